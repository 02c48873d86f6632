#!/usr/bin/env bash
# GPT-345M DP8 (reference benchmarks/test_tipc/gpt/.../data_parallel N1C8)
DIR=$(dirname "$0")
bash "$DIR/../../benchmark_common/run_benchmark.sh" \
  paddlefleetx_amd/configs/nlp/gpt/pretrain_gpt_345M_single_card.yaml 8 \
  "Distributed.dp_degree=8 Global.global_batch_size=64 Global.local_batch_size=8 Global.micro_batch_size=8 Model.num_layers=4 Model.hidden_dropout_prob=0.0 Model.attention_probs_dropout_prob=0.0"
