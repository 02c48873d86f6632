#!/usr/bin/env bash
# GPT DP2-MP2-PP2 hybrid (reference gpt_bs16_fp16_DP2-MP2-PP2.sh)
DIR=$(dirname "$0")
bash "$DIR/../../benchmark_common/run_benchmark.sh" \
  paddlefleetx_amd/configs/nlp/gpt/pretrain_gpt_base.yaml 8 \
  "Distributed.dp_degree=2 Distributed.mp_degree=2 Distributed.pp_degree=2 Global.global_batch_size=16 Global.local_batch_size=8 Global.micro_batch_size=2 Model.num_layers=4 Model.hidden_dropout_prob=0.0 Model.attention_probs_dropout_prob=0.0"
