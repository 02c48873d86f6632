#!/usr/bin/env bash
# context parallel with the ring-attention backend
DIR=$(dirname "$0")
bash "$DIR/../../benchmark_common/run_benchmark.sh" \
  paddlefleetx_amd/configs/nlp/gpt/pretrain_gpt_base.yaml 8 \
  "Distributed.cp_degree=8 Model.cp_backend=ring Global.global_batch_size=16 Global.local_batch_size=16 Global.micro_batch_size=4 Model.num_layers=4 Model.hidden_dropout_prob=0.0 Model.attention_probs_dropout_prob=0.0"
