#!/usr/bin/env bash
# auto-generated TIPC-style topology benchmark (see benchmark_common/run_benchmark.sh)
DIR=$(dirname "$0")
bash "$DIR/../../benchmark_common/run_benchmark.sh" \
  paddlefleetx_amd/configs/nlp/gpt/pretrain_gpt_345M_single_card.yaml 1 \
  "Model.num_layers=4 Model.hidden_dropout_prob=0.0 Model.attention_probs_dropout_prob=0.0"
