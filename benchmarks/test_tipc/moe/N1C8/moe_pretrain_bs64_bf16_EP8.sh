#!/usr/bin/env bash
# auto-generated TIPC-style topology benchmark (see benchmark_common/run_benchmark.sh)
DIR=$(dirname "$0")
bash "$DIR/../../benchmark_common/run_benchmark.sh" \
  paddlefleetx_amd/configs/nlp/moe/pretrain_moe_345M_64experts_ep8.yaml 8 \
  "Model.num_layers=4"
