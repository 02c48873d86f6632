#!/usr/bin/env bash
# auto-generated TIPC-style topology benchmark (see benchmark_common/run_benchmark.sh)
DIR=$(dirname "$0")
bash "$DIR/../../benchmark_common/run_benchmark.sh" \
  paddlefleetx_amd/configs/vis/vit/ViT_base_patch16_224_pretrain_dp8.yaml 8 \
  ""
