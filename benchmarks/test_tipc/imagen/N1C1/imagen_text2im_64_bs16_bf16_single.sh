#!/usr/bin/env bash
DIR=$(dirname "$0")
bash "$DIR/../../benchmark_common/run_benchmark.sh" \
  paddlefleetx_amd/configs/mm/imagen/text2im_397M_64x64_single_card.yaml 1 \
  "" 20
