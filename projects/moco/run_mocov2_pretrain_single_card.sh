#!/usr/bin/env bash
# reference projects/moco/run_mocov2_pretrain_single_card.sh
cd "$(dirname "$0")/../.."
python tools/train.py -c paddlefleetx_amd/configs/vis/moco/mocov2_pretrain_single_card.yaml "$@"
