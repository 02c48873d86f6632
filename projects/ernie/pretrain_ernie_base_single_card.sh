#!/usr/bin/env bash
# reference projects/ernie/pretrain_ernie_base_3D.sh (single-card fold)
cd "$(dirname "$0")/../.."
python tools/train.py -c paddlefleetx_amd/configs/nlp/ernie/pretrain_ernie_base_3D.yaml \
  -o Distributed.dp_degree=1 -o Distributed.mp_degree=1 -o Distributed.pp_degree=1 "$@"
