#!/usr/bin/env bash
# reference projects/gpt/pretrain_gpt_1.3B_dp8.sh — one 8xMI355X node
cd "$(dirname "$0")/../.."
python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 \
  tools/train.py -c paddlefleetx_amd/configs/nlp/gpt/pretrain_gpt_1.3B_dp8.yaml "$@"
