#!/usr/bin/env bash
# ERNIE-MoE 64-expert EP8 over the full xGMI mesh (driver config #4)
cd "$(dirname "$0")/../.."
python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 \
  tools/train.py -c paddlefleetx_amd/configs/nlp/ernie/pretrain_ernie_moe_64experts_ep8.yaml "$@"
