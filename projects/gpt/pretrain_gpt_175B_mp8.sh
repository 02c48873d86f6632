#!/usr/bin/env bash
# 175B TP8 sharding-stage3 sizing on one 8xMI355X node (288 GB HBM/GPU)
cd "$(dirname "$0")/../.."
python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 \
  tools/train.py -c paddlefleetx_amd/configs/nlp/gpt/pretrain_gpt_175B_mp8_pp16.yaml "$@"
