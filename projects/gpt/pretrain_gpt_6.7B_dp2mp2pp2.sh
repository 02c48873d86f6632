#!/usr/bin/env bash
# the BASELINE.json north-star topology: 6.7B bf16 DP2xTP2xPP2 on 8xMI355X
cd "$(dirname "$0")/../.."
python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 \
  tools/train.py -c paddlefleetx_amd/configs/nlp/gpt/pretrain_gpt_6.7B_dp2mp2pp2.yaml "$@"
