#!/usr/bin/env bash
# ViT-Huge/14 DP8 bf16 (driver config #5)
cd "$(dirname "$0")/../.."
python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 \
  tools/train.py -c paddlefleetx_amd/configs/vis/vit/ViT_huge_patch14_224_pretrain_dp8.yaml "$@"
