#!/usr/bin/env bash
# reference projects/gpt/pretrain_gpt_6.7B_sharding16.sh
cd "$(dirname "$0")/../.."
python tools/train.py -c paddlefleetx_amd/configs/nlp/gpt/pretrain_gpt_6.7B_sharding16.yaml "$@"
