#!/usr/bin/env bash
# GPT-13B on ONE MI355X: full Adam state resident in 288 GB HBM3E, no
# sharding (the reference runs 13B only sharded across nodes)
cd "$(dirname "$0")/../.."
python tools/train.py -c paddlefleetx_amd/configs/nlp/gpt/pretrain_gpt_13B_single_card.yaml "$@"
