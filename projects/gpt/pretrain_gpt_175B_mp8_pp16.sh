#!/usr/bin/env bash
# reference projects/gpt/pretrain_gpt_175B_mp8_pp16.sh
cd "$(dirname "$0")/../.."
python tools/train.py -c paddlefleetx_amd/configs/nlp/gpt/pretrain_gpt_175B_mp8_pp16.yaml "$@"
