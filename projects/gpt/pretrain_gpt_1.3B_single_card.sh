#!/usr/bin/env bash
# reference projects/gpt/pretrain_gpt_1.3B_single_card.sh
cd "$(dirname "$0")/../.."
python tools/train.py -c paddlefleetx_amd/configs/nlp/gpt/pretrain_gpt_1.3B_single_card.yaml "$@"
