#!/usr/bin/env bash
# reference projects/gpt/auto_gpt_1.3B_single_card.sh
cd "$(dirname "$0")/../.."
python tools/auto.py -c paddlefleetx_amd/configs/nlp/gpt/pretrain_gpt_1.3B_single_card.yaml "$@"
