#!/usr/bin/env bash
# reference projects/gpt/auto_export_gpt_345M_single_card.sh
cd "$(dirname "$0")/../.."
python tools/auto_export.py -c paddlefleetx_amd/configs/nlp/gpt/pretrain_gpt_345M_single_card.yaml "$@"
