#!/usr/bin/env bash
# reference projects/imagen/run_text2im_397M_64x64_single_card.sh
cd "$(dirname "$0")/../.."
python tools/train.py -c paddlefleetx_amd/configs/mm/imagen/text2im_397M_64x64_single_card.yaml "$@"
