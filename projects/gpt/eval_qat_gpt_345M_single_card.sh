#!/usr/bin/env bash
# reference projects/gpt/eval_qat_gpt_345M_single_card.sh
cd "$(dirname "$0")/../.."
python tools/eval.py -c paddlefleetx_amd/configs/nlp/gpt/qat_gpt_345M_single_card.yaml "$@"
