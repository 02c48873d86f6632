#!/usr/bin/env bash
# reference projects/gpt/auto_gpt_1.3B_dp8_tuning.sh (accumulate-steps tuning pass)
cd "$(dirname "$0")/../.."
python tools/auto.py --tune -c paddlefleetx_amd/configs/nlp/gpt/pretrain_gpt_1.3B_dp8.yaml "$@"
