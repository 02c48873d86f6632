#!/usr/bin/env bash
# Protein folding (Evoformer + IPA structure module) on synthetic data.
set -e
cd "$(dirname "$0")/../.."
python tools/train.py -c paddlefleetx_amd/configs/folding/pretrain_folding_tiny.yaml "$@"
