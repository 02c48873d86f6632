#!/usr/bin/env bash
# CE-style convergence run (reference CE_ prefix scripts parse loss:)
DIR=$(dirname "$0")
bash "$DIR/../../benchmark_common/run_benchmark.sh" \
  paddlefleetx_amd/configs/nlp/gpt/finetune_gpt_345M_single_card_glue.yaml 1 \
  "Model.num_layers=4 Model.task=mrpc" 20
