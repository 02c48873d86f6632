#!/usr/bin/env bash
DIR=$(dirname "$0")
bash "$DIR/../../benchmark_common/run_benchmark.sh" \
  paddlefleetx_amd/configs/nlp/ernie/pretrain_ernie_base_3D.yaml 1 \
  "Distributed.dp_degree=1 Distributed.mp_degree=1 Distributed.pp_degree=1 Model.num_hidden_layers=4"
