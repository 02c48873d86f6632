#!/usr/bin/env bash
# auto-generated TIPC-style topology benchmark (see benchmark_common/run_benchmark.sh)
DIR=$(dirname "$0")
bash "$DIR/../../benchmark_common/run_benchmark.sh" \
  paddlefleetx_amd/configs/nlp/ernie/pretrain_ernie_base_3D.yaml 8 \
  "Distributed.dp_degree=8 Global.global_batch_size=32 Global.local_batch_size=4 Global.micro_batch_size=4 Model.num_hidden_layers=4"
