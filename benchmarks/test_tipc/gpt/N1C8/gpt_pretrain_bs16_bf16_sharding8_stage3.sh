#!/usr/bin/env bash
DIR=$(dirname "$0")
bash "$DIR/../../benchmark_common/run_benchmark.sh" \
  paddlefleetx_amd/configs/nlp/gpt/pretrain_gpt_345M_single_card.yaml 8 \
  "Model.num_layers=4 Distributed.sharding.sharding_degree=8 Distributed.sharding.sharding_stage=3 Global.micro_batch_size=2 Global.local_batch_size=2 Global.global_batch_size=16"
