#!/usr/bin/env bash
# hybrid topology + interleaved virtual pipeline stages
DIR=$(dirname "$0")
bash "$DIR/../../benchmark_common/run_benchmark.sh" \
  paddlefleetx_amd/configs/nlp/gpt/pretrain_gpt_base.yaml 8 \
  "Distributed.dp_degree=2 Distributed.mp_degree=2 Distributed.pp_degree=2 Distributed.pipeline.virtual_pp_degree=2 Global.global_batch_size=16 Global.local_batch_size=8 Global.micro_batch_size=2 Engine.accumulate_steps=4 Model.num_layers=4 Model.hidden_dropout_prob=0.0 Model.attention_probs_dropout_prob=0.0"
