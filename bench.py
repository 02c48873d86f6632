#!/usr/bin/env python3
"""Flagship benchmark: GPT-3 6.7B pretrain step (BASELINE.json north star).

Single node, one rank per GPU over RCCL:
  python bench.py --gpus 1 --steps 10 --warmup 3
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N ...

Parallelism by N (weak scaling toward the 8-GPU DP2xTP2xPP2 target):
  1 -> single | 2 -> DP2 | 4 -> DP4 | 8 -> DP2xTP2xPP2

Rank 0 prints ONE JSON line with the whole-job tokens/s.
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch
import torch.distributed as dist


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=None)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--model", type=str, default="GPT-6.7B",
                   choices=["GPT-TEST", "GPT-345M", "GPT-1.3B", "GPT-6.7B",
                            "GPT-13B"])
    p.add_argument("--micro-batch", type=int, default=None)
    p.add_argument("--acc-steps", type=int, default=None)
    p.add_argument("--seq-len", type=int, default=1024)
    p.add_argument("--hip-graph", action="store_true",
                   help="capture the micro-step in a hipGraph (launch-"
                        "bound small-model configs)")
    p.add_argument("--dropout", type=float, default=0.0,
                   help="hidden+attention dropout (the reference's "
                        "pretrain configs run 0.1; in-kernel Philox "
                        "attention dropout on the flash path)")
    return p.parse_args()


MODELS = {
    # tiny config for CPU rehearsal of the multi-rank code paths
    # (tests/test_bench_rehearsal_cpu.py); never a reportable number
    "GPT-TEST": dict(hidden_size=64, num_layers=4, num_attention_heads=4),
    "GPT-345M": dict(hidden_size=1024, num_layers=24, num_attention_heads=16),
    "GPT-1.3B": dict(hidden_size=2048, num_layers=24, num_attention_heads=16),
    "GPT-6.7B": dict(hidden_size=4096, num_layers=32, num_attention_heads=32),
    # fits in ONE MI355X's 288 GB with full Adam state resident
    # (params 26 + master 52 + m/v 104 + grads 26 GB) — no sharding
    "GPT-13B": dict(hidden_size=5120, num_layers=40, num_attention_heads=40),
}


def topology_for(n):
    """(dp, tp, pp). N=8 is the named DP2xTP2xPP2 config; intermediate N
    scale out over DP (weak scaling: per-GPU work fixed)."""
    return {1: (1, 1, 1), 2: (2, 1, 1), 4: (4, 1, 1), 8: (2, 2, 2)}.get(
        n, (n, 1, 1))  # fallback: pure DP


def main():
    args = parse_args()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    n = args.gpus or world
    assert world in (1, n), f"launched with WORLD_SIZE={world} but --gpus={n}"
    dp, tp, pp = topology_for(n)

    from paddlefleetx_amd.utils.config import get_config
    from paddlefleetx_amd.parallel.env import init_dist_env, get_hcg
    from paddlefleetx_amd.models import build_module
    from paddlefleetx_amd.core import EagerEngine

    # defaults from the measured 1-GPU sweep (profiles/): micro=16 puts
    # the hipBLASLt GEMMs at M=16384 (same-box A/B vs micro8: 26234 vs
    # 25904 tok/s; micro32 gains nothing more and doubles activation
    # memory — gpurun_out/r2_m16_ab.txt, r2_m32.txt). Under pipeline
    # parallel trade GEMM width for more micro-batches
    # (1F1B bubble = (pp-1)/(acc+pp-1)).
    if pp > 1:
        micro = args.micro_batch or 4
        acc = args.acc_steps or 8
    elif args.model == "GPT-13B":
        # measured sweet spot (profiles/r02_13b_single_gpu.txt): micro4
        # saturates the GEMMs; micro16 would overrun 288 GB
        micro = args.micro_batch or 4
        acc = args.acc_steps or 4
    else:
        micro = args.micro_batch or 16
        acc = args.acc_steps or 2
    local_bs = micro * acc
    seq = args.seq_len
    cfg_path = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                            "paddlefleetx_amd/configs/nlp/gpt/pretrain_gpt_base.yaml")
    shape = MODELS[args.model]
    overrides = [
        f"Model.hidden_size={shape['hidden_size']}",
        f"Model.num_layers={shape['num_layers']}",
        f"Model.num_attention_heads={shape['num_attention_heads']}",
        f"Model.max_position_embeddings={seq}",
        f"Model.hidden_dropout_prob={args.dropout}",
        f"Model.attention_probs_dropout_prob={args.dropout}",
        f"Global.micro_batch_size={micro}",
        f"Global.local_batch_size={local_bs}",
        "Global.eval_freq=", "Global.save_steps=",
        f"Distributed.dp_degree={dp}", f"Distributed.mp_degree={tp}",
        f"Distributed.pp_degree={pp}",
    ]
    if args.hip_graph:
        overrides.append("Engine.hip_graph=True")
    cfg = get_config(cfg_path, overrides=overrides)
    hcg = init_dist_env(cfg)
    rank = hcg.global_rank
    device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")

    module = build_module(cfg)
    engine = EagerEngine(cfg, module)

    vocab = cfg["Model"]["padded_vocab_size"]
    torch.manual_seed(1234 + hcg.get_data_world_rank())
    batch = (torch.randint(0, vocab, (local_bs, seq), device=device),
             torch.arange(seq, device=device).unsqueeze(0).repeat(local_bs, 1),
             torch.randint(0, vocab, (local_bs, seq), device=device),
             torch.ones(local_bs, seq, device=device))

    def barrier_sync():
        if dist.is_initialized():
            dist.barrier()
        if device.type == "cuda":
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        engine._fit_impl(batch)
    barrier_sync()
    t0 = time.time()
    loss = None
    for _ in range(args.steps):
        loss = engine._fit_impl(batch)
    barrier_sync()
    elapsed = time.time() - t0

    # max over ranks
    if dist.is_initialized():
        t = torch.tensor(elapsed, device=device if device.type == "cuda" else None)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t)

    global_batch = cfg["Global"]["global_batch_size"]
    tokens_per_step = global_batch * seq
    toks_per_s = tokens_per_step * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        metric = "tokens/sec/GPU GPT-3-6.7B pretrain, DP2×TP2×PP2 at 1/2/4/8 MI355X"
        if args.model != "GPT-6.7B":
            metric = metric.replace("GPT-3-6.7B", args.model)
        out = {
            "metric": metric,
            "value": round(toks_per_s, 1),
            "unit": "tokens/s",
            "n_gpus": n,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "dropout": args.dropout,
            "config": {
                "model": args.model, "global_batch": global_batch,
                "seq_len": seq, "micro_batch": micro,
                "parallelism": f"dp{dp}tp{tp}pp{pp}",
                "loss": float(loss) if loss is not None else None,
            },
        }
        print(json.dumps(out))


if __name__ == "__main__":
    main()
