#!/usr/bin/env python3
"""Per-family throughput probes for the BASELINE config list: ERNIE-MoE
(config #4 at EP1 on one GPU; EP8 is the driver's 8-GPU tier) and
ViT-Huge/14 (config #5, per-GPU fold of DP8).

    python benchmarks/bench_family.py --family moe|vit [--steps 8]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--family", choices=["moe", "vit"], required=True)
    p.add_argument("--steps", type=int, default=8)
    p.add_argument("--warmup", type=int, default=2)
    args = p.parse_args()

    from paddlefleetx_amd.parallel.env import set_hcg, set_seed
    from paddlefleetx_amd.parallel.topology import HybridTopology
    import torch.distributed as dist
    if not dist.is_initialized():
        set_hcg(HybridTopology())
    set_seed(1234)
    from paddlefleetx_amd.utils.config import get_config
    from paddlefleetx_amd.models import build_module
    from paddlefleetx_amd.core import EagerEngine

    repo = os.path.join(os.path.dirname(__file__), "..")
    dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")

    if args.family == "moe":
        # ERNIE-MoE-style GPT-MoE: 64 experts on ONE GPU (EP1; the a2a
        # degenerates, the fused dispatch + gshard gate run for real)
        cfg = get_config(os.path.join(
            repo, "paddlefleetx_amd/configs/nlp/moe/"
            "pretrain_moe_345M_64experts_ep8.yaml"),
            overrides=["Distributed.expert_parallel_degree=1",
                       "Distributed.dp_degree=1",
                       "Distributed.world_size=1",
                       "Global.micro_batch_size=8",
                       "Global.local_batch_size=8"])
        module = build_module(cfg)
        engine = EagerEngine(cfg, module)
        seq = int(cfg["Model"]["max_position_embeddings"])
        bs = int(cfg["Global"]["local_batch_size"])
        vocab = cfg["Model"]["padded_vocab_size"]
        batch = (torch.randint(0, vocab, (bs, seq), device=dev),
                 torch.arange(seq, device=dev).repeat(bs, 1),
                 torch.randint(0, vocab, (bs, seq), device=dev),
                 torch.ones(bs, seq, device=dev))
        unit, per_step = "tokens/s", bs * seq
    else:
        cfg = get_config(os.path.join(
            repo, "paddlefleetx_amd/configs/vis/vit/"
            "ViT_huge_patch14_224_pretrain_dp8.yaml"),
            overrides=["Distributed.dp_degree=1",
                       "Distributed.world_size=1",
                       "Global.micro_batch_size=32",
                       "Global.local_batch_size=32",
                       "Global.global_batch_size=32"])
        module = build_module(cfg)
        engine = EagerEngine(cfg, module)
        bs = int(cfg["Global"]["local_batch_size"])
        mdtype = next(module.model.parameters()).dtype
        batch = (torch.randn(bs, 3, 224, 224, device=dev, dtype=mdtype),
                 torch.randint(0, 1000, (bs,), device=dev))
        unit, per_step = "imgs/s", bs

    for _ in range(args.warmup):
        engine._fit_impl(batch)
    if dev.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.time()
    loss = None
    for _ in range(args.steps):
        loss = engine._fit_impl(batch)
    if dev.type == "cuda":
        torch.cuda.synchronize()
    dt = (time.time() - t0) / args.steps
    print(f"{args.family}: {per_step / dt:.1f} {unit} "
          f"({dt * 1000:.1f} ms/step, loss {float(loss):.3f})")


if __name__ == "__main__":
    main()
