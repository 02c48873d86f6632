#!/usr/bin/env python3
"""Attention kernel micro-benchmark (the bench.py hot shape).

    python benchmarks/bench_attn.py [--B 8] [--H 32] [--S 1024] [--D 128]
Prints fwd/bwd ms and effective TFLOP/s (causal-adjusted).
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--B", type=int, default=8)
    p.add_argument("--H", type=int, default=32)
    p.add_argument("--S", type=int, default=1024)
    p.add_argument("--D", type=int, default=128)
    p.add_argument("--iters", type=int, default=20)
    p.add_argument("--causal", type=int, default=1)
    args = p.parse_args()

    from paddlefleetx_amd.ops import hip_ext
    ext = hip_ext()
    B, H, S, D = args.B, args.H, args.S, args.D
    q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    do = torch.randn_like(q)
    scale = D ** -0.5
    causal = bool(args.causal)

    def timeit(fn, n):
        for _ in range(5):
            fn()
        torch.cuda.synchronize()
        t0 = time.time()
        for _ in range(n):
            fn()
        torch.cuda.synchronize()
        return (time.time() - t0) / n

    o, lse = ext.attn_fwd(q, k, v, causal, scale)
    t_fwd = timeit(lambda: ext.attn_fwd(q, k, v, causal, scale), args.iters)
    t_bwd = timeit(lambda: ext.attn_bwd(do, q, k, v, o, lse, causal, scale),
                   args.iters)

    frac = 0.5 + 0.5 / (S / 64) if causal else 1.0  # causal-valid fraction
    f_fwd = 4 * B * H * S * S * D * frac
    f_bwd = f_fwd * 2.5
    print(f"attn B{B} H{H} S{S} D{D} causal={causal}: "
          f"fwd {t_fwd*1e3:.3f} ms = {f_fwd/t_fwd/1e12:.1f} TF/s eff | "
          f"bwd {t_bwd*1e3:.3f} ms = {f_bwd/t_bwd/1e12:.1f} TF/s eff")


if __name__ == "__main__":
    main()
