#!/usr/bin/env python3
"""Micro-bench the fused memory-bound ops at the GPT-6.7B bench shapes."""

import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch


def t(fn, n=30, warm=5):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / n * 1e6  # us


def main():
    from paddlefleetx_amd.ops import hip_ext
    ext = hip_ext()
    dev = "cuda"
    bf = torch.bfloat16

    N, H = 8192, 4096
    x = torch.randn(N, H, device=dev, dtype=bf)
    r = torch.randn(N, H, device=dev, dtype=bf)
    w = torch.randn(H, device=dev, dtype=bf)
    b = torch.randn(H, device=dev, dtype=bf)
    dy = torch.randn(N, H, device=dev, dtype=bf)

    y, mean, rstd = ext.layernorm_fwd(x, w, b, 1e-5)
    print(f"ln_fwd        [8192,4096]: {t(lambda: ext.layernorm_fwd(x, w, b, 1e-5)):8.1f} us")
    print(f"ln_fwd_res    [8192,4096]: {t(lambda: ext.layernorm_fwd_residual(x, r, w, b, 1e-5)):8.1f} us")
    print(f"ln_bwd        [8192,4096]: {t(lambda: ext.layernorm_bwd(dy, x, w, mean, rstd)):8.1f} us")
    print(f"ln_bwd_res    [8192,4096]: {t(lambda: ext.layernorm_bwd_residual(dy, x, w, mean, rstd, r)):8.1f} us")
    print(f"torch add     [8192,4096]: {t(lambda: x + r):8.1f} us")

    F, FH = 8192, 16384
    xf = torch.randn(F, FH, device=dev, dtype=bf)
    bb = torch.randn(FH, device=dev, dtype=bf)
    dyf = torch.randn(F, FH, device=dev, dtype=bf)
    print(f"bias_gelu_fwd [8192,16384]: {t(lambda: ext.bias_gelu_fwd(xf, bb)):8.1f} us")
    print(f"bias_gelu_bwd [8192,16384]: {t(lambda: ext.bias_gelu_bwd(dyf, xf, bb)):8.1f} us")
    print(f"colsum        [8192,16384]: {t(lambda: ext.colsum(dyf)):8.1f} us")
    print(f"torch f32sum0 [8192,16384]: {t(lambda: dyf.float().sum(0)):8.1f} us")

    # wgrad GEMM: beta=1 addmm_ into a bf16 grad view vs mm + add
    for (o, i) in [(12288, 4096), (4096, 4096), (16384, 4096), (4096, 16384)]:
        g = torch.zeros(o, i, device=dev, dtype=bf)
        dyw = torch.randn(N, o, device=dev, dtype=bf)
        xw = torch.randn(N, i, device=dev, dtype=bf)
        t_fused = t(lambda: g.addmm_(dyw.t(), xw))
        t_sep = t(lambda: g.add_(dyw.t().mm(xw)))
        t_mm = t(lambda: dyw.t().mm(xw))
        print(f"wgrad [{o:5d},{i:5d}]: addmm_ {t_fused:8.1f} us | mm+add {t_sep:8.1f} us | mm alone {t_mm:8.1f} us")

    # MoE dispatch: fused HIP vs torch argsort+index_select
    T, D2, E, K = 8192, 4096, 64, 2
    xm = torch.randn(T, D2, device=dev, dtype=bf)
    ids = torch.randint(0, E, (T * K,), device=dev)
    def torch_disp():
        sel = (ids >= 0).nonzero(as_tuple=True)[0]
        active = ids[sel]
        perm = torch.argsort(active, stable=True)
        sel_sorted = sel[perm]
        torch.bincount(active[perm], minlength=E)
        return xm.index_select(0, sel_sorted // K)
    print(f"moe_dispatch fused  (T8192 D4096 E64 k2): {t(lambda: ext.moe_dispatch(xm, ids, E, K)):8.1f} us")
    print(f"moe_dispatch torch  (argsort+gather):     {t(torch_disp):8.1f} us")

    # delta kernel (inside attn bwd) via attn micro shapes
    B, Hh, S, D = 8, 32, 1024, 128
    q = torch.randn(B, Hh, S, D, device=dev, dtype=bf)
    k2 = torch.randn_like(q); v2 = torch.randn_like(q); do = torch.randn_like(q)
    o2, lse = ext.attn_fwd(q, k2, v2, True, D ** -0.5)
    print(f"attn_bwd (B8 H32 S1024 D128): {t(lambda: ext.attn_bwd(do, q, k2, v2, o2, lse, True, D ** -0.5), n=20):8.1f} us")


if __name__ == "__main__":
    main()
