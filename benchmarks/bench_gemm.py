#!/usr/bin/env python3
"""hipBLASLt GEMM rates for the GPT-6.7B hot shapes (micro=8, seq=1024)."""
import sys, time
import torch

def rate(fn, flops, n=20):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.time()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    dt = (time.time() - t0) / n
    return flops / dt / 1e12, dt * 1e3

M, H = 8192, 4096
shapes = [
    ("qkv fwd   [8192,4096]x[4096,12288]", (M, H), (3*H, H)),
    ("out fwd   [8192,4096]x[4096,4096]",  (M, H), (H, H)),
    ("up fwd    [8192,4096]x[4096,16384]", (M, H), (4*H, H)),
    ("down fwd  [8192,16384]x[16384,4096]",(M, 4*H), (H, 4*H)),
    ("logits    [8192,4096]x[4096,50304]", (M, H), (50304, H)),
]
total_t = 0.0
for name, xs, ws in shapes:
    x = torch.randn(*xs, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(*ws, device="cuda", dtype=torch.bfloat16)
    fl = 2 * xs[0] * xs[1] * ws[0]
    tf, ms = rate(lambda: torch.nn.functional.linear(x, w), fl)
    # dgrad: dy [M, N] x W [N, K]
    dy = torch.randn(xs[0], ws[0], device="cuda", dtype=torch.bfloat16)
    tf_d, ms_d = rate(lambda: dy @ w, fl)
    # wgrad: dy^T [N, M] x x [M, K]
    tf_w, ms_w = rate(lambda: dy.t() @ x, fl)
    total_t += ms + ms_d + ms_w
    print(f"{name}: fwd {tf:6.0f} TF/s ({ms:.2f}ms) | dgrad {tf_d:6.0f} ({ms_d:.2f}ms) | wgrad {tf_w:6.0f} ({ms_w:.2f}ms)")
print(f"sum per micro-step GEMM time: {total_t:.1f} ms")
