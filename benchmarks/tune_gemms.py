#!/usr/bin/env python3
"""Offline TunableOp GEMM tuning for every hot shape of the bench
topologies (N=1: tp1 micro8; N=8: tp2 pp2 micro4), run on ONE GPU.

Each Linear(M, in, out) triggers the same hipBLASLt GEMMs the model
will issue (fwd tn / dgrad nn / wgrad nt), so tuning here covers the
distributed runs without needing multiple GPUs.

  PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1 \
  PYTORCH_TUNABLEOP_FILENAME=gpurun_out/tune.csv \
  python benchmarks/tune_gemms.py
then merge gpurun_out/tune0.csv rows into configs/tunableop_gfx950.csv.
"""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch
import torch.nn.functional as F

H = 4096          # GPT-6.7B hidden
V = 50304         # padded vocab
SEQ = 1024


def shapes():
    out = set()
    for tp, micro in ((1, 8), (1, 16), (2, 4), (2, 8)):
        m = micro * SEQ
        for i, o in (
                (H, 3 * H // tp),      # col QKV
                (H // tp, H),          # row out-proj
                (H, 4 * H // tp),      # col FC1
                (4 * H // tp, H),      # row FC2
                (H, V // tp),          # tied logits (parallel_matmul)
        ):
            out.add((m, i, o))
    return sorted(out)


def main():
    assert torch.cuda.is_available()
    torch.backends.cuda.matmul.allow_tf32 = False
    dev = torch.device("cuda")
    for (m, i, o) in shapes():
        x = torch.randn(m, i, device=dev, dtype=torch.bfloat16,
                        requires_grad=True)
        w = torch.randn(o, i, device=dev, dtype=torch.bfloat16,
                        requires_grad=True)
        y = F.linear(x, w)          # tn fwd
        y.backward(torch.randn_like(y))  # nn dgrad + nt wgrad
        torch.cuda.synchronize()
        print(f"tuned M={m} in={i} out={o}")
    print("done")


if __name__ == "__main__":
    main()
