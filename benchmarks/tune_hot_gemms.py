#!/usr/bin/env python3
"""Long-budget TunableOp tuning of ONLY the GPT-6.7B hot GEMM shapes.

Run on a GPU box with:
  PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1 \
  PYTORCH_TUNABLEOP_FILENAME=out.csv \
  PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS=200 \
  python benchmarks/tune_hot_gemms.py
Then merge out.csv into the shipped table by BEST MEASURED TIME per key
(tools-side; see gpurun_out notes). The quick online retune of the whole
bench regressed e2e (ROADMAP) — per-shape long tuning + best-of merge is
the safe form.
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch
import torch.nn.functional as F

_ap = argparse.ArgumentParser()
_ap.add_argument("--m", type=int, default=8192,
                 help="GEMM M = micro_batch x seq (8192=micro8, "
                      "16384=micro16, 32768=micro32)")
M = _ap.parse_args().m
H = 4096
SHAPES = [  # (out_features, in_features) of the 6.7B layer GEMMs
    (3 * H, H),        # qkv
    (H, H),            # attention out
    (4 * H, H),        # ffn up
    (H, 4 * H),        # ffn down
    (50304, H),        # logits (tied embedding)
]


def main():
    assert torch.cuda.is_available()
    dev = "cuda"
    for out_f, in_f in SHAPES:
        x = torch.randn(M, in_f, device=dev, dtype=torch.bfloat16,
                        requires_grad=True)
        w = torch.randn(out_f, in_f, device=dev, dtype=torch.bfloat16,
                        requires_grad=True)
        b = torch.randn(out_f, device=dev, dtype=torch.bfloat16)
        for _ in range(3):
            y = F.linear(x, w, b)             # fwd (GemmAndBias)
            y.backward(torch.randn_like(y))   # dgrad + wgrad
            x.grad = None
            w.grad = None
        # the fused-wgrad form (beta=1 addmm into a grad view)
        g = torch.zeros(out_f, in_f, device=dev, dtype=torch.bfloat16)
        dy = torch.randn(M, out_f, device=dev, dtype=torch.bfloat16)
        for _ in range(3):
            g.addmm_(dy.t(), x.detach())
        torch.cuda.synchronize()
        print(f"tuned shapes for ({out_f}, {in_f})")


if __name__ == "__main__":
    main()
