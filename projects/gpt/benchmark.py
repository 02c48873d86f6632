#!/usr/bin/env python3
"""Inference latency benchmark over an exported model (reference
projects/gpt/benchmark.py:44-85: warm the predictor, then time `--iter`
decode runs and print ms/run).

    python projects/gpt/benchmark.py --model-dir ./exported_model \
        [--seq-len 128] [--iter 20] [--batch 1]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import torch

from paddlefleetx_amd.core.inference_engine import InferenceEngine


def parse_args():
    p = argparse.ArgumentParser("gpt-inference-benchmark")
    p.add_argument("--model-dir", default="./exported_model")
    p.add_argument("--mp-degree", type=int, default=1)
    p.add_argument("--seq-len", type=int, default=128)
    p.add_argument("--batch", type=int, default=1)
    p.add_argument("--iter", type=int, default=20)
    p.add_argument("--max-dec-len", type=int, default=None,
                   help="override the exported Generation max_dec_len")
    return p.parse_args()


def main():
    args = parse_args()
    gen_cfg = {"max_dec_len": args.max_dec_len} if args.max_dec_len else None
    engine = InferenceEngine(args.model_dir, mp_degree=args.mp_degree,
                             generation_cfg=gen_cfg)
    ids = [[100] * args.seq_len for _ in range(args.batch)]

    for _ in range(3):  # warmup (reference runs 10; decode is costlier)
        out = engine.predict(ids)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    start = time.perf_counter()
    for _ in range(args.iter):
        out = engine.predict(ids)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    end = time.perf_counter()
    ms = 1000 * (end - start) / args.iter
    new_tokens = max(1, out.shape[-1] if hasattr(out, "shape") else 1)
    print(f"batch {args.iter} run time: {ms:.2f}ms "
          f"({ms / new_tokens:.2f} ms/token, batch {args.batch}, "
          f"prompt {args.seq_len})")


if __name__ == "__main__":
    main()
