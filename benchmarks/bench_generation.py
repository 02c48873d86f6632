#!/usr/bin/env python3
"""KV-cache decode + top-p sampling benchmark (ms/token over batch 1-16).

Counterpart of the reference's inference latency harness
(ppfleetx projects/gpt/benchmark.py:44-85: batch sweep, ms/batch print).

    python benchmarks/bench_generation.py [--model GPT-1.3B] \
        [--prompt-len 128] [--gen-len 64]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

SHAPES = {
    "GPT-345M": dict(hidden_size=1024, num_layers=24, num_attention_heads=16),
    "GPT-1.3B": dict(hidden_size=2048, num_layers=24, num_attention_heads=16),
    "GPT-6.7B": dict(hidden_size=4096, num_layers=32, num_attention_heads=32),
}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="GPT-1.3B", choices=sorted(SHAPES))
    p.add_argument("--prompt-len", type=int, default=128)
    p.add_argument("--gen-len", type=int, default=64)
    p.add_argument("--batches", type=str, default="1,2,4,8,16")
    p.add_argument("--top-p", type=float, default=0.9)
    args = p.parse_args()

    from paddlefleetx_amd.parallel.env import set_hcg
    from paddlefleetx_amd.parallel.topology import HybridTopology
    import torch.distributed as dist
    if not dist.is_initialized():
        set_hcg(HybridTopology())
    from paddlefleetx_amd.models.gpt.generation import GPTForGeneration
    from paddlefleetx_amd.models.gpt.model import GPTModel

    dev = "cuda" if torch.cuda.is_available() else "cpu"
    dtype = torch.bfloat16 if dev == "cuda" else torch.float32
    shape = SHAPES[args.model]
    vocab = 50304
    max_len = args.prompt_len + args.gen_len + 8
    torch.manual_seed(0)
    with torch.device(dev):
        gpt = GPTModel(vocab_size=vocab, max_position_embeddings=max_len,
                       dtype=dtype, **shape)
        model = GPTForGeneration(gpt, {
            "max_dec_len": args.gen_len,
            "decoding_strategy": "sampling",
            "top_p": args.top_p,
            "use_topp_sampling": True,
            "eos_token_id": vocab - 1,
        })
    model.eval()

    print(f"# decode bench {args.model} prompt={args.prompt_len} "
          f"gen={args.gen_len} top_p={args.top_p} ({dev})")
    for bs in [int(b) for b in args.batches.split(",")]:
        ids = torch.randint(0, vocab - 2, (bs, args.prompt_len), device=dev)
        with torch.no_grad():
            model(ids)  # warmup (compile caches, allocator)
            if dev == "cuda":
                torch.cuda.synchronize()
            t0 = time.time()
            reps = 3
            for _ in range(reps):
                out = model(ids)
            if dev == "cuda":
                torch.cuda.synchronize()
        dt = (time.time() - t0) / reps
        # GPTForGeneration returns only the GENERATED ids
        ntok = out.shape[1] if out.shape[1] <= args.gen_len \
            else out.shape[1] - args.prompt_len
        ms_tok = dt * 1000.0 / max(1, ntok)
        print(f"bs={bs:3d}: {dt*1000:8.1f} ms/gen ({ntok} new tok) = "
              f"{ms_tok:7.2f} ms/token, {bs*ntok/dt:8.0f} tok/s")


if __name__ == "__main__":
    main()
