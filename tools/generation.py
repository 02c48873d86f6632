#!/usr/bin/env python3
"""Text generation entry (reference tasks/gpt/generation.py:35-63).

    python tools/generation.py -c configs/.../generation_gpt_345M_single_card.yaml \
        -o Generation.top_p=0.9 --input-ids 464,3290,318
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from paddlefleetx_amd.models import build_module
from paddlefleetx_amd.parallel.env import init_dist_env
from paddlefleetx_amd.utils.config import get_config
from paddlefleetx_amd.utils.log import logger


def parse_args():
    p = argparse.ArgumentParser("generation")
    p.add_argument("-c", "--config", required=True)
    p.add_argument("-o", "--override", action="append", default=[])
    p.add_argument("--input-ids", type=str, default=None,
                   help="comma-separated prompt token ids (no tokenizer run)")
    p.add_argument("--text", type=str, default="The quick brown fox",
                   help="prompt text, encoded with the GPT BPE tokenizer")
    return p.parse_args()


def main():
    args = parse_args()
    cfg = get_config(args.config, overrides=args.override, show=True)
    init_dist_env(cfg)
    module = build_module(cfg)
    ckpt = cfg["Engine"].get("save_load", {}).get("ckpt_dir")
    if ckpt:
        from paddlefleetx_amd.core import EagerEngine
        EagerEngine(cfg, module, mode="eval").load(ckpt)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    module.model.to(device)
    if args.input_ids:
        ids = [int(t) for t in args.input_ids.split(",")]
    else:
        from paddlefleetx_amd.data.tokenizers import GPTTokenizer
        tok = GPTTokenizer.gpt2_tokenizer()
        ids = tok.encode(args.text)
    input_ids = torch.tensor([ids], dtype=torch.long, device=device)
    out = module.generate(input_ids)
    logger.info(f"prompt ids: {ids}")
    logger.info(f"generated ids: {out[0].tolist()}")
    if not args.input_ids:
        logger.info(f"generated text: {tok.decode(out[0].tolist())!r}")


if __name__ == "__main__":
    main()
