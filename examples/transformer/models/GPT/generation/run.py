#!/usr/bin/env python3
"""Decomposed generation (reference examples/transformer/models/GPT/
generation/run.py + impls.py): build the generation module and decode a
prompt, no engine.

    python examples/transformer/models/GPT/generation/run.py \
        [-c config.yaml] [-o key=val ...] [--input-ids 5,17,101]
"""

import argparse
import os
import sys

REPO = os.path.abspath(os.path.join(os.path.dirname(__file__),
                                    "..", "..", "..", "..", ".."))
sys.path.insert(0, REPO)

import torch

from paddlefleetx_amd.models import build_module
from paddlefleetx_amd.parallel.env import init_dist_env
from paddlefleetx_amd.utils.config import get_config
from paddlefleetx_amd.utils.log import logger


def main():
    p = argparse.ArgumentParser("gpt-generation-decomposed")
    p.add_argument("-c", "--config", default=os.path.join(
        REPO, "paddlefleetx_amd/configs/nlp/gpt/"
              "generation_gpt_345M_single_card.yaml"))
    p.add_argument("-o", "--override", action="append", default=[])
    p.add_argument("--input-ids", type=str, default="5,17,101")
    args = p.parse_args()
    cfg = get_config(args.config, overrides=args.override)
    init_dist_env(cfg)
    module = build_module(cfg)
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    module.model.to(device)
    ids = torch.tensor([[int(t) for t in args.input_ids.split(",")]],
                       dtype=torch.long, device=device)
    out = module.generate(ids)
    logger.info(f"generated ids: {out[0].tolist()}")


if __name__ == "__main__":
    main()
