#!/usr/bin/env python3
"""Decomposed offline LM evaluation (reference examples/transformer/
models/GPT/offline-eval/run.py + impls.py): sliding-window perplexity
over an eval set, no engine.

    python examples/transformer/models/GPT/offline-eval/run.py \
        [-c config.yaml] [-o key=val ...] [--max-iters N]
"""

import argparse
import math
import os
import sys

REPO = os.path.abspath(os.path.join(os.path.dirname(__file__),
                                    "..", "..", "..", "..", ".."))
sys.path.insert(0, REPO)

import torch

from paddlefleetx_amd.data import build_dataloader
from paddlefleetx_amd.models import build_module
from paddlefleetx_amd.parallel.env import init_dist_env
from paddlefleetx_amd.utils.config import get_config
from paddlefleetx_amd.utils.log import logger


def main():
    p = argparse.ArgumentParser("gpt-offline-eval-decomposed")
    p.add_argument("-c", "--config", default=os.path.join(
        REPO, "paddlefleetx_amd/configs/nlp/gpt/"
              "pretrain_gpt_345M_single_card.yaml"))
    p.add_argument("-o", "--override", action="append", default=[])
    p.add_argument("--max-iters", type=int, default=8)
    args = p.parse_args()
    cfg = get_config(args.config, overrides=args.override)
    init_dist_env(cfg)
    module = build_module(cfg)
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    module.model.to(device)
    module.model.eval()
    loader = build_dataloader(cfg, "Eval") or build_dataloader(cfg, "Train")

    total_loss, total_tokens = 0.0, 0
    with torch.no_grad():
        for i, batch in enumerate(loader):
            if i >= args.max_iters:
                break
            batch = tuple(t.to(device) for t in batch)
            loss = module.validation_step(batch)
            n = batch[3].sum() if len(batch) > 3 else batch[0].numel()
            total_loss += float(loss) * float(n)
            total_tokens += float(n)
    avg = total_loss / max(1.0, total_tokens)
    logger.eval(f"[offline-eval] avg loss: {avg:.6f}, "
                f"ppl: {math.exp(min(20.0, avg)):.3f}, "
                f"tokens: {int(total_tokens)}")


if __name__ == "__main__":
    main()
