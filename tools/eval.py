#!/usr/bin/env python3
"""Offline evaluation entry (reference tools/eval.py):
WikiText PPL / LAMBADA accuracy via GPTEvalModule, or plain validation
loop for other modules.

    python tools/eval.py -c cfg.yaml -o Offline_Eval.cloze_eval=False
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from paddlefleetx_amd.core import EagerEngine
from paddlefleetx_amd.data import build_dataloader
from paddlefleetx_amd.models import build_module
from paddlefleetx_amd.parallel.env import init_dist_env
from paddlefleetx_amd.utils.config import get_config


def parse_args():
    p = argparse.ArgumentParser("eval")
    p.add_argument("-c", "--config", required=True)
    p.add_argument("-o", "--override", action="append", default=[])
    return p.parse_args()


def main():
    args = parse_args()
    cfg = get_config(args.config, overrides=args.override, show=True)
    init_dist_env(cfg)
    module = build_module(cfg)
    loader = build_dataloader(cfg, "Eval")
    engine = EagerEngine(cfg, module, mode="eval")
    engine.eval_iters = int(cfg.get("Offline_Eval", {}).get("max_iters",
                                                            10 ** 9))
    engine.evaluate(loader)
    if hasattr(module, "validation_epoch_end"):
        module.validation_epoch_end()


if __name__ == "__main__":
    main()
