#!/usr/bin/env python3
"""Training entry point: `python tools/train.py -c cfg.yaml -o Key.sub=val`.

Reference surface: /root/reference/tools/train.py:44-73.
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
    p = argparse.ArgumentParser("train")
    p.add_argument("-c", "--config", required=True, help="yaml config path")
    p.add_argument("-o", "--override", action="append", default=[],
                   help="override config option a.b.c=v (repeatable)")
    return p.parse_args()


def main():
    args = parse_args()
    cfg = get_config(args.config, overrides=args.override, show=True)
    init_dist_env(cfg)
    module = build_module(cfg)
    train_loader = build_dataloader(cfg, "Train") if "Data" in cfg else None
    valid_loader = build_dataloader(cfg, "Eval") \
        if "Data" in cfg and "Eval" in cfg["Data"] else None
    engine = EagerEngine(cfg, module, mode="train")
    engine.fit(train_loader, valid_loader)


if __name__ == "__main__":
    main()
