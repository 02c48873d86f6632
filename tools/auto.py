#!/usr/bin/env python3
"""Auto-parallel training entry (reference tools/auto.py -> AutoEngine).

    python tools/auto.py -c cfg.yaml [--tune]
The Distributed section is PLANNED from the model size and world size;
any user-specified degrees are overridden by the plan.
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from paddlefleetx_amd.core.auto_engine import AutoEngine
from paddlefleetx_amd.data import build_dataloader
from paddlefleetx_amd.utils.config import get_config


def parse_args():
    p = argparse.ArgumentParser("auto")
    p.add_argument("-c", "--config", required=True)
    p.add_argument("-o", "--override", action="append", default=[])
    p.add_argument("--tune", action="store_true",
                   help="run the accumulate-steps tuning pass first")
    return p.parse_args()


def main():
    args = parse_args()
    cfg = get_config(args.config, overrides=args.override, show=True)
    engine = AutoEngine(cfg)
    train_loader = build_dataloader(engine.configs, "Train")
    valid_loader = build_dataloader(engine.configs, "Eval") \
        if "Eval" in engine.configs.get("Data", {}) else None
    if args.tune and train_loader is not None:
        cands = (cfg.get("Tuning", {}) or {}).get("candidates")
        engine.tune(train_loader, candidates=cands)
    engine.fit(train_loader, valid_loader)


if __name__ == "__main__":
    main()
