#!/usr/bin/env python3
"""Export through the auto-parallel engine (reference tools/auto_export.py
-> AutoEngine.export_from_prog, auto_engine.py:165-209): the Distributed
section is PLANNED from model size + world size before the model is built
and exported.

    python tools/auto_export.py -c cfg.yaml [--output-dir DIR]
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from paddlefleetx_amd.core.auto_engine import AutoEngine
from paddlefleetx_amd.utils.config import get_config
from paddlefleetx_amd.utils.export import export_inference_model


def parse_args():
    p = argparse.ArgumentParser("auto_export")
    p.add_argument("-c", "--config", required=True)
    p.add_argument("-o", "--override", action="append", default=[])
    p.add_argument("--output-dir", default="./exported_model")
    return p.parse_args()


def main():
    args = parse_args()
    cfg = get_config(args.config, overrides=args.override, show=True)
    engine = AutoEngine(cfg, mode="eval")
    ckpt = engine.configs["Engine"].get("save_load", {}).get("ckpt_dir")
    if ckpt:
        engine.load(ckpt)
    extra = {"plan": engine.plan}
    if "Generation" in engine.configs:
        extra["generation"] = dict(engine.configs["Generation"])
    export_inference_model(engine.module.model,
                           dict(engine.configs["Model"]),
                           args.output_dir, extra=extra)


if __name__ == "__main__":
    main()
