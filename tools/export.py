#!/usr/bin/env python3
"""Export a trained model for inference (reference tools/export.py).

    python tools/export.py -c cfg.yaml -o Engine.save_load.ckpt_dir=...
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from paddlefleetx_amd.core import EagerEngine
from paddlefleetx_amd.models import build_module
from paddlefleetx_amd.parallel.env import init_dist_env
from paddlefleetx_amd.utils.config import get_config
from paddlefleetx_amd.utils.export import export_inference_model


def parse_args():
    p = argparse.ArgumentParser("export")
    p.add_argument("-c", "--config", required=True)
    p.add_argument("-o", "--override", action="append", default=[])
    p.add_argument("--output-dir", default="./exported_model")
    return p.parse_args()


def main():
    args = parse_args()
    cfg = get_config(args.config, overrides=args.override, show=True)
    init_dist_env(cfg)
    module = build_module(cfg)
    ckpt = cfg["Engine"].get("save_load", {}).get("ckpt_dir")
    if ckpt:
        EagerEngine(cfg, module, mode="eval").load(ckpt)
    extra = {}
    if "Generation" in cfg:
        extra["generation"] = dict(cfg["Generation"])
    export_inference_model(module.model, dict(cfg["Model"]),
                           args.output_dir, extra=extra)


if __name__ == "__main__":
    main()
