#!/usr/bin/env python3
"""Inference over an exported model (reference tools/inference.py,
tasks/gpt/inference.py:35-60).

    python tools/inference.py -c cfg.yaml --model-dir ./exported_model \
        --input-ids 464,3290,318
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from paddlefleetx_amd.core.inference_engine import InferenceEngine
from paddlefleetx_amd.parallel.env import init_dist_env
from paddlefleetx_amd.utils.config import get_config
from paddlefleetx_amd.utils.log import logger


def parse_args():
    p = argparse.ArgumentParser("inference")
    p.add_argument("-c", "--config", required=True)
    p.add_argument("-o", "--override", action="append", default=[])
    p.add_argument("--model-dir", default="./exported_model")
    p.add_argument("--input-ids", type=str, default="464,3290,318")
    return p.parse_args()


def main():
    args = parse_args()
    cfg = get_config(args.config, overrides=args.override, show=True)
    init_dist_env(cfg)
    inf = cfg.get("Inference", {}) or {}
    model_dir = args.model_dir if args.model_dir != "./exported_model" \
        else inf.get("model_dir", args.model_dir)
    mp = int(inf.get("mp_degree") or
             cfg.get("Distributed", {}).get("mp_degree", 1) or 1)
    gen = dict(cfg.get("Generation") or {})
    if inf.get("fp8"):
        gen["fp8"] = True  # opt-in serving fp8 (InferenceEngine)
    engine = InferenceEngine(model_dir, mp_degree=mp,
                             generation_cfg=gen or None)
    ids = [int(t) for t in args.input_ids.split(",")]
    out = engine.predict(ids)
    logger.info(f"input ids: {ids}")
    logger.info(f"generated ids: {out[0].tolist()}")


if __name__ == "__main__":
    main()
