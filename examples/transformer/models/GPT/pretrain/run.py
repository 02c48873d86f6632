#!/usr/bin/env python3
"""Decomposed pretraining loop (no EagerEngine).

Reference: examples/transformer/models/GPT/pretrain/run.py:36-260 +
impls.py:118-246 — the newer PaddleFleetX API assembles the training loop
from `components` builders instead of the engine. Same here: every piece
the engine automates is called explicitly, so the control flow is fully
visible and hackable.

    python examples/transformer/models/GPT/pretrain/run.py \
        [-c config.yaml] [-o key=val ...]
"""

import argparse
import os
import sys
import time

REPO = os.path.abspath(os.path.join(os.path.dirname(__file__),
                                    "..", "..", "..", "..", ".."))
sys.path.insert(0, REPO)

import torch

from paddlefleetx_amd.data import build_dataloader
from paddlefleetx_amd.models import build_module
from paddlefleetx_amd.optims import build_lr_scheduler, build_optimizer
from paddlefleetx_amd.parallel.env import init_dist_env
from paddlefleetx_amd.utils.config import get_config
from paddlefleetx_amd.utils.log import logger


def parse_args():
    p = argparse.ArgumentParser("gpt-pretrain-decomposed")
    p.add_argument("-c", "--config",
                   default=os.path.join(
                       REPO, "paddlefleetx_amd/configs/nlp/gpt/"
                             "pretrain_gpt_345M_single_card.yaml"))
    p.add_argument("-o", "--override", action="append", default=[])
    p.add_argument("--max-steps", type=int, default=20)
    return p.parse_args()


def main():
    args = parse_args()
    # -- components (impls.py:118-246 equivalents) --
    cfg = get_config(args.config, overrides=args.override)
    hcg = init_dist_env(cfg)
    module = build_module(cfg)
    loader = build_dataloader(cfg, "Train")
    lr_sched = build_lr_scheduler(cfg["Optimizer"].get("lr", {}))
    optimizer = build_optimizer(cfg["Optimizer"], module.model,
                                lr_value=lr_sched.get_lr())
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    module.model.to(device)
    acc_steps = int(cfg["Engine"].get("accumulate_steps", 1))

    # -- explicit fit loop (run.py:36-260 equivalent) --
    module.model.train()
    t0 = time.time()
    for step, batch in enumerate(loader):
        if step >= args.max_steps:
            break
        batch = tuple(t.to(device) if torch.is_tensor(t) else t
                      for t in batch)
        micros = [tuple(t.chunk(acc_steps)[i] if torch.is_tensor(t) else t
                        for t in batch) for i in range(acc_steps)]
        total = 0.0
        for mb in micros:
            loss = module.training_step(mb)
            (loss / acc_steps).backward()
            total += float(loss.detach())
        optimizer.reduce_gradients(hcg.get_data_parallel_group()) \
            if hasattr(optimizer, "reduce_gradients") else None
        lr_sched.step()
        if hasattr(optimizer, "step") and "lr" in \
                optimizer.step.__code__.co_varnames:
            optimizer.step(lr=lr_sched.get_lr())
        else:
            optimizer.step()
        optimizer.zero_grad()
        if step % 5 == 0:
            cost = (time.time() - t0) / (step + 1)
            gbs = cfg["Global"]["global_batch_size"]
            seq = cfg["Model"].get("max_position_embeddings", 1024)
            logger.train(f"[train] batch: {step}, loss: {total/acc_steps:.6f},"
                         f" avg_batch_cost: {cost:.5f} sec, "
                         f"ips: {gbs*seq/cost:.0f} tokens/s")


if __name__ == "__main__":
    main()
