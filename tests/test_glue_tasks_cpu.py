"""Per-task GLUE coverage: every task's tsv column layout parsed and
label-mapped (reference per-task readers glue_dataset.py:48-841), plus a
CE-style finetune convergence check on a learnable task (reference
convergence tests, run_benchmark.sh:160-162 CE_ prefix semantics — real
GLUE corpora are not downloadable in this environment, so the
convergence assertion runs on a separable synthetic task)."""

import csv
import os
import tempfile

import pytest
import torch

from paddlefleetx_amd.data.glue_dataset import (GLUE_METRICS, GLUE_TASKS,
                                                GLUEDataset)


class ToyTok:
    eos_token_id = 0
    pad_token_id = 0

    def encode(self, text):
        return [1 + (ord(c) % 50) for c in text[:16]]


def _write_tsv(path, task, rows):
    spec = GLUE_TASKS[task]
    ncols = max(max(spec["cols"]), spec["label_col"] if spec["label_col"] >= 0
                else 0) + 2
    with open(path, "w", encoding="utf-8", newline="") as f:
        w = csv.writer(f, delimiter="\t", quoting=csv.QUOTE_NONE,
                       escapechar="\\")
        if spec["header"]:
            w.writerow([f"col{i}" for i in range(ncols)])
        for a, b, label in rows:
            row = ["x"] * ncols
            row[spec["cols"][0]] = a
            if len(spec["cols"]) > 1:
                row[spec["cols"][1]] = b
            row[spec["label_col"] if spec["label_col"] >= 0 else ncols - 1] \
                = label
            w.writerow(row)


@pytest.mark.parametrize("task", sorted(GLUE_TASKS))
def test_glue_task_reader(task):
    spec = GLUE_TASKS[task]
    labels = spec["labels"]
    rows = []
    for i in range(6):
        lab = "2.5" if labels is None else labels[i % len(labels)]
        rows.append((f"sentence a {i}", f"sentence b {i}", lab))
    with tempfile.TemporaryDirectory() as td:
        _write_tsv(os.path.join(td, "dev.tsv"), task, rows)
        ds = GLUEDataset(task, td, split="dev", tokenizer=ToyTok(),
                         max_length=32)
        assert len(ds) == 6
        ids, mask, label = ds[0]
        assert ids.shape == (32,) and mask.shape == (32,)
        if labels is None:  # regression (STS-B)
            assert label.dtype == torch.float32 and float(label) == 2.5
        else:
            assert label.dtype == torch.long
            got = sorted({int(ds[i][2]) for i in range(6)})
            assert got == list(range(min(len(labels), 6)))[:len(got)]
        # the task is wired to its reference metric
        assert task in GLUE_METRICS


@pytest.mark.timeout(600)
def test_glue_finetune_ce_convergence():
    """A linearly-separable 2-class token task: finetune accuracy must
    reach >= 0.9 within a few hundred steps (convergence machinery,
    not dataset fidelity — no GLUE downloads offline)."""
    from paddlefleetx_amd.parallel.env import set_hcg
    from paddlefleetx_amd.parallel.topology import HybridTopology
    import torch.distributed as dist
    if not dist.is_initialized():
        set_hcg(HybridTopology())
    from paddlefleetx_amd.models import build_module
    from paddlefleetx_amd.core import EagerEngine
    from paddlefleetx_amd.utils.config import get_config
    repo = os.path.join(os.path.dirname(__file__), "..")
    cfg = get_config(os.path.join(
        repo, "paddlefleetx_amd/configs/nlp/gpt/pretrain_gpt_345M_single_card.yaml"),
        overrides=["Model.name=GPTFinetuneModule", "Model.task=sst2",
                   "Model.hidden_size=32", "Model.num_layers=2",
                   "Model.num_attention_heads=4", "Model.vocab_size=128",
                   "Model.max_position_embeddings=16",
                   "Model.hidden_dropout_prob=0.0",
                   "Model.attention_probs_dropout_prob=0.0",
                   "Global.micro_batch_size=16", "Global.local_batch_size=16",
                   "Engine.mix_precision.enable=False",
                   "Optimizer.lr.name=ConstantLR",
                   "Optimizer.lr.learning_rate=0.002"])
    module = build_module(cfg)
    engine = EagerEngine(cfg, module)

    def make_batch(step):
        g = torch.Generator().manual_seed(step)
        label = torch.randint(0, 2, (16,), generator=g)
        # class k sentences END with token 10+k — separable at the
        # classification read position (last real token)
        ids = torch.randint(20, 128, (16, 16), generator=g)
        ids[:, -1] = 10 + label
        mask = torch.ones(16, 16, dtype=torch.long)
        return ids, mask, label

    for s in range(120):
        engine._fit_impl(make_batch(s))
    module.model.eval()
    correct = total = 0
    with torch.no_grad():
        for s in range(1000, 1004):
            ids, mask, label = make_batch(s)
            logits = module.model(ids, attention_mask=mask)
            correct += int((logits.argmax(-1) == label).sum())
            total += 16
    acc = correct / total
    assert acc >= 0.9, acc
