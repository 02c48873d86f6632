"""GLUE task datasets (tsv readers) + synthetic variant.

Reference: ppfleetx/data/dataset/glue_dataset.py — per-task readers
(CoLA :48, SST2 :135, MNLI :228, ...). One parametrized reader here: the
task table carries the tsv column layout and label set of each task.
Samples: (input_ids, label) with text pairs joined by the tokenizer's eos.
"""

from __future__ import annotations

import csv
import os
from typing import List, Optional, Tuple

import numpy as np
import torch
from torch.utils.data import Dataset

# task -> (sentence columns (train/dev), label column, labels or None=regression)
GLUE_TASKS = {
    "cola": dict(cols=(3,), label_col=1, labels=["0", "1"], header=False),
    "sst2": dict(cols=(0,), label_col=1, labels=["0", "1"], header=True),
    "mrpc": dict(cols=(3, 4), label_col=0, labels=["0", "1"], header=True),
    "stsb": dict(cols=(7, 8), label_col=9, labels=None, header=True),
    "qqp": dict(cols=(3, 4), label_col=5, labels=["0", "1"], header=True),
    "mnli": dict(cols=(8, 9), label_col=-1,
                 labels=["contradiction", "entailment", "neutral"],
                 header=True),
    "qnli": dict(cols=(1, 2), label_col=-1,
                 labels=["entailment", "not_entailment"], header=True),
    "rte": dict(cols=(1, 2), label_col=-1,
                labels=["entailment", "not_entailment"], header=True),
    "wnli": dict(cols=(1, 2), label_col=-1, labels=["0", "1"], header=True),
}

# which metric each task reports (reference finetune configs)
GLUE_METRICS = {
    "cola": "Mcc", "sst2": "Accuracy", "mrpc": "AccuracyAndF1",
    "stsb": "PearsonAndSpearman", "qqp": "AccuracyAndF1", "mnli": "Accuracy",
    "qnli": "Accuracy", "rte": "Accuracy", "wnli": "Accuracy",
}


class GLUEDataset(Dataset):
    """Reads <root>/<split>.tsv for `task`, tokenizes with a GPT tokenizer."""

    collate_fn = None  # padding happens in __getitem__ (fixed max_length)

    def __init__(self, task: str, root: str, split: str = "train",
                 tokenizer=None, max_length: int = 128, mode: str = "Train",
                 **unused):
        task = task.lower()
        assert task in GLUE_TASKS, f"unknown GLUE task {task}"
        self.spec = GLUE_TASKS[task]
        self.task = task
        self.max_length = max_length
        if tokenizer is None:
            from paddlefleetx_amd.data.tokenizers import GPTTokenizer
            tokenizer = GPTTokenizer.gpt2_tokenizer()
        self.tokenizer = tokenizer
        self.samples: List[Tuple[str, Optional[str], float]] = []
        path = os.path.join(root, f"{split}.tsv")
        with open(path, encoding="utf-8") as f:
            reader = csv.reader(f, delimiter="\t", quoting=csv.QUOTE_NONE)
            rows = list(reader)
        if self.spec["header"]:
            rows = rows[1:]
        labels = self.spec["labels"]
        for row in rows:
            if not row:
                continue
            cols = self.spec["cols"]
            a = row[cols[0]]
            b = row[cols[1]] if len(cols) > 1 else None
            raw = row[self.spec["label_col"]]
            label = float(raw) if labels is None else labels.index(raw)
            self.samples.append((a, b, label))

    def __len__(self):
        return len(self.samples)

    def __getitem__(self, idx):
        a, b, label = self.samples[idx]
        ids = self.tokenizer.encode(a)
        if b is not None:
            ids = ids + [self.tokenizer.eos_token_id] + self.tokenizer.encode(b)
        ids = ids[:self.max_length]
        pad = self.max_length - len(ids)
        mask = [1] * len(ids) + [0] * pad
        ids = ids + [self.tokenizer.pad_token_id] * pad
        label_t = torch.tensor(label, dtype=torch.float32
                               if self.spec["labels"] is None else torch.long)
        return (torch.tensor(ids, dtype=torch.long),
                torch.tensor(mask, dtype=torch.long), label_t)


class SyntheticGLUEDataset(Dataset):
    """Random (ids, mask, label) tuples for plumbing tests/benchmarks."""

    collate_fn = None

    def __init__(self, num_samples: int = 1000, max_length: int = 128,
                 vocab_size: int = 50304, num_classes: int = 2,
                 regression: bool = False, mode: str = "Train", **unused):
        self.num_samples = num_samples
        self.max_length = max_length
        self.vocab_size = vocab_size
        self.num_classes = num_classes
        self.regression = regression

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(int(idx))
        n = int(torch.randint(4, self.max_length, (1,), generator=g))
        ids = torch.randint(0, self.vocab_size, (self.max_length,),
                            generator=g)
        mask = torch.zeros(self.max_length, dtype=torch.long)
        mask[:n] = 1
        if self.regression:
            label = torch.rand((), generator=g) * 5.0
        else:
            label = torch.randint(0, self.num_classes, (), generator=g)
        return ids, mask, label
