"""Composable collate helpers (reference ppfleetx/data/sampler/collate.py
Stack:27 / Pad:70 / Tuple:173 / Dict:248 and
ppfleetx/data/utils/batch_collate_fn.py DataCollatorWithPadding:150).

Each helper is a callable over a list of per-sample fields; `Tuple` /
`Dict` compose them over tuple- or dict-shaped samples, so a dataset can
declare e.g. ``collate_fn = Tuple(Stack(), Pad(pad_val=0), Stack())``.
"""

from __future__ import annotations

from typing import Any, Callable, Dict as TDict, List, Optional, Sequence

import numpy as np
import torch

__all__ = ["Stack", "Pad", "Tuple", "Dict", "DataCollatorWithPadding"]


def _to_tensor(x, dtype=None):
    if isinstance(x, torch.Tensor):
        t = x
    elif isinstance(x, np.ndarray):
        t = torch.from_numpy(x)
    else:
        t = torch.as_tensor(x)
    return t.to(dtype) if dtype is not None else t


class Stack:
    """Stack equal-shape fields along a new leading batch axis."""

    def __init__(self, dtype=None):
        self.dtype = dtype

    def __call__(self, data: Sequence[Any]) -> torch.Tensor:
        return torch.stack([_to_tensor(d, self.dtype) for d in data])


class Pad:
    """Pad variable-length 1-D (or leading-dim variable N-D) fields to
    the batch max length with `pad_val`. `ret_length` additionally
    returns the original lengths (reference collate.py:70 semantics)."""

    def __init__(self, pad_val=0, axis: int = 0, ret_length: bool = False,
                 dtype=None):
        self.pad_val = pad_val
        self.axis = axis
        self.ret_length = ret_length
        self.dtype = dtype

    def __call__(self, data: Sequence[Any]):
        ts = [_to_tensor(d, self.dtype) for d in data]
        lengths = torch.tensor([t.shape[self.axis] for t in ts],
                               dtype=torch.long)
        max_len = int(lengths.max()) if len(ts) else 0
        out = []
        for t in ts:
            pad_n = max_len - t.shape[self.axis]
            if pad_n > 0:
                pad_shape = list(t.shape)
                pad_shape[self.axis] = pad_n
                filler = torch.full(pad_shape, self.pad_val, dtype=t.dtype)
                t = torch.cat([t, filler], dim=self.axis)
            out.append(t)
        batch = torch.stack(out)
        return (batch, lengths) if self.ret_length else batch


class Tuple:
    """Apply the i-th sub-collate to the i-th field of tuple samples."""

    def __init__(self, *fns: Callable):
        if len(fns) == 1 and isinstance(fns[0], (list, tuple)):
            fns = tuple(fns[0])
        self.fns = fns

    def __call__(self, samples: Sequence[Sequence[Any]]):
        assert all(len(s) == len(self.fns) for s in samples), \
            "sample arity != number of collate fns"
        return tuple(fn([s[i] for s in samples])
                     for i, fn in enumerate(self.fns))


class Dict:
    """Apply per-key sub-collates to dict samples; keys missing a collate
    pass through as a plain list."""

    def __init__(self, fns: TDict[str, Callable]):
        self.fns = fns

    def __call__(self, samples: Sequence[TDict[str, Any]]):
        keys = samples[0].keys()
        out = {}
        for k in keys:
            vals = [s[k] for s in samples]
            out[k] = self.fns[k](vals) if k in self.fns else vals
        return out


class DataCollatorWithPadding:
    """Tokenizer-driven batch collate: pads dict-encoded samples
    (input_ids / token_type_ids / attention_mask / labels) to the batch
    max via `tokenizer.pad` (reference batch_collate_fn.py:150)."""

    def __init__(self, tokenizer, padding: bool = True,
                 max_length: Optional[int] = None,
                 return_tensors: str = "pt"):
        self.tokenizer = tokenizer
        self.padding = padding
        self.max_length = max_length
        self.return_tensors = return_tensors

    def __call__(self, features: List[TDict[str, Any]]):
        from paddlefleetx_amd.data.tokenizers.tokenization_base import \
            BatchEncoding
        labels = None
        if features and "labels" in features[0]:
            features = [dict(f) for f in features]
            labels = [f.pop("labels") for f in features]
        # list-of-dicts -> BatchEncoding (dict of key -> list of rows)
        enc = BatchEncoding({k: [f[k] for f in features]
                             for k in features[0].keys()})
        batch = self.tokenizer.pad(enc, padding=self.padding,
                                   max_length=self.max_length)
        if labels is not None:
            batch["labels"] = labels
        if self.return_tensors:
            batch.convert_to_tensors(self.return_tensors)
        return batch
