"""Composable collate helpers (Stack/Pad/Tuple/Dict,
DataCollatorWithPadding) and the tensor-native RandAugment transform."""

import torch

from paddlefleetx_amd.data.collate import (DataCollatorWithPadding, Dict,
                                           Pad, Stack, Tuple)


def test_stack_and_pad():
    s = Stack()
    out = s([torch.tensor([1, 2]), torch.tensor([3, 4])])
    assert out.shape == (2, 2) and out[1, 0] == 3

    p = Pad(pad_val=-1, ret_length=True)
    batch, lengths = p([[1, 2, 3], [4], [5, 6]])
    assert batch.shape == (3, 3)
    assert batch[1].tolist() == [4, -1, -1]
    assert lengths.tolist() == [3, 1, 2]


def test_tuple_and_dict_compose():
    t = Tuple(Stack(), Pad(pad_val=0))
    a, b = t([(torch.tensor([1]), [1, 2]), (torch.tensor([2]), [3])])
    assert a.tolist() == [[1], [2]]
    assert b.tolist() == [[1, 2], [3, 0]]

    d = Dict({"x": Stack()})
    out = d([{"x": torch.tensor(1), "y": "a"}, {"x": torch.tensor(2),
                                                "y": "b"}])
    assert out["x"].tolist() == [1, 2]
    assert out["y"] == ["a", "b"]  # un-collated keys pass through


def test_data_collator_with_padding():
    from paddlefleetx_amd.data.tokenizers.ernie_tokenizer import \
        ErnieTokenizer
    vocab = {t: i for i, t in enumerate(
        ["[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]", "a", "b", "c"])}
    tok = ErnieTokenizer(vocab)
    coll = DataCollatorWithPadding(tok)
    feats = [
        {"input_ids": [2, 5, 6, 3], "token_type_ids": [0, 0, 0, 0],
         "labels": 1},
        {"input_ids": [2, 7, 3], "token_type_ids": [0, 0, 0], "labels": 0},
    ]
    batch = coll(feats)
    assert batch["input_ids"].shape == (2, 4)
    assert batch["input_ids"][1].tolist() == [2, 7, 3, 0]  # [PAD]=0
    assert batch["token_type_ids"].shape == (2, 4)
    assert batch["labels"].tolist() == [1, 0]


def test_randaugment_shapes_and_range():
    from paddlefleetx_amd.data.vision_dataset import RandAugment
    g = torch.Generator().manual_seed(0)
    ra = RandAugment(num_ops=2, magnitude=9, generator=g)
    x = torch.rand(3, 24, 24)
    for _ in range(40):  # cycle through the op table
        y = ra(x)
        assert y.shape == x.shape
        assert torch.isfinite(y).all()
        assert float(y.min()) >= -1e-5 and float(y.max()) <= 1.0 + 1e-5


def test_randaugment_identity_at_zero_magnitude():
    from paddlefleetx_amd.data.vision_dataset import RandAugment
    ra = RandAugment(num_ops=2, magnitude=0)
    x = torch.rand(3, 16, 16)
    # the magnitude-scaled ops reduce to identity at m=0 (affine ones via
    # identity-grid resampling); posterize/solarize/autocontrast are
    # magnitude-independent by construction and excluded
    for op in ("identity", "brightness", "contrast", "sharpness", "rotate",
               "shear_x", "shear_y", "translate_x", "translate_y"):
        y = ra._apply(x, op, 1.0)
        assert torch.allclose(y, x, atol=1e-5), op
