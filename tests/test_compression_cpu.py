"""Compression: pruning + int8 weight quantization."""

import torch
import torch.nn as nn

from paddlefleetx_amd.utils.compression_helper import (QuantizedLinear,
                                                       prune_model,
                                                       quant_model,
                                                       quantization_error)


def _net():
    torch.manual_seed(0)
    return nn.Sequential(nn.Linear(16, 32), nn.ReLU(), nn.Linear(32, 8))


def test_structured_prune_zeroes_channels():
    net = _net()
    report = prune_model(net, ratio=0.25, structured=True)
    assert len(report) == 2
    w = net[0].weight.data
    zero_rows = (w.abs().sum(dim=1) == 0).sum()
    assert zero_rows == 8  # 25% of 32 output channels


def test_unstructured_prune_ratio():
    net = _net()
    report = prune_model(net, ratio=0.5, structured=False)
    for name, sparsity in report.items():
        assert 0.45 <= sparsity <= 0.55


def test_quantization_roundtrip_error_small():
    net = _net().eval()
    import copy
    qnet = copy.deepcopy(net)
    n = quant_model(qnet)
    assert n == 2
    assert isinstance(qnet[0], QuantizedLinear)
    x = torch.randn(4, 16)
    err = quantization_error(net, qnet, (x,))
    ref_mag = net(x).abs().mean()
    assert err < 0.05 * float(ref_mag) + 0.02


def test_quant_include_filter():
    net = _net()
    n = quant_model(net, include=["0"])
    assert n == 1
    assert isinstance(net[0], QuantizedLinear)
    assert isinstance(net[2], nn.Linear)
