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


# ---------------------------------------------------------------------------
# QAT: fake-quant training path (reference compression_helper.py:210)
# ---------------------------------------------------------------------------

def test_qat_fake_quant_ste():
    import torch
    from paddlefleetx_amd.utils.compression_helper import fake_quant
    x = torch.randn(16, requires_grad=True)
    scale = torch.tensor(0.1)
    y = fake_quant(x, scale)
    # forward snaps to the grid
    assert torch.allclose(y / scale, torch.round(y / scale), atol=1e-5)
    # straight-through gradient
    y.sum().backward()
    assert torch.allclose(x.grad, torch.ones_like(x))


def test_qat_wrap_train_convert():
    import torch
    import torch.nn as nn
    from paddlefleetx_amd.utils.compression_helper import (QATLinear,
                                                           QuantizedLinear,
                                                           convert_qat,
                                                           qat_model)
    torch.manual_seed(0)
    model = nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 4))
    n = qat_model(model)
    assert n == 2 and isinstance(model[0], QATLinear)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    x = torch.randn(32, 8)
    target = torch.randn(32, 4)
    losses = []
    for _ in range(30):
        opt.zero_grad()
        loss = ((model(x) - target) ** 2).mean()
        loss.backward()
        opt.step()
        losses.append(float(loss))
    assert losses[-1] < losses[0]  # trains THROUGH the fake quant
    # observer accumulated an activation range
    assert float(model[0].act_absmax) > 0
    # convert to int8 inference layers; outputs close to the QAT eval
    model.eval()
    with torch.no_grad():
        y_qat = model(x)
    convert_qat(model)
    assert isinstance(model[0], QuantizedLinear)
    with torch.no_grad():
        y_int8 = model(x)
    assert (y_qat - y_int8).abs().mean() < 0.2


def test_engine_compress_qat_path():
    import torch
    from paddlefleetx_amd.parallel.env import set_hcg
    from paddlefleetx_amd.parallel.topology import HybridTopology
    import torch.distributed as dist
    if not dist.is_initialized():
        set_hcg(HybridTopology())
    from paddlefleetx_amd.utils.config import get_config
    from paddlefleetx_amd.models import build_module
    from paddlefleetx_amd.core import EagerEngine
    import os
    repo = os.path.join(os.path.dirname(__file__), "..")
    cfg = get_config(os.path.join(
        repo, "paddlefleetx_amd/configs/nlp/gpt/pretrain_gpt_345M_single_card.yaml"),
        overrides=["Model.hidden_size=32", "Model.num_layers=1",
                   "Model.num_attention_heads=4", "Model.vocab_size=128",
                   "Model.max_position_embeddings=32",
                   "Global.micro_batch_size=2", "Global.local_batch_size=2",
                   "Engine.mix_precision.enable=False"])
    cfg["Compress"] = {"Quantization": {"enable_qat": True,
                                        "include": ["ffn"]}}
    module = build_module(cfg)
    engine = EagerEngine(cfg, module)
    engine.compress_model()
    from paddlefleetx_amd.utils.compression_helper import QATLinear
    qat_layers = [m for m in module.model.modules()
                  if isinstance(m, QATLinear)]
    assert qat_layers
