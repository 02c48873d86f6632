"""ResNet + MoCo."""

import pytest
import torch

from paddlefleetx_amd.parallel.env import set_hcg
from paddlefleetx_amd.parallel.topology import HybridTopology


@pytest.fixture(autouse=True)
def _env():
    set_hcg(HybridTopology())
    yield


def test_resnet18_forward():
    from paddlefleetx_amd.models.resnet import resnet18
    m = resnet18(class_num=10)
    y = m(torch.randn(2, 3, 64, 64))
    assert y.shape == (2, 10)


def test_resnet50_features():
    from paddlefleetx_amd.models.resnet import resnet50
    m = resnet50(class_num=0, with_pool=False)
    f = m.forward_features(torch.randn(1, 3, 64, 64))
    assert f.shape[1] == 2048  # bottleneck expansion 4
    assert m.num_features == 2048


def test_moco_forward_and_queue():
    from paddlefleetx_amd.models.moco import (MoCo, MoCoClassifier,
                                              MoCoV2Projector)
    from paddlefleetx_amd.models.resnet import resnet18

    def enc():
        return resnet18(class_num=0, with_pool=False)

    feats = 512
    m = MoCo(enc(), MoCoV2Projector(True, feats, feats),
             MoCoClassifier(False, feats, 32),
             enc(), MoCoV2Projector(True, feats, feats),
             MoCoClassifier(False, feats, 32), dim=32, K=64, m=0.99, T=0.07)
    x1 = torch.randn(4, 3, 32, 32)
    x2 = torch.randn(4, 3, 32, 32)
    ptr0 = int(m.queue_ptr[0])
    logits, labels = m(x1, x2)
    assert logits.shape == (4, 1 + 64)
    assert labels.tolist() == [0, 0, 0, 0]
    assert int(m.queue_ptr[0]) == (ptr0 + 4) % 64
    # momentum params must stay grad-free
    assert all(not p.requires_grad for p in m.momentum_encoder.parameters())
    loss = torch.nn.functional.cross_entropy(logits, labels)
    loss.backward()
    assert next(m.base_encoder.parameters()).grad is not None


def test_momentum_update_moves_toward_base():
    from paddlefleetx_amd.models.moco import (MoCo, MoCoClassifier,
                                              MoCoV2Projector)
    from paddlefleetx_amd.models.resnet import resnet18

    def enc():
        return resnet18(class_num=0, with_pool=False)
    m = MoCo(enc(), MoCoV2Projector(True, 512, 512),
             MoCoClassifier(False, 512, 16),
             enc(), MoCoV2Projector(True, 512, 512),
             MoCoClassifier(False, 512, 16), dim=16, K=32, m=0.5)
    pb = next(m.base_encoder.parameters())
    pm = next(m.momentum_encoder.parameters())
    with torch.no_grad():
        pb.add_(1.0)
    before = (pm - pb).abs().mean()
    m._update_momentum_encoder()
    after = (pm - pb).abs().mean()
    assert after < before


def test_moco_module_builds():
    from paddlefleetx_amd.models import build_module
    cfg = {"Global": {"global_batch_size": 2},
           "Engine": {"mix_precision": {"enable": False}},
           "Model": {"name": "MOCOModule",
                     "model": {"backbone": "resnet18", "dim": 16, "K": 32,
                               "v2": True}}}
    mod = build_module(cfg)
    x = torch.randn(2, 3, 32, 32)
    loss = mod.training_step(((x, x), None))
    assert loss.ndim == 0
