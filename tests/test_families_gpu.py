"""GPU smokes for the model families not yet exercised on hardware by
test_models_gpu.py: DeBERTa-v2, Imagen (train loss + DDPM sample),
protein folding (engine step on the shipped tiny config), MoCo/ResNet
and T5ForConditionalGeneration. Each is a tiny forward+backward on
cuda:0 — the point is that every family in SURVEY §2.3 runs end-to-end
on the MI355X path, not just under gloo/CPU."""

import os

import pytest
import torch

from paddlefleetx_amd.parallel.env import set_hcg, set_seed
from paddlefleetx_amd.parallel.topology import HybridTopology

pytestmark = pytest.mark.gpu

REPO = os.path.join(os.path.dirname(__file__), "..")


@pytest.fixture(autouse=True)
def _env():
    set_hcg(HybridTopology())
    set_seed(1234)
    yield


def test_debertav2_step_gpu():
    from paddlefleetx_amd.models.debertav2 import DebertaV2Model
    m = DebertaV2Model(vocab_size=200, hidden_size=64, num_hidden_layers=2,
                       num_attention_heads=4, intermediate_size=128,
                       max_position_embeddings=64, position_buckets=16,
                       hidden_dropout_prob=0.0,
                       attention_probs_dropout_prob=0.0
                       ).cuda().to(torch.bfloat16)
    ids = torch.randint(0, 200, (2, 32), device="cuda")
    out = m(ids)
    assert out.shape == (2, 32, 64)
    loss = out.float().square().mean()
    loss.backward()
    assert torch.isfinite(loss)
    assert m.rel_embeddings.weight.grad is not None
    assert torch.isfinite(m.rel_embeddings.weight.grad.float()).all()


def test_imagen_loss_and_sample_gpu():
    from paddlefleetx_amd.models.imagen import ImagenModel, Unet
    unet = Unet(dim=16, dim_mults=(1, 2), num_resnet_blocks=1,
                layer_attns=(False, True), layer_cross_attns=(False, True),
                attn_heads=2, attn_dim_head=8, text_embed_dim=32, groups=4)
    m = ImagenModel(unet=unet, image_size=16, text_embed_dim=32,
                    text_encoder_kwargs=dict(vocab_size=64, d_kv=8, d_ff=64,
                                             num_layers=1, num_heads=4,
                                             dropout_rate=0.0)).cuda()
    imgs = torch.randn(2, 3, 16, 16, device="cuda")
    ids = torch.randint(0, 64, (2, 6), device="cuda")
    loss = m(imgs, text_ids=ids)
    assert loss.ndim == 0 and torch.isfinite(loss)
    loss.backward()
    assert m.unet.init_conv.weight.grad is not None
    out = m.sample(text_ids=ids, batch_size=2, steps=3)
    assert out.shape == (2, 3, 16, 16) and torch.isfinite(out).all()


def test_folding_engine_step_gpu():
    """FoldingModule (Evoformer + IPA structure module + FAPE/torsion
    losses) through the EagerEngine on the shipped tiny config — the
    engine places it on cuda:0 by itself."""
    from paddlefleetx_amd.utils.config import get_config
    from paddlefleetx_amd.models import build_module
    from paddlefleetx_amd.core import EagerEngine
    from paddlefleetx_amd.data import build_dataloader
    cfg = get_config(os.path.join(
        REPO, "paddlefleetx_amd/configs/folding/pretrain_folding_tiny.yaml"),
        overrides=["Model.num_evoformer_blocks=1",
                   "Model.num_structure_layers=1",
                   "Data.Train.dataset.num_res=16",
                   "Data.Train.dataset.num_samples=8"])
    module = build_module(cfg)
    engine = EagerEngine(cfg, module)
    assert engine.device.type == "cuda"
    loader = build_dataloader(cfg, "Train")
    batch = next(iter(loader))
    l0 = engine._fit_impl(batch)
    l1 = engine._fit_impl(batch)
    assert torch.isfinite(l0) and torch.isfinite(l1)


def test_moco_step_gpu():
    from paddlefleetx_amd.models.moco import (MoCo, MoCoClassifier,
                                              MoCoV2Projector)
    from paddlefleetx_amd.models.resnet import resnet18

    def enc():
        return resnet18(class_num=0, with_pool=False)

    feats = 512
    m = MoCo(enc(), MoCoV2Projector(True, feats, feats),
             MoCoClassifier(False, feats, 32),
             enc(), MoCoV2Projector(True, feats, feats),
             MoCoClassifier(False, feats, 32),
             dim=32, K=64, m=0.99, T=0.07).cuda()
    x1 = torch.randn(4, 3, 32, 32, device="cuda")
    x2 = torch.randn(4, 3, 32, 32, device="cuda")
    logits, labels = m(x1, x2)
    assert logits.shape == (4, 1 + 64)
    loss = torch.nn.functional.cross_entropy(logits, labels)
    loss.backward()
    assert torch.isfinite(loss)
    assert next(m.base_encoder.parameters()).grad is not None


def test_t5_conditional_generation_gpu():
    from paddlefleetx_amd.models.t5 import T5ForConditionalGeneration
    m = T5ForConditionalGeneration(
        vocab_size=100, d_model=64, d_kv=16, d_ff=128, num_layers=2,
        num_heads=4, dropout_rate=0.0).cuda()
    src = torch.randint(2, 100, (2, 12), device="cuda")
    tgt = torch.randint(2, 100, (2, 8), device="cuda")
    loss, logits = m(src, labels=tgt)
    assert logits.shape == (2, 8, 100)
    loss.backward()
    assert torch.isfinite(loss)
    seq = m.generate(src, max_length=6)
    assert seq.shape[0] == 2 and seq.shape[1] <= 7
    assert seq.is_cuda
