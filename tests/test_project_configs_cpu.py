"""New project configs (imagen / moco / qat / prune) load and run one
training step on CPU with tiny overrides."""

import os

import pytest
import torch

REPO = os.path.join(os.path.dirname(__file__), "..")
CFG = os.path.join(REPO, "paddlefleetx_amd", "configs")


def _cfg(rel, overrides):
    import sys
    sys.path.insert(0, REPO)
    from paddlefleetx_amd.utils.config import get_config
    return get_config(os.path.join(CFG, rel), overrides=overrides)


def test_imagen_config_one_step():
    from paddlefleetx_amd.data import build_dataloader
    from paddlefleetx_amd.models import build_module
    cfg = _cfg("mm/imagen/text2im_397M_64x64_single_card.yaml", [
        "Model.unet_name=Unet",
        "Model.image_size=16",
        "Model.text_embed_dim=32",
        "Model.text_encoder_kwargs.num_layers=1",
        "Model.text_encoder_kwargs.num_heads=2",
        "Model.text_encoder_kwargs.d_ff=64",
        "Model.text_encoder_kwargs.vocab_size=512",
        "Model.unet_kwargs.dim=16",
        "Model.unet_kwargs.dim_mults=[1,2]",
        "Model.unet_kwargs.layer_attns=[False,True]",
        "Model.unet_kwargs.layer_cross_attns=[False,True]",
        "Model.unet_kwargs.attn_heads=2",
        "Model.unet_kwargs.attn_dim_head=8",
        "Engine.mix_precision.enable=False",
        "Global.global_batch_size=2", "Global.local_batch_size=2",
        "Global.micro_batch_size=2",
        "Data.Train.dataset.num_samples=4",
        "Data.Train.dataset.image_size=16",
        "Data.Train.dataset.text_len=8",
        "Data.Train.sampler.batch_size=2",
        "Data.Train.loader.num_workers=0",
    ])
    mod = build_module(cfg)
    dl = build_dataloader(cfg, "Train")
    batch = next(iter(dl))
    loss = mod.training_step(batch)
    assert torch.isfinite(loss)
    loss.backward()


def test_moco_config_one_step():
    from paddlefleetx_amd.data import build_dataloader
    from paddlefleetx_amd.models import build_module
    cfg = _cfg("vis/moco/mocov2_pretrain_single_card.yaml", [
        "Model.model.backbone=resnet18",
        "Model.model.K=32", "Model.model.dim=16",
        "Engine.mix_precision.enable=False",
        "Global.global_batch_size=2", "Global.local_batch_size=2",
        "Global.micro_batch_size=2",
        "Data.Train.dataset.num_samples=4",
        "Data.Train.dataset.image_size=32",
        "Data.Train.sampler.batch_size=2",
        "Data.Train.loader.num_workers=0",
    ])
    mod = build_module(cfg)
    dl = build_dataloader(cfg, "Train")
    batch = next(iter(dl))
    loss = mod.training_step(batch)
    assert torch.isfinite(loss)
    loss.backward()
    # Momentum optimizer + CosineAnnealingDecay resolve
    from paddlefleetx_amd.optims import build_optimizer
    from paddlefleetx_amd.optims.lr_scheduler import build_lr_scheduler
    opt = build_optimizer(dict(cfg["Optimizer"]), mod.model, lr_value=0.03)
    opt.step()
    sched = build_lr_scheduler(dict(cfg["Optimizer"]["lr"]))
    assert sched.get_lr() > 0


@pytest.mark.parametrize("rel", [
    "nlp/gpt/qat_gpt_345M_single_card.yaml",
    "nlp/gpt/prune_gpt_345M_single_card.yaml",
    "nlp/gpt/pretrain_gpt_1.3B_single_card.yaml",
    "nlp/gpt/pretrain_gpt_13B_single_card.yaml",
])
def test_gpt_variant_configs_load(rel):
    cfg = _cfg(rel, ["Model.hidden_size=64", "Model.num_layers=2",
                     "Model.num_attention_heads=4", "Model.vocab_size=128",
                     "Model.max_position_embeddings=32",
                     "Engine.mix_precision.enable=False",
                     "Global.global_batch_size=2",
                     "Global.local_batch_size=2",
                     "Global.micro_batch_size=2"])
    from paddlefleetx_amd.models import build_module
    mod = build_module(cfg)
    if "Compress" in cfg:
        from paddlefleetx_amd.core import EagerEngine
        eng = EagerEngine(cfg, mod, mode="train")
        eng.compress_model()
