"""Export -> InferenceEngine round trip + profiler guard."""

import pytest
import torch

from paddlefleetx_amd.parallel.env import set_hcg
from paddlefleetx_amd.parallel.topology import HybridTopology


@pytest.fixture(autouse=True)
def _env():
    set_hcg(HybridTopology())
    yield


def test_export_and_inference_roundtrip(tmp_path):
    from paddlefleetx_amd.core.inference_engine import InferenceEngine
    from paddlefleetx_amd.models.gpt.generation import GPTForGeneration
    from paddlefleetx_amd.models.gpt.model import GPTModel
    from paddlefleetx_amd.utils.export import export_inference_model

    torch.manual_seed(0)
    mcfg = dict(vocab_size=128, hidden_size=32, num_layers=1,
                num_attention_heads=2, max_position_embeddings=64,
                hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0,
                fused_attn=False)
    gen_cfg = {"max_dec_len": 4, "decoding_strategy": "greedy_search",
               "eos_token_id": 127}
    gpt = GPTModel(**mcfg)
    model = GPTForGeneration(gpt, gen_cfg)
    out_dir = str(tmp_path / "exported")
    export_inference_model(model, mcfg, out_dir,
                           extra={"generation": gen_cfg})

    engine = InferenceEngine(out_dir, mp_degree=1)
    ids = torch.randint(0, 127, (1, 6))
    ref = model(ids)
    out = engine.predict(ids)
    assert torch.equal(ref, out.cpu())


def test_profiler_guard_disabled_and_enabled(tmp_path):
    from paddlefleetx_amd.utils.profiler import ProfilerGuard
    g = ProfilerGuard(None)
    g.step()
    g.stop_and_summary()  # inert

    g2 = ProfilerGuard({"enable": True, "scheduler": [0, 2],
                        "profiler_log": str(tmp_path / "plog")})
    x = torch.randn(8, 8)
    for _ in range(3):
        (x @ x).sum()
        g2.step()
    g2.stop_and_summary()


def test_inference_engine_model_generic_vit(tmp_path):
    """Non-GPT family export -> model-generic InferenceEngine rebuild
    (reference inference_engine.py:144-271 loads any exported program)."""
    from paddlefleetx_amd.parallel.env import set_hcg
    from paddlefleetx_amd.parallel.topology import HybridTopology
    import torch.distributed as dist
    if not dist.is_initialized():
        set_hcg(HybridTopology())
    from paddlefleetx_amd.core.inference_engine import InferenceEngine
    from paddlefleetx_amd.utils.export import export_inference_model
    from paddlefleetx_amd.models import build_module

    mcfg = {"module": "GeneralClsModule", "name": "GeneralClsModule",
            "model": {"name": "ViT_tiny_patch16_224", "class_num": 10},
            "loss": {"train": {"name": "CELoss"}},
            "metric": {"name": "TopkAcc"}}
    module = build_module({"Model": mcfg})
    module.model = module.model.float()
    out_dir = str(tmp_path / "vit_exported")
    export_inference_model(module.model, mcfg, out_dir)
    engine = InferenceEngine(out_dir, mp_degree=1)
    x = torch.randn(2, 3, 224, 224)
    with torch.no_grad():
        ref = module.model(x)
    out = engine.predict(x)
    assert torch.allclose(ref, out.cpu(), atol=1e-5)
