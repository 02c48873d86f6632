"""GPU end-to-end steps for the non-GPT families (ViT, ERNIE, MoE, T5,
MoCo, generation) — all on the gfx950 kernel path."""

import pytest
import torch

from paddlefleetx_amd.parallel.env import set_hcg
from paddlefleetx_amd.parallel.topology import HybridTopology

pytestmark = pytest.mark.gpu


@pytest.fixture(autouse=True)
def _env():
    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    set_hcg(HybridTopology())
    yield


def test_vit_base_step_uses_flash_kernel():
    from paddlefleetx_amd.models.vit import build_vit
    m = build_vit("ViT_base_patch16_224", class_num=10,
                  dtype=torch.bfloat16).cuda()
    x = torch.randn(2, 3, 224, 224, device="cuda", dtype=torch.bfloat16)
    y = m(x)
    assert y.shape == (2, 10)
    y.float().sum().backward()
    assert m.blocks[0].attn.qkv.weight.grad is not None
    # head_dim 64 -> the non-causal gfx950 kernel must be the live path
    from paddlefleetx_amd.ops import has_hip_ext
    assert has_hip_ext()


def test_vit_matches_cpu_reference_small():
    """bf16 GPU ViT forward ~= fp32 CPU forward (loose tol)."""
    from paddlefleetx_amd.models.vit import ViT
    torch.manual_seed(0)
    m32 = ViT(img_size=32, patch_size=8, embed_dim=64, depth=2, num_heads=1,
              class_num=5, qkv_bias=True)  # head_dim 64 -> kernel path
    m16 = ViT(img_size=32, patch_size=8, embed_dim=64, depth=2, num_heads=1,
              class_num=5, qkv_bias=True)
    m16.load_state_dict(m32.state_dict())
    m16 = m16.to("cuda", torch.bfloat16).eval()
    m32.eval()
    x = torch.randn(2, 3, 32, 32)
    ref = m32(x)
    out = m16(x.to("cuda", torch.bfloat16)).float().cpu()
    assert torch.allclose(ref, out, atol=0.15, rtol=0.1), \
        (ref - out).abs().max()


def test_ernie_pretrain_step_gpu():
    from paddlefleetx_amd.models.ernie import (ErnieForPretraining,
                                               ErnieModel,
                                               ErniePretrainingCriterion)
    m = ErnieForPretraining(ErnieModel(
        vocab_size=1000, hidden_size=256, num_hidden_layers=2,
        num_attention_heads=4, intermediate_size=512,
        max_position_embeddings=128, hidden_dropout_prob=0.0,
        attention_probs_dropout_prob=0.0,
        dtype=torch.bfloat16)).cuda()
    ids = torch.randint(0, 1000, (4, 128), device="cuda")
    tt = torch.zeros_like(ids)
    labels = torch.randint(0, 1000, (4, 128), device="cuda")
    nsp = torch.randint(0, 2, (4,), device="cuda")
    pred, rel = m(ids, tt)
    mlm, nsp_l = ErniePretrainingCriterion()(pred, rel, labels, nsp)
    (mlm + nsp_l).backward()
    assert torch.isfinite(mlm) and torch.isfinite(nsp_l)


def test_moe_layer_gpu_matches_loop():
    from paddlefleetx_amd.models.moe import MoELayer, NaiveGate
    torch.manual_seed(3)
    layer = MoELayer(64, 128, num_experts=4,
                     gate=NaiveGate(64, 4, top_k=2)).cuda()
    x = torch.randn(32, 64, device="cuda")
    y = layer(x)
    layer.gate.eval()
    idx, score = layer.gate(x)
    ref = torch.zeros_like(x)
    with torch.no_grad():
        for t in range(32):
            for k in range(2):
                ref[t] += score[t, k] * \
                    layer.experts[int(idx[t, k])](x[t:t + 1])[0]
    assert torch.allclose(y, ref, atol=1e-4), (y - ref).abs().max()


def test_t5_encoder_gpu():
    from paddlefleetx_amd.models.t5 import T5EncoderModel
    m = T5EncoderModel(vocab_size=512, d_model=128, d_kv=32, d_ff=256,
                       num_layers=2, num_heads=4, dropout_rate=0.0).cuda()
    ids = torch.randint(0, 512, (2, 64), device="cuda")
    out = m(ids)
    out.sum().backward()
    assert out.shape == (2, 64, 128)


def test_generation_greedy_gpu():
    from paddlefleetx_amd.models.gpt.generation import GPTForGeneration
    from paddlefleetx_amd.models.gpt.model import GPTModel
    torch.manual_seed(0)
    gpt = GPTModel(vocab_size=512, hidden_size=256, num_layers=2,
                   num_attention_heads=2, max_position_embeddings=128,
                   hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0,
                   dtype=torch.bfloat16).cuda()
    gen = GPTForGeneration(gpt, {"max_dec_len": 8,
                                 "decoding_strategy": "greedy_search",
                                 "eos_token_id": 511})
    ids = torch.randint(0, 511, (2, 16), device="cuda")
    out1 = gen(ids)
    out2 = gen(ids)
    assert torch.equal(out1, out2)
    assert out1.shape[1] <= 8


def test_topp_generation_gpu():
    from paddlefleetx_amd.models.gpt.generation import GPTForGeneration
    from paddlefleetx_amd.models.gpt.model import GPTModel
    gpt = GPTModel(vocab_size=512, hidden_size=256, num_layers=1,
                   num_attention_heads=2, max_position_embeddings=64,
                   hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0,
                   dtype=torch.bfloat16).cuda()
    gen = GPTForGeneration(gpt, {"max_dec_len": 6, "top_p": 0.8,
                                 "use_topp_sampling": True,
                                 "eos_token_id": 511})
    out = gen(torch.randint(0, 511, (2, 8), device="cuda"))
    assert out.shape[0] == 2 and out.shape[1] <= 6


def test_engine_hipgraph_matches_eager():
    dev = "cuda"
    """Engine.hip_graph: captured micro-step replay == eager execution
    (same losses and updated params over several steps)."""
    from paddlefleetx_amd.parallel.env import set_hcg, set_seed
    from paddlefleetx_amd.parallel.topology import HybridTopology
    import torch.distributed as dist
    if not dist.is_initialized():
        set_hcg(HybridTopology())
    from paddlefleetx_amd.models import build_module
    from paddlefleetx_amd.core import EagerEngine
    from paddlefleetx_amd.utils.config import get_config
    import os
    repo = os.path.join(os.path.dirname(__file__), "..")

    def run(graph):
        set_seed(1234)
        cfg = get_config(os.path.join(
            repo, "paddlefleetx_amd/configs/nlp/gpt/"
            "pretrain_gpt_345M_single_card.yaml"),
            overrides=["Model.hidden_size=256", "Model.num_layers=2",
                       "Model.num_attention_heads=4",
                       "Model.vocab_size=1024",
                       "Model.max_position_embeddings=128",
                       "Model.hidden_dropout_prob=0.0",
                       "Model.attention_probs_dropout_prob=0.0",
                       "Global.micro_batch_size=2",
                       "Global.local_batch_size=4",
                       f"Engine.hip_graph={graph}"])
        torch.manual_seed(7)
        module = build_module(cfg)
        engine = EagerEngine(cfg, module)
        losses = []
        for s in range(6):
            torch.manual_seed(100 + s)
            batch = (torch.randint(0, 1024, (4, 128), device=dev),
                     torch.arange(128, device=dev).repeat(4, 1),
                     torch.randint(0, 1024, (4, 128), device=dev),
                     torch.ones(4, 128, device=dev))
            losses.append(float(engine._fit_impl(batch)))
        if graph:
            assert engine._graph is not None, "graph never captured"
        params = [b.model_flat.clone() for b in engine.optimizer.buckets]
        return losses, params

    l0, p0 = run(False)
    l1, p1 = run(True)
    for a, b in zip(l0, l1):
        assert abs(a - b) < 5e-3, (l0, l1)
    for a, b in zip(p0, p1):
        assert torch.allclose(a.float(), b.float(), atol=1e-2), \
            (a - b).float().abs().max()


def test_recompute_dropout_exact_on_flash_path():
    """GPU analogue of the checkpoint_rng_context fix: the flash
    attention dropout SEED is drawn from the mp-tracker stream, so the
    recompute re-forward must replay the same seed (and the same
    in-kernel Philox masks). Grads exact vs no recompute."""
    from paddlefleetx_amd.models.gpt.model import (GPTForPretraining,
                                                   GPTModel,
                                                   GPTPretrainingCriterion)
    from paddlefleetx_amd.parallel.env import set_seed
    grads = {}
    for rec in (False, True):
        set_seed(1234)
        torch.manual_seed(3)
        m = GPTForPretraining(GPTModel(
            vocab_size=512, hidden_size=256, num_layers=2,
            num_attention_heads=2, max_position_embeddings=128,
            fused_attn=True, hidden_dropout_prob=0.1,
            attention_probs_dropout_prob=0.1,
            use_recompute=rec, recompute_granularity="full",
            dtype=torch.bfloat16)).cuda()
        m.train()
        torch.manual_seed(7)
        tokens = torch.randint(0, 512, (2, 128), device="cuda")
        labels = torch.randint(0, 512, (2, 128), device="cuda")
        loss = GPTPretrainingCriterion()(m(tokens), labels,
                                         torch.ones(2, 128, device="cuda"))
        loss.backward()
        grads[rec] = m.gpt.layers[0].attn.qkv.weight.grad.float().clone()
    assert torch.equal(grads[False], grads[True])
