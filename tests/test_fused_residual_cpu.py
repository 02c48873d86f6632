"""layernorm_residual autograd (fused add+LN) == unfused add + layernorm."""

import torch

from paddlefleetx_amd.ops import layernorm, layernorm_residual


def test_layernorm_residual_matches_unfused():
    torch.manual_seed(3)
    N, H = 12, 64
    a = torch.randn(N, H, requires_grad=True)
    r = torch.randn(N, H, requires_grad=True)
    w = torch.randn(H, requires_grad=True)
    b = torch.randn(H, requires_grad=True)

    y, s = layernorm_residual(a, r, w, b, 1e-5)
    loss = (y * y).sum() + (s * s * 0.5).sum()
    loss.backward()

    a2 = a.detach().clone().requires_grad_(True)
    r2 = r.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    s2 = a2 + r2
    y2 = layernorm(s2, w2, b2, 1e-5)
    ((y2 * y2).sum() + (s2 * s2 * 0.5).sum()).backward()

    assert torch.allclose(y, y2, atol=1e-5)
    assert torch.allclose(s, s2, atol=1e-6)
    for g1, g2 in [(a.grad, a2.grad), (r.grad, r2.grad),
                   (w.grad, w2.grad), (b.grad, b2.grad)]:
        assert torch.allclose(g1, g2, atol=1e-4), (g1 - g2).abs().max()


def test_layernorm_residual_in_decoder_layer():
    """The decoder layer with the fused ln2 site still matches a manual
    pre-LN computation."""
    from paddlefleetx_amd.parallel.env import set_hcg
    from paddlefleetx_amd.parallel.topology import HybridTopology
    import torch.distributed as dist
    if not dist.is_initialized():
        set_hcg(HybridTopology())
    from paddlefleetx_amd.models.gpt.model import TransformerDecoderLayer
    torch.manual_seed(5)
    layer = TransformerDecoderLayer(32, 4, 128, fused_attn=False,
                                    dtype=torch.float32)
    x = torch.randn(2, 8, 32, requires_grad=True)
    z = layer(x)
    # manual: pre-LN attention + pre-LN FFN
    h1 = layer.ln1(x)
    a, _ = layer.attn(h1)
    y = x + a
    z2 = y + layer.ffn(layer.ln2(y))
    assert torch.allclose(z, z2, atol=1e-5), (z - z2).abs().max()
    z.sum().backward()
    assert x.grad is not None and torch.isfinite(x.grad).all()
