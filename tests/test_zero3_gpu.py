"""ZeRO-3 on hardware: 175B-scale-model memory validation + prefetch
overlap correctness.

The sizing test runs a LAYER SLICE of the BASELINE "GPT-3 175B TP8
sharding-stage3" config at the real hidden size (12288, TP8 shard ->
per-rank 1536-wide attention / 6144-wide FFN... here we keep full-width
layers and shard the PARAMS 8-ways with the fake-world rehearsal mode,
which reproduces stage-3's allocation behavior exactly on one GPU), and
extrapolates the 96-layer footprint against 288 GB HBM.
"""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _layer_stack(num_layers, hidden, heads, dtype):
    from paddlefleetx_amd.models.gpt.model import TransformerDecoderLayer
    import torch.nn as nn

    class Stack(nn.Module):
        def __init__(self):
            super().__init__()
            self.layers = nn.ModuleList([
                TransformerDecoderLayer(hidden, heads, 4 * hidden,
                                        fused_attn=True, dtype=dtype)
                for _ in range(num_layers)])

        def forward(self, x):
            for l in self.layers:
                x = l(x)
            return x

    return Stack()


def test_zero3_175b_scale_model_memory():
    from paddlefleetx_amd.parallel.env import set_hcg
    from paddlefleetx_amd.parallel.topology import HybridTopology
    import torch.distributed as dist
    if not dist.is_initialized():
        set_hcg(HybridTopology())
    from paddlefleetx_amd.parallel.zero3 import (GroupShardedStage3,
                                                 Stage3AdamW)
    torch.cuda.empty_cache()
    torch.cuda.reset_peak_memory_stats()
    H, HEADS, L, WORLD = 12288, 96, 4, 8  # 175B layer geometry
    with torch.device("cuda"):
        model = _layer_stack(L, H, HEADS, torch.bfloat16)
    layer_params = sum(p.numel() for p in model.layers[0].parameters())
    wrapped = GroupShardedStage3(model, group=None, fake_world=WORLD)
    opt = Stage3AdamW(wrapped, lr=1e-4)

    # resident per layer must be ~1/8 of the full layer (plus grad shard)
    resident = torch.cuda.memory_allocated()
    per_layer_bytes = layer_params * 2  # bf16
    # shard(2B) + grad shard(2B) + fp32 master/m/v (12B) on 1/8
    expected_layer_resident = per_layer_bytes / WORLD * (2 + 12) / 2
    assert resident < L * expected_layer_resident * 1.6 + 2 ** 30, \
        (resident / 2 ** 30, L * expected_layer_resident / 2 ** 30)

    x = torch.randn(1, 2048, H, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    y = wrapped(x)
    y.float().pow(2).mean().backward()
    opt.reduce_and_step()
    peak = torch.cuda.max_memory_allocated()

    # extrapolate the full 96-layer model: resident states scale with L,
    # transient (one gathered layer + activations) does not
    resident_96 = resident / L * 96
    transient = peak - resident
    total_96 = resident_96 + transient
    hbm = torch.cuda.get_device_properties(0).total_memory
    print(f"[sizing] resident/4L={resident/2**30:.1f} GiB, "
          f"peak={peak/2**30:.1f} GiB, 96L on-device projection="
          f"{total_96/2**30:.1f} GiB of {hbm/2**30:.0f} GiB")
    # MEASURED sizing fact: with fp32 optimizer states ON DEVICE the
    # 96-layer projection (~340 GiB) exceeds 288 GB HBM — the 175B
    # single-node config REQUIRES the CPU-offload path below (or a
    # second sharding axis across nodes)
    assert total_96 > hbm, "sizing assumption changed — revisit"
    # storage really released after the step
    for u in wrapped.units:
        if u.name != "<rest>":
            assert not u.live

    # ---- offload variant: fp32 master/m/v in pinned host memory ----
    del opt
    torch.cuda.empty_cache()
    torch.cuda.reset_peak_memory_stats()
    opt2 = Stage3AdamW(wrapped, lr=1e-4, offload=True)
    resident_off = torch.cuda.memory_allocated()
    y = wrapped(x)
    y.float().pow(2).mean().backward()
    opt2.reduce_and_step()
    peak_off = torch.cuda.max_memory_allocated()
    total_96_off = resident_off / L * 96 + (peak_off - resident_off)
    print(f"[sizing+offload] resident/4L={resident_off/2**30:.1f} GiB, "
          f"peak={peak_off/2**30:.1f} GiB, 96L projection="
          f"{total_96_off/2**30:.1f} GiB of {hbm/2**30:.0f} GiB")
    assert total_96_off < hbm * 0.75,         (total_96_off / 2 ** 30, hbm / 2 ** 30)


def test_zero3_prefetch_matches_sync():
    """Prefetched (side-stream) gathers produce the same outputs as the
    synchronous path."""
    from paddlefleetx_amd.parallel.env import set_hcg
    from paddlefleetx_amd.parallel.topology import HybridTopology
    import torch.distributed as dist
    if not dist.is_initialized():
        set_hcg(HybridTopology())
    from paddlefleetx_amd.parallel.zero3 import GroupShardedStage3
    torch.manual_seed(3)
    with torch.device("cuda"):
        m1 = _layer_stack(3, 256, 4, torch.bfloat16)
    m2 = _layer_stack(3, 256, 4, torch.bfloat16).cuda()
    m2.load_state_dict(m1.state_dict())
    w1 = GroupShardedStage3(m1, group=None, prefetch=True)
    w2 = GroupShardedStage3(m2, group=None, prefetch=False)
    x = torch.randn(2, 64, 256, device="cuda", dtype=torch.bfloat16)
    y1 = w1(x)
    y2 = w2(x)
    torch.cuda.synchronize()
    assert torch.equal(y1, y2)
    y1.float().sum().backward()
    torch.cuda.synchronize()
