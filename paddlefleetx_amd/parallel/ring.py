"""Ring attention over the context-parallel group (long-context path).

Alternative CP backend to Ulysses (parallel/cp.py): instead of trading
sequence for heads with an all-to-all, each rank keeps its sequence
shard of Q resident and the K/V shards travel around the ring
(cp p2p hops on xGMI), merged with a numerically stable
log-sum-exp combine. Per-step traffic is 2·B·Sl·H·D bytes to ONE
neighbour — on the 8×MI355X mesh each hop has a dedicated xGMI link, and
compute of block j overlaps the in-flight transfer of block j+1.

The reference has no ring attention (SURVEY §5 long-context: "No ring
attention ... max context bounded by max_position_embeddings") — this is
an MI355X-native extension: 288 GB HBM + ring K/V makes 32k+ context
feasible at 6.7B.

Causality with sequential sharding: rank r's Q rows are globally after
every row held by ranks < r, so a visiting block from rank s needs
  s < r : full (non-causal) attention
  s == r: causal attention (the local diagonal block)
  s > r : skipped entirely (still rotated to keep the ring in step).

Block math reuses the gfx950 flash kernels unchanged: forward merges
per-block (o_j, lse_j); backward recomputes each block's P from the
GLOBAL lse (so per-block dq/dk/dv sum to the exact full-attention
grads), with the (dk, dv) accumulators travelling with their K/V block —
after cp rotations they arrive back at the owning rank.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.distributed as dist

from paddlefleetx_amd.parallel.env import get_hcg

__all__ = ["RingAttention", "ring_attention"]


def _block_fwd(q, k, v, causal, scale):
    from paddlefleetx_amd.ops import _reference as ref
    from paddlefleetx_amd.ops import hip_ext, use_hip
    if use_hip(q):
        return hip_ext().attn_fwd(q, k, v, causal, scale)
    return ref.attention_fwd(q, k, v, causal, scale)


def _block_bwd(do, q, k, v, o, lse, causal, scale):
    from paddlefleetx_amd.ops import _reference as ref
    from paddlefleetx_amd.ops import hip_ext, use_hip
    if use_hip(q):
        return hip_ext().attn_bwd(do, q, k, v, o, lse, causal, scale)
    return ref.attention_bwd(do, q, k, v, o, lse, causal, scale)


def _rotate_start(tensors, g):
    """Post the ring exchange (send to next, recv from previous) and
    return (requests, out_tensors, sent_refs) WITHOUT waiting — the
    caller overlaps block compute with the transfer and calls
    `_rotate_wait` when it needs the data. `sent_refs` keeps the
    contiguous send copies alive until completion."""
    ranks = g.ranks
    me = g.rank
    nxt = ranks[(me + 1) % g.world_size]
    prv = ranks[(me - 1) % g.world_size]
    outs = [torch.empty_like(t) for t in tensors]
    sends = [t.contiguous() for t in tensors]
    ops = []
    for t, o in zip(sends, outs):
        ops.append(dist.P2POp(dist.isend, t, nxt))
        ops.append(dist.P2POp(dist.irecv, o, prv))
    return dist.batch_isend_irecv(ops), outs, sends


def _rotate_wait(reqs):
    for r in reqs:
        r.wait()


def _halves(rank: int, cp: int):
    """Zigzag layout: rank r's local sequence = global half-chunks
    (r, 2cp-1-r) concatenated — balances the causal triangle so every
    rank attends the same number of key blocks."""
    return (rank, 2 * cp - 1 - rank)


def _pair_mode(gq: int, gk: int):
    """full / causal / skip relation between two global half-chunks."""
    if gq > gk:
        return "full"
    if gq == gk:
        return "causal"
    return None


class _RingAttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale, zigzag):
        g = get_hcg().get_context_parallel_group()
        cp = g.world_size
        r = g.rank if cp > 1 else 0
        Sl = q.shape[2]
        assert not zigzag or Sl % 2 == 0

        def merge(acc, o_j, lse_j):
            if acc[0] is None:
                return [o_j.float(), lse_j]
            o_acc, lse_acc = acc
            lse_new = torch.logaddexp(lse_acc, lse_j)
            w_old = torch.exp(lse_acc - lse_new)[..., None]
            w_new = torch.exp(lse_j - lse_new)[..., None]
            return [o_acc * w_old + o_j.float() * w_new, lse_new]

        cur_k, cur_v = k, v
        if not zigzag:
            acc = [None, None]
            for j in range(cp):
                src = (r - j) % cp
                # post block j+1's transfer BEFORE computing block j so
                # the hop overlaps the flash kernel
                pend = _rotate_start([cur_k, cur_v], g) if j < cp - 1 \
                    else None
                if src <= r:
                    acc = merge(acc, *_block_fwd(q, cur_k, cur_v, src == r,
                                                 scale))
                if pend is not None:
                    _rotate_wait(pend[0])
                    cur_k, cur_v = pend[1]
            out, lse = acc[0].to(q.dtype), acc[1]
        else:
            h = Sl // 2
            qh = (q[:, :, :h], q[:, :, h:])
            accs = [[None, None], [None, None]]
            for j in range(cp):
                src = (r - j) % cp
                pend = _rotate_start([cur_k, cur_v], g) if j < cp - 1 \
                    else None
                for qi, gq in enumerate(_halves(r, cp)):
                    for ki, gk in enumerate(_halves(src, cp)):
                        mode = _pair_mode(gq, gk)
                        if mode is None:
                            continue
                        kk = cur_k[:, :, ki * h:(ki + 1) * h]
                        vv = cur_v[:, :, ki * h:(ki + 1) * h]
                        accs[qi] = merge(accs[qi], *_block_fwd(
                            qh[qi].contiguous(), kk.contiguous(),
                            vv.contiguous(), mode == "causal", scale))
                if pend is not None:
                    _rotate_wait(pend[0])
                    cur_k, cur_v = pend[1]
            out = torch.cat([accs[0][0], accs[1][0]], dim=2).to(q.dtype)
            lse = torch.cat([accs[0][1], accs[1][1]], dim=2)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.scale = scale
        ctx.zigzag = zigzag
        return out

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        g = get_hcg().get_context_parallel_group()
        cp = g.world_size
        r = g.rank if cp > 1 else 0
        do = do.contiguous()

        dq_acc = torch.zeros_like(q, dtype=torch.float32)
        cur_k, cur_v = k, v
        cur_dk = torch.zeros_like(k, dtype=torch.float32)
        cur_dv = torch.zeros_like(v, dtype=torch.float32)
        dkv_pend = None  # in-flight (dk, dv) hop from the previous step
        h = q.shape[2] // 2
        for j in range(cp):
            src = (r - j) % cp
            # k/v for the NEXT step travel during this step's block
            # backward. The (dk, dv) accumulators posted at the end of
            # the PREVIOUS step are also still in flight here: the block
            # backward doesn't read them, so the wait is deferred until
            # just before the += below and the hop overlaps compute too.
            kv_pend = _rotate_start([cur_k, cur_v], g) if cp > 1 else None
            blocks = []  # (dq_slice, dk_slice, dq_j, dk_j, dv_j)
            if not ctx.zigzag:
                if src <= r:
                    blocks.append((slice(None), slice(None),
                                   *_block_bwd(do, q, cur_k, cur_v, o,
                                               lse, src == r, ctx.scale)))
            else:
                for qi, gq in enumerate(_halves(r, cp)):
                    qs = slice(qi * h, (qi + 1) * h)
                    for ki, gk in enumerate(_halves(src, cp)):
                        mode = _pair_mode(gq, gk)
                        if mode is None:
                            continue
                        ks = slice(ki * h, (ki + 1) * h)
                        blocks.append((qs, ks, *_block_bwd(
                            do[:, :, qs].contiguous(),
                            q[:, :, qs].contiguous(),
                            cur_k[:, :, ks].contiguous(),
                            cur_v[:, :, ks].contiguous(),
                            o[:, :, qs].contiguous(),
                            lse[:, :, qs].contiguous(),
                            mode == "causal", ctx.scale)))
            if dkv_pend is not None:
                _rotate_wait(dkv_pend[0])
                cur_dk, cur_dv = dkv_pend[1]
                dkv_pend = None
            for qs, ks, dq_j, dk_j, dv_j in blocks:
                dq_acc[:, :, qs] += dq_j.float()
                cur_dk[:, :, ks] += dk_j.float()
                cur_dv[:, :, ks] += dv_j.float()
            if cp > 1:
                # after cp hops each (k, v, dk, dv) quartet is back at
                # its owning rank; the last dk/dv hop is awaited below
                _rotate_wait(kv_pend[0])
                cur_k, cur_v = kv_pend[1]
                dkv_pend = _rotate_start([cur_dk, cur_dv], g)
        if dkv_pend is not None:
            _rotate_wait(dkv_pend[0])
            cur_dk, cur_dv = dkv_pend[1]
        return (dq_acc.to(q.dtype), cur_dk.to(k.dtype), cur_dv.to(v.dtype),
                None, None)


def ring_attention(q, k, v, scale: Optional[float] = None,
                   zigzag: bool = False):
    """Causal ring attention; q,k,v [B, H, S/cp, D] sequence shards.
    zigzag=False: sequential sharding (rank r owns chunk r).
    zigzag=True: rank r owns half-chunks (r, 2cp-1-r) — every rank then
    attends the same number of key blocks, levelling the causal-triangle
    load imbalance of sequential sharding (slice batches with
    `zigzag_slice`)."""
    scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
    return _RingAttnFn.apply(q.contiguous(), k.contiguous(), v.contiguous(),
                             scale, zigzag)


def zigzag_slice(t: torch.Tensor, cp: int, rank: int,
                 dim: int = 1) -> torch.Tensor:
    """Take rank's zigzag shard (half-chunks rank and 2cp-1-rank) of a
    full-sequence tensor along `dim`."""
    chunks = t.chunk(2 * cp, dim=dim)
    return torch.cat([chunks[rank], chunks[2 * cp - 1 - rank]],
                     dim=dim).contiguous()


class RingAttention(torch.nn.Module):
    """Drop-in alternative to UlyssesAttention: q,k,v [B, S/cp, h, D] ->
    o [B, S/cp, h, D] (same interface/layout as parallel/cp.py)."""

    def __init__(self, scale: Optional[float] = None, causal: bool = True,
                 zigzag: bool = False):
        super().__init__()
        assert causal, "ring attention: causal only (GPT pretraining path)"
        self.scale = scale
        self.zigzag = zigzag

    def forward(self, q, k, v):
        q = q.permute(0, 2, 1, 3)  # [B, h, Sl, D]
        k = k.permute(0, 2, 1, 3)
        v = v.permute(0, 2, 1, 3)
        o = ring_attention(q, k, v, self.scale, zigzag=self.zigzag)
        return o.permute(0, 2, 1, 3)
