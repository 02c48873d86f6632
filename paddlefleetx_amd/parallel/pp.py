"""Native 1F1B pipeline parallelism over RCCL p2p.

The reference gets its pipeline engine from Paddle (`PipelineLayer` +
`forward_backward_pipeline`, hybrid_model.py:1055-1206 /
eager_engine.py:507-517). This is a from-scratch implementation:

- LayerDesc/SharedLayerDesc model description, `layer:` seg_method
  partitioning (uniform split of the matched layers, first/last aux
  layers pinned to the edge stages).
- Non-interleaved 1F1B schedule: warmup fwd, steady 1F1B, cooldown bwd;
  micro-batch count = accumulate_steps.
- Activation/grad exchange via dist.batch_isend_irecv on the pp axis
  (xGMI p2p on a single node), static [micro_b, s, h] bf16 buffers.
- Tied first/last embedding: weight broadcast at init + grad allreduce
  over the {first, last} embedding group after the schedule
  (hybrid_model.py:1116/1169 SharedLayerDesc semantics).
"""

from __future__ import annotations

import re
from typing import Any, Callable, Dict, List, Optional, Tuple

import torch
import torch.distributed as dist
import torch.nn as nn

from paddlefleetx_amd.parallel.env import get_hcg
from paddlefleetx_amd.utils.log import logger

__all__ = ["LayerDesc", "SharedLayerDesc", "PipelineModule"]


def _get_dotted(obj, path: str):
    for p in path.split("."):
        obj = getattr(obj, p)
    return obj


def _set_dotted(obj, path: str, value):
    parts = path.split(".")
    for p in parts[:-1]:
        obj = getattr(obj, p)
    setattr(obj, parts[-1], value)


class LayerDesc:
    def __init__(self, layer_cls, *args, **kwargs):
        self.layer_cls = layer_cls
        self.args = args
        self.kwargs = kwargs

    def build(self) -> nn.Module:
        return self.layer_cls(*self.args, **self.kwargs)

    @property
    def name(self):
        return self.layer_cls.__name__


class SharedLayerDesc(LayerDesc):
    """Layer whose `shared_weight_attr` parameter is tied across the stages
    that hold a desc with the same key (reference SharedLayerDesc)."""

    def __init__(self, key: str, layer_cls, *args,
                 forward_func: Optional[Callable] = None,
                 shared_weight_attr: str = "weight", **kwargs):
        super().__init__(layer_cls, *args, **kwargs)
        self.key = key
        self.forward_func = forward_func
        self.shared_weight_attr = shared_weight_attr


class _FuncLayer(nn.Module):
    """Wraps a SharedLayerDesc's alternate forward (e.g. tied logits)."""

    def __init__(self, module: nn.Module, fn: Callable):
        super().__init__()
        self.inner = module
        self.fn = fn

    def forward(self, *args):
        return self.fn(self.inner, *args)


class PipelineModule(nn.Module):
    """Builds only this pp stage's layers from the full desc list."""

    def __init__(self, descs: List[LayerDesc], loss_shape_hint=None,
                 seg_method: str = "uniform", act_dtype=torch.bfloat16):
        super().__init__()
        hcg = get_hcg()
        self.hcg = hcg
        self.pp_rank = hcg.get_pipe_parallel_rank()
        self.pp_size = hcg.get_pipe_parallel_world_size()
        self.act_dtype = act_dtype
        self.descs = descs

        start, end = self._partition(descs, seg_method)
        self.stage_start, self.stage_end = start, end
        self.layers = nn.ModuleList()
        self._shared_keys: Dict[str, Tuple[nn.Module, str]] = {}
        for i in range(start, end):
            d = descs[i]
            m = d.build()
            if isinstance(d, SharedLayerDesc):
                if d.key in self._shared_keys:
                    # reuse the first-built module's weight (same stage)
                    first, attr = self._shared_keys[d.key]
                    _set_dotted(m, attr, _get_dotted(first, attr))
                else:
                    self._shared_keys[d.key] = (m, d.shared_weight_attr)
                if d.forward_func is not None:
                    m = _FuncLayer(m, d.forward_func)
            self.layers.append(m)
        logger.info(f"pp stage {self.pp_rank}/{self.pp_size}: layers "
                    f"[{start}, {end}) of {len(descs)}")
        self._shared_groups = None  # built lazily after dist init

    # ------------------------------------------------------------------
    def _partition(self, descs, seg_method) -> Tuple[int, int]:
        n = len(descs)
        if seg_method.startswith("layer:"):
            pat = seg_method.split(":", 1)[1]
            idx = [i for i, d in enumerate(descs)
                   if re.match(pat, d.name)]
            per = len(idx) // self.pp_size
            rem = len(idx) % self.pp_size
            # distribute matched layers; leading descs join stage 0,
            # trailing descs join the last stage
            counts = [per + (1 if s < rem else 0) for s in range(self.pp_size)]
            bounds = [0]
            for c in counts:
                bounds.append(bounds[-1] + c)
            first_matched = idx[0]
            lo = first_matched + bounds[self.pp_rank]
            hi = first_matched + bounds[self.pp_rank + 1]
            if self.pp_rank == 0:
                lo = 0
            if self.pp_rank == self.pp_size - 1:
                hi = n
            return lo, hi
        per = n // self.pp_size
        rem = n % self.pp_size
        counts = [per + (1 if s < rem else 0) for s in range(self.pp_size)]
        lo = sum(counts[:self.pp_rank])
        return lo, lo + counts[self.pp_rank]

    # ------------------------------------------------------------------
    def _build_shared_groups(self):
        if self._shared_groups is not None:
            return
        self._shared_groups = []
        hcg = self.hcg
        if self.pp_size == 1 or not dist.is_initialized():
            return
        # group of global ranks {stage0, stageN-1} within this pp slice
        r_first = hcg._pp_global_rank(0)
        r_last = hcg._pp_global_rank(self.pp_size - 1)
        # every rank must execute the same new_group calls: enumerate all
        # (dp, sharding, mp) coordinates deterministically
        import itertools
        mine = None
        for dp_i, sd_i, mp_i in itertools.product(
                range(hcg.dp_degree), range(hcg.sharding_degree),
                range(hcg.mp_degree)):
            def grank(pp_i):
                return ((pp_i * hcg.dp_degree + dp_i) * hcg.sharding_degree
                        + sd_i) * hcg.mp_degree + mp_i
            ranks = [grank(0), grank(self.pp_size - 1)]
            g = dist.new_group(ranks=ranks)
            if hcg.global_rank in ranks:
                mine = (g, ranks)
        if mine and (self.pp_rank == 0 or self.pp_rank == self.pp_size - 1):
            self._shared_groups.append(mine)
            # initial weight sync: broadcast stage-0 copy
            for key, (m, attr) in self._shared_keys.items():
                w = _get_dotted(m, attr)
                dist.broadcast(w.data, src=mine[1][0], group=mine[0])

    def sync_shared_grads(self):
        """Allreduce tied-embedding grads over the {first,last} group."""
        if not self._shared_groups:
            return
        g, ranks = self._shared_groups[0]
        for key, (m, attr) in self._shared_keys.items():
            w = _get_dotted(m, attr)
            buf = getattr(w, "main_grad", None)
            if buf is None:
                buf = w.grad
            if buf is not None:
                dist.all_reduce(buf, group=g)

    # ------------------------------------------------------------------
    def forward(self, *inputs):
        """Run the local stage's layers (single-stage semantics)."""
        x = inputs
        for layer in self.layers:
            x = layer(*x) if isinstance(x, tuple) else layer(x)
        return x

    # ------------------------------------------------------------------
    # 1F1B schedule
    # ------------------------------------------------------------------
    def _p2p(self, ops):
        reqs = dist.batch_isend_irecv(ops)
        for r in reqs:
            r.wait()

    def _comm_shape(self, micro_b, seq, hidden):
        return (micro_b, seq, hidden)

    def forward_backward_pipeline(self, batch, loss_fn, accumulate_steps: int,
                                  scale: float = 1.0) -> torch.Tensor:
        """batch = (tokens, position_ids, labels, loss_mask) full local batch."""
        self._build_shared_groups()
        hcg = self.hcg
        pp_group = hcg.get_pipe_parallel_group()
        prev = hcg.pp_prev_rank()
        nxt = hcg.pp_next_rank()
        M = accumulate_steps
        device = next(self.parameters()).device

        tokens, position_ids, labels, loss_mask = batch
        micro_b = tokens.shape[0] // M
        seq = tokens.shape[1]
        # hidden size from config stashed on the module
        hidden = self.hidden_size
        shape = self._comm_shape(micro_b, seq, hidden)

        def micro(t, i):
            return t[i * micro_b:(i + 1) * micro_b]

        is_first = self.pp_rank == 0
        is_last = self.pp_rank == self.pp_size - 1

        def recv_forward():
            if is_first:
                return None
            buf = torch.empty(shape, dtype=self.act_dtype, device=device)
            self._p2p([dist.P2POp(dist.irecv, buf, prev)])
            buf.requires_grad_(True)
            return buf

        def send_forward(out):
            if not is_last:
                self._p2p([dist.P2POp(dist.isend, out.detach().contiguous(), nxt)])

        def recv_backward():
            if is_last:
                return None
            buf = torch.empty(shape, dtype=self.act_dtype, device=device)
            self._p2p([dist.P2POp(dist.irecv, buf, nxt)])
            return buf

        def send_backward(in_grad):
            if not is_first and in_grad is not None:
                self._p2p([dist.P2POp(dist.isend, in_grad.contiguous(), prev)])

        def send_forward_recv_backward(out):
            if is_last:
                return None
            buf = torch.empty(shape, dtype=self.act_dtype, device=device)
            self._p2p([dist.P2POp(dist.isend, out.detach().contiguous(), nxt),
                       dist.P2POp(dist.irecv, buf, nxt)])
            return buf

        def send_backward_recv_forward(in_grad):
            if is_first:
                return None
            buf = torch.empty(shape, dtype=self.act_dtype, device=device)
            self._p2p([dist.P2POp(dist.isend, in_grad.contiguous(), prev),
                       dist.P2POp(dist.irecv, buf, prev)])
            buf.requires_grad_(True)
            return buf

        def forward_step(inp, i):
            if is_first:
                x = (micro(tokens, i), micro(position_ids, i))
            else:
                x = (inp,)
            out = self.forward(*x)
            if is_last:
                loss = loss_fn(out, micro(labels, i), micro(loss_mask, i))
                return loss
            return out

        def backward_step(inp, out, out_grad):
            if is_last:
                (out * (scale / M)).backward()
            else:
                out.backward(gradient=out_grad)
            return inp.grad if inp is not None else None

        num_warmup = min(self.pp_size - self.pp_rank - 1, M)
        num_steady = M - num_warmup
        inputs: List = []
        outputs: List = []
        losses: List[torch.Tensor] = []
        fwd_i = 0

        for _ in range(num_warmup):
            inp = recv_forward()
            out = forward_step(inp, fwd_i)
            send_forward(out)
            inputs.append(inp)
            outputs.append(out)
            fwd_i += 1

        inp = recv_forward() if num_steady > 0 else None
        for i in range(num_steady):
            out = forward_step(inp, fwd_i)
            fwd_i += 1
            if is_last:
                losses.append(out.detach())
                out_grad = None
            else:
                out_grad = send_forward_recv_backward(out)
            inputs.append(inp)
            outputs.append(out)
            inp0 = inputs.pop(0)
            out0 = outputs.pop(0)
            in_grad = backward_step(inp0, out0, out_grad)
            last_iter = (i == num_steady - 1)
            if last_iter:
                send_backward(in_grad)
                inp = None
            elif is_first:
                inp = None
            else:
                inp = send_backward_recv_forward(in_grad)

        for _ in range(num_warmup):
            inp0 = inputs.pop(0)
            out0 = outputs.pop(0)
            out_grad = recv_backward()
            in_grad = backward_step(inp0, out0, out_grad)
            send_backward(in_grad)

        self.sync_shared_grads()

        # average loss across micro-batches on the last stage, broadcast so
        # every rank can log it
        if is_last and losses:
            loss = torch.stack(losses).mean()
        else:
            loss = torch.zeros((), device=device)
        if dist.is_initialized() and self.pp_size > 1:
            src = hcg._pp_global_rank(self.pp_size - 1)
            dist.broadcast(loss, src=src, group=pp_group.group)
        return loss.cpu()

    @torch.no_grad()
    def eval_pipeline(self, batch, loss_fn, accumulate_steps: int):
        """Forward-only pipeline for evaluation."""
        self._build_shared_groups()
        hcg = self.hcg
        prev, nxt = hcg.pp_prev_rank(), hcg.pp_next_rank()
        M = accumulate_steps
        device = next(self.parameters()).device
        tokens, position_ids, labels, loss_mask = batch
        micro_b = tokens.shape[0] // M
        seq = tokens.shape[1]
        shape = self._comm_shape(micro_b, seq, self.hidden_size)
        is_first = self.pp_rank == 0
        is_last = self.pp_rank == self.pp_size - 1
        losses = []
        for i in range(M):
            if is_first:
                x = (tokens[i * micro_b:(i + 1) * micro_b],
                     position_ids[i * micro_b:(i + 1) * micro_b])
                out = self.forward(*x)
            else:
                buf = torch.empty(shape, dtype=self.act_dtype, device=device)
                self._p2p([dist.P2POp(dist.irecv, buf, prev)])
                out = self.forward(buf)
            if is_last:
                losses.append(loss_fn(out, labels[i * micro_b:(i + 1) * micro_b],
                                      loss_mask[i * micro_b:(i + 1) * micro_b]))
            else:
                self._p2p([dist.P2POp(dist.isend, out.contiguous(), nxt)])
        if is_last and losses:
            loss = torch.stack(losses).mean()
        else:
            loss = torch.zeros((), device=device)
        if dist.is_initialized() and self.pp_size > 1:
            dist.broadcast(loss, src=hcg._pp_global_rank(self.pp_size - 1),
                           group=hcg.get_pipe_parallel_group().group)
        return loss.cpu()
