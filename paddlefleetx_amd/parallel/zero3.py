"""ZeRO stage-3: parameter + gradient + optimizer-state sharding.

Reference: paddle `group_sharded_parallel(model, optimizer, level="p_g_os")`
consumed by ppfleetx (core/engine/eager_engine.py:281-307,
distributed/apis/strategy.py:37-70). Rebuilt natively on RCCL:

Each wrapped unit (by default every transformer layer plus the remainder)
flattens its params into one flat bf16 buffer sharded 1/N per rank.
Full-parameter storage exists only while the unit is live:
  - forward pre-hook: all-gather the unit's flat buffer (storage resize
    0 -> full), params become views;
  - forward post-hook: release (storage resize -> 0) — the autograd graph
    saves views of the same storage, so release is real;
  - backward: a pre-backward gather re-materializes the params, and a
    per-param gradient countdown reduce-scatters grads onto the shard and
    releases the full storage again.
The optimizer (Stage3AdamW) steps on the 1/N fp32 master shard with the
same fused HIP kernel as FusedAdamW.

MI355X sizing note: with 288 GB HBM per GPU, stage-3 is the 175B-scale
tool (BASELINE config "GPT-3 175B TP8 sharding-stage3"): bf16 params
350 GB shard 8-ways to 44 GB + 66 GB optimizer shard.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch
import torch.distributed as dist
import torch.nn as nn

from paddlefleetx_amd.ops import fused_adamw_flat
from paddlefleetx_amd.utils.log import logger

ALIGN = 64


class _Unit:
    """One gather/release granule: a module's own (direct+child) params."""

    def __init__(self, name: str, module: nn.Module,
                 params: List[Tuple[str, nn.Parameter]], group, rank: int,
                 world: int):
        self.name = name
        self.module = module
        self.group = group
        self.rank, self.world = rank, world
        self.params = [p for _, p in params]
        self.param_names = [n for n, _ in params]
        offs, total = [], 0
        for p in self.params:
            offs.append(total)
            total += (p.numel() + ALIGN - 1) // ALIGN * ALIGN
        total = (total + ALIGN * world - 1) // (ALIGN * world) * (ALIGN * world)
        self.offsets = offs
        self.numel = total
        self.shard_len = total // world
        device = self.params[0].device
        dtype = self.params[0].dtype
        # full flat buffer; storage released when the unit is not live
        self.flat = torch.zeros(total, dtype=dtype, device=device)
        for p, off in zip(self.params, offs):
            self.flat[off:off + p.numel()].copy_(p.data.reshape(-1))
            p.data = self.flat[off:off + p.numel()].view(p.shape)
        # persistent 1/N shard
        self.shard = self.flat[self.rank * self.shard_len:
                               (self.rank + 1) * self.shard_len].clone()
        self.grad_shard = torch.zeros_like(self.shard)
        self.live = True
        self._grads_pending = 0
        self._pending_evt = None

    # -- storage control ----------------------------------------------------
    def release(self):
        if not self.live:
            return
        self.flat.untyped_storage().resize_(0)
        self.live = False

    def gather(self):
        if self.live:
            return
        if self._pending_evt is not None:
            # async prefetch already in flight on the side stream: make
            # the compute stream wait instead of re-gathering
            torch.cuda.current_stream().wait_event(self._pending_evt)
            self._pending_evt = None
            self.live = True
            return
        elems = self.numel
        self.flat.untyped_storage().resize_(
            elems * self.flat.element_size())
        if self.world > 1 and dist.is_initialized():
            dist.all_gather_into_tensor(self.flat, self.shard,
                                        group=self.group)
        else:
            # single-process (incl. fake-world sizing rehearsal): replicate
            self.flat.view(self.world, self.shard_len).copy_(self.shard)
        self.live = True

    def gather_async(self, stream) -> None:
        """Prefetch the all-gather on a side HIP stream so it overlaps the
        previous unit's compute (reference group_sharded broadcast_overlap,
        eager_engine.py:303-307)."""
        if self.live or self._pending_evt is not None or stream is None:
            return
        cur = torch.cuda.current_stream()
        elems = self.numel
        self.flat.untyped_storage().resize_(
            elems * self.flat.element_size())
        evt0 = torch.cuda.Event()
        evt0.record(cur)
        with torch.cuda.stream(stream):
            stream.wait_event(evt0)
            if self.world > 1 and dist.is_initialized():
                dist.all_gather_into_tensor(self.flat, self.shard,
                                            group=self.group)
            else:
                self.flat.view(self.world, self.shard_len).copy_(self.shard)
            evt = torch.cuda.Event()
            evt.record(stream)
        self._pending_evt = evt

    def writeback_from_shard(self):
        """After an optimizer step on `shard`, broadcast into flat if live."""
        if self.live:
            if self.world > 1 and dist.is_initialized():
                dist.all_gather_into_tensor(self.flat, self.shard,
                                            group=self.group)
            else:
                self.flat.view(self.world, self.shard_len).copy_(self.shard)

    # -- gradients ----------------------------------------------------------
    def reduce_grads(self):
        """Reduce-scatter the unit's full grads onto grad_shard (+=)."""
        device = self.shard.device
        full = torch.zeros(self.numel, dtype=self.shard.dtype, device=device)
        for p, off in zip(self.params, self.offsets):
            if p.grad is not None:
                full[off:off + p.numel()].copy_(p.grad.reshape(-1))
                p.grad = None
        if self.world > 1 and dist.is_initialized():
            out = torch.empty_like(self.grad_shard)
            dist.reduce_scatter_tensor(out, full, group=self.group)
            self.grad_shard.add_(out)
        else:
            self.grad_shard.add_(
                full[self.rank * self.shard_len:
                     (self.rank + 1) * self.shard_len])


class GroupShardedStage3(nn.Module):
    """Wrap `model` for ZeRO-3 over `group`. Units = modules matching
    `unit_classes` (default: any module whose class name contains
    'DecoderLayer' or 'EncoderLayer' or 'Block'), plus one unit for the
    remaining params."""

    def __init__(self, model: nn.Module, group=None,
                 unit_classes: Tuple[str, ...] = ("DecoderLayer",
                                                  "EncoderLayer", "Block"),
                 prefetch: bool = True, fake_world: int = 0):
        super().__init__()
        self.model = model
        self._prefetch = prefetch and torch.cuda.is_available()
        self._side_stream = torch.cuda.Stream() if self._prefetch else None
        gi = group
        self.group = gi.group if hasattr(gi, "group") else gi
        self.rank = gi.rank if hasattr(gi, "rank") else (
            dist.get_rank(self.group) if dist.is_initialized() else 0)
        self.world = gi.world_size if hasattr(gi, "world_size") else (
            dist.get_world_size(self.group) if dist.is_initialized() else 1)
        if fake_world > 1 and self.world == 1:
            # single-process MEMORY rehearsal of an N-way sharded run:
            # shards are 1/N sized and gathers replicate — values are not
            # meaningful across "ranks" but allocation behavior is exact
            # (the 175B TP8-stage3 sizing check, BASELINE config #3)
            self.world = fake_world

        self.units: List[_Unit] = []
        claimed = set()
        for name, mod in model.named_modules():
            if any(c in type(mod).__name__ for c in unit_classes):
                ps = [(f"{name}.{pn}", p)
                      for pn, p in mod.named_parameters()
                      if p.requires_grad and id(p) not in claimed]
                if not ps:
                    continue
                for _, p in ps:
                    claimed.add(id(p))
                self.units.append(_Unit(name, mod, ps, self.group, self.rank,
                                        self.world))
        rest = [(n, p) for n, p in model.named_parameters()
                if p.requires_grad and id(p) not in claimed]
        if rest:
            self.units.append(_Unit("<rest>", model, rest, self.group,
                                    self.rank, self.world))
        self._param_unit: Dict[int, _Unit] = {}
        for u in self.units:
            for p in u.params:
                self._param_unit[id(p)] = u
        self._install_hooks()
        # release everything except <rest> (embeddings etc. used at edges)
        for u in self.units:
            if u.name != "<rest>":
                u.release()
        n = sum(u.numel for u in self.units)
        logger.info(f"ZeRO-3: {len(self.units)} units, {n/1e6:.1f}M params, "
                    f"shard 1/{self.world}")

    # -- hooks ---------------------------------------------------------------
    def _install_hooks(self):
        for u in self.units:
            if u.name == "<rest>":
                continue
            u.module.register_forward_pre_hook(self._make_pre(u))
            u.module.register_forward_hook(self._make_post(u))
        for u in self.units:
            for p in u.params:
                p.register_post_accumulate_grad_hook(self._make_grad_hook(u))

    def _make_pre(self, u: _Unit):
        def pre(mod, args):
            u.gather()
            if self._prefetch:
                # overlap the NEXT unit's all-gather with this unit's
                # forward compute
                i = self.units.index(u)
                if i + 1 < len(self.units) and \
                        self.units[i + 1].name != "<rest>":
                    self.units[i + 1].gather_async(self._side_stream)
            return None
        return pre

    def _make_post(self, u: _Unit):
        wrapper = self

        def post(mod, args, out):
            if not torch.is_grad_enabled() or not self.model.training:
                u.release()
                return out
            # re-gather before this unit's backward runs: hook the output
            u._grads_pending = len([p for p in u.params if p.requires_grad])

            def regather(_grad):
                u.gather()
                if self._prefetch:
                    # backward walks units in reverse: prefetch the
                    # PREVIOUS unit's params under this unit's backward
                    i = self.units.index(u)
                    if i - 1 >= 0 and self.units[i - 1].name != "<rest>":
                        self.units[i - 1].gather_async(self._side_stream)
                return _grad

            if torch.is_tensor(out):
                out.register_hook(regather)
            elif isinstance(out, (tuple, list)) and torch.is_tensor(out[0]):
                out[0].register_hook(regather)
            u.release()
            return out
        return post

    def _make_grad_hook(self, u: _Unit):
        def hook(p):
            u._grads_pending -= 1
            if u._grads_pending <= 0:
                u.reduce_grads()
                if u.name != "<rest>":
                    u.release()
        return hook

    def forward(self, *args, **kwargs):
        return self.model(*args, **kwargs)

    def __getattr__(self, name):
        try:
            return super().__getattr__(name)
        except AttributeError:
            return getattr(self.model, name)

    # -- for checkpointing ---------------------------------------------------
    def gather_full_params(self):
        for u in self.units:
            u.gather()

    def state_dict(self, *a, **k):
        self.gather_full_params()
        return self.model.state_dict(*a, **k)

    def load_state_dict(self, sd, strict: bool = True):
        """Gather, copy the full state in-place (params view the flat
        buffers, so the copy lands there), then refresh every unit's
        persistent shard from its slice of the loaded flat."""
        self.gather_full_params()
        ret = self.model.load_state_dict(sd, strict=strict)
        for u in self.units:
            u.shard.copy_(u.flat[u.rank * u.shard_len:
                                 (u.rank + 1) * u.shard_len])
        return ret


class Stage3AdamW:
    """AdamW on the per-unit fp32 master shards (fused HIP kernel)."""

    def __init__(self, wrapper: GroupShardedStage3, lr: float = 1e-4,
                 beta1: float = 0.9, beta2: float = 0.95,
                 epsilon: float = 1e-8, weight_decay: float = 0.01,
                 offload: bool = False, **unused):
        self.w = wrapper
        self.lr = lr
        self.beta1, self.beta2, self.eps = beta1, beta2, epsilon
        self.weight_decay = weight_decay
        self._step = 0
        # CPU offload (reference group_sharded_parallel(offload=True),
        # eager_engine.py:281-307): fp32 master/m/v live in pinned host
        # memory; the step runs on CPU and only the bf16 shard returns
        self.offload = offload
        self.state = []
        for u in wrapper.units:
            master = u.shard.float()
            if offload:
                master = master.cpu()
                if torch.cuda.is_available():
                    master = master.pin_memory()
            self.state.append({
                "master": master,
                "exp_avg": torch.zeros_like(master),
                "exp_avg_sq": torch.zeros_like(master),
            })

    def zero_grad(self, set_to_none: bool = False):
        for u in self.w.units:
            u.grad_shard.zero_()

    def grad_global_norm(self) -> torch.Tensor:
        dev = self.w.units[0].shard.device
        sq = torch.zeros((), dtype=torch.float32, device=dev)
        for u in self.w.units:
            sq += torch.linalg.vector_norm(u.grad_shard,
                                           dtype=torch.float32) ** 2
        if self.w.world > 1 and dist.is_initialized():
            dist.all_reduce(sq, group=self.w.group)
        return sq.sqrt()

    def reduce_and_step(self, lr: Optional[float] = None,
                        grad_clip: Optional[float] = None,
                        loss_scale: float = 1.0, dp_group=None):
        if lr is not None:
            self.lr = lr
        if dp_group is not None and getattr(dp_group, "world_size", 1) > 1:
            pg = dp_group.group if hasattr(dp_group, "group") else dp_group
            for u in self.w.units:
                dist.all_reduce(u.grad_shard, group=pg)
            loss_scale = loss_scale * dp_group.world_size
        # the sharding group is also a data-replica axis: the reduce-scatter
        # summed `world` replicas' grads (env.get_data_world_size semantics)
        if self.w.world > 1:
            loss_scale = loss_scale * self.w.world
        if loss_scale != 1.0:
            for u in self.w.units:
                u.grad_shard.div_(loss_scale)
        if grad_clip:
            norm = self.grad_global_norm()
            scale = grad_clip / (float(norm) + 1e-6)
            if scale < 1.0:
                for u in self.w.units:
                    u.grad_shard.mul_(scale)
        self._step += 1
        for u, st in zip(self.w.units, self.state):
            if self.offload:
                g = u.grad_shard.to(st["master"].device,
                                    non_blocking=False).float()
                model_out = torch.empty_like(st["master"])
                fused_adamw_flat(st["master"], g, st["exp_avg"],
                                 st["exp_avg_sq"], model_out, self.lr,
                                 self.beta1, self.beta2, self.eps,
                                 self.weight_decay, self._step)
                u.shard.copy_(model_out.to(u.shard.dtype))
            else:
                fused_adamw_flat(st["master"], u.grad_shard, st["exp_avg"],
                                 st["exp_avg_sq"], u.shard, self.lr,
                                 self.beta1, self.beta2, self.eps,
                                 self.weight_decay, self._step)
            u.writeback_from_shard()
        self.zero_grad()

    def state_dict(self):
        return {"step": self._step, "lr": self.lr, "state": self.state}

    def load_state_dict(self, sd):
        self._step = sd["step"]
        self.lr = sd.get("lr", self.lr)
        for st, s in zip(self.state, sd["state"]):
            st["master"].copy_(s["master"])
            st["exp_avg"].copy_(s["exp_avg"])
            st["exp_avg_sq"].copy_(s["exp_avg_sq"])
        for u, st in zip(self.w.units, self.state):
            u.shard.copy_(st["master"].to(u.shard.dtype))
            u.writeback_from_shard()
