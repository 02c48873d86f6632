"""EagerEngine: the trainer.

Reference: ppfleetx/core/engine/eager_engine.py (fit :422, _fit_impl :479,
micro-batch accumulation :522-561, optim update :563-578, save/load
:717-825 with the epoch_X_step_Y/mp_XX_sharding_XX_pp_XX layout).

MI355X-native differences: AMP-O2 = bf16 weights + fp32 flat master/grad
buffers inside FusedAdamW (one HIP kernel per bucket); DP gradient
allreduce runs directly on the fused fp32 buffers over RCCL; fp16 path
keeps a dynamic loss scale with a cross-rank found_inf allreduce
(reference amp.py:193-234); pipeline parallel delegates to the native
1F1B scheduler.
"""

from __future__ import annotations

import os
import time
from typing import Any, Dict, Optional

import torch
import torch.distributed as dist

from paddlefleetx_amd.optims import build_lr_scheduler, build_optimizer
from paddlefleetx_amd.optims.optimizer import FusedAdamW
from paddlefleetx_amd.parallel import sp as sp_ops
from paddlefleetx_amd.parallel.env import get_hcg
from paddlefleetx_amd.utils.log import logger


class BasicEngine:
    def fit(self, *a, **k):
        raise NotImplementedError

    def evaluate(self, *a, **k):
        raise NotImplementedError


def _split_micro(batch, n: int):
    """Split each tensor of the batch along dim0 into n micro-batches."""
    if n == 1:
        return [batch]
    parts = [torch.chunk(t, n, dim=0) if torch.is_tensor(t) else [t] * n
             for t in batch]
    return [tuple(p[i] for p in parts) for i in range(n)]


class EagerEngine(BasicEngine):
    def __init__(self, configs, module, mode: str = "train"):
        self.configs = configs
        self.module = module
        self.mode = mode
        self.hcg = get_hcg()

        e = configs["Engine"]
        self.accumulate_steps = int(e.get("accumulate_steps", 1))
        self.max_steps = e.get("max_steps")
        self.logging_freq = int(e.get("logging_freq", 10))
        self.eval_freq = e.get("eval_freq")
        self.eval_iters = int(e.get("eval_iters", 10))
        self.num_train_epochs = int(e.get("num_train_epochs", 1))
        sl = e.get("save_load", {})
        self.save_steps = sl.get("save_steps")
        self.output_dir = sl.get("output_dir", "./output")

        mpc = e.get("mix_precision", {})
        self.amp_enable = bool(mpc.get("enable", True))
        self.amp_dtype = {"bfloat16": torch.bfloat16, "float16": torch.float16,
                          "float32": torch.float32}[mpc.get("dtype", "bfloat16")]
        self.loss_scale = float(mpc.get("scale_loss", 32768.0)) \
            if self.amp_dtype == torch.float16 else 1.0
        self._scale_growth_interval = 1000
        self._good_steps = 0

        self.device = torch.device("cuda") if torch.cuda.is_available() \
            else torch.device("cpu")
        if self.device.type == "cuda":
            from paddlefleetx_amd.utils.tunable import enable_tuned_gemms
            enable_tuned_gemms()
        self.module.to(self.device)

        self.is_pipeline = self.hcg.get_pipe_parallel_world_size() > 1
        if self.is_pipeline:
            assert hasattr(self.module.model, "forward_backward_pipeline"), \
                "pp_degree>1 requires a pipeline model"

        # ZeRO sharding
        sh = configs.get("Distributed", {}).get("sharding", {})
        self.sharding_stage = int(sh.get("sharding_stage", 1))
        self.sharding_degree = self.hcg.get_sharding_parallel_world_size()

        opt_cfg = configs.get("Optimizer", {})
        self.lr_scheduler = build_lr_scheduler(opt_cfg.get("lr", {}))
        self.grad_clip_norm = None
        gc = opt_cfg.get("grad_clip", None)
        if isinstance(gc, dict):
            self.grad_clip_norm = gc.get("clip_norm", 1.0)
        elif isinstance(gc, (int, float)):
            self.grad_clip_norm = float(gc)

        if mode == "train":
            sharding_group = self.hcg.get_sharding_parallel_group() \
                if self.sharding_degree > 1 else None
            if self.sharding_stage == 3 and self.sharding_degree > 1:
                # ZeRO-3: wrap the model for parameter sharding
                # (reference group_sharded_parallel level="p_g_os",
                # eager_engine.py:281-307)
                from paddlefleetx_amd.parallel.zero3 import (
                    GroupShardedStage3, Stage3AdamW)
                assert not self.is_pipeline, \
                    "sharding stage 3 excludes pipeline parallel " \
                    "(reference eager_engine.py:276)"
                self.module.model = GroupShardedStage3(
                    self.module.model, group=sharding_group)
                ocfg = {k: v for k, v in opt_cfg.items()
                        if k in ("weight_decay", "beta1", "beta2", "epsilon")}
                sh_cfg = configs.get("Distributed", {}).get("sharding", {})
                self.optimizer = Stage3AdamW(
                    self.module.model, lr=self.lr_scheduler.get_lr(),
                    offload=bool(sh_cfg.get("offload", False)), **ocfg)
            else:
                self.optimizer = build_optimizer(
                    opt_cfg, self.module.model,
                    lr_value=self.lr_scheduler.get_lr(),
                    sharding_group=sharding_group,
                    sharding_stage=self.sharding_stage)
            # broadcast initial params across dp (and sharding) so replicas agree
            self._sync_params()
            # DP grad allreduce overlapped with the last backward
            # (reference reduce_overlap knob, eager_engine.py:303-307)
            self._overlap_reduce = False
            want_overlap = bool(configs.get("Distributed", {})
                                .get("reduce_overlap", True))
            if want_overlap and isinstance(self.optimizer, FusedAdamW) \
                    and not self.is_pipeline:
                self._overlap_reduce = self.optimizer.enable_overlap(
                    self.hcg.get_data_parallel_group())
            # arm in-GEMM weight-grad accumulation (ops/linear.py): grads
            # land in the bucket views inside the wgrad GEMM (beta=1),
            # skipping autograd's per-micro dW materialize + add
            if isinstance(self.optimizer, FusedAdamW) \
                    and not self.optimizer._fp32_main_grad:
                from paddlefleetx_amd.ops.linear import set_wgrad_fusion
                set_wgrad_fusion(True)
        else:
            self.optimizer = None
            self._overlap_reduce = False

        self._load_recovery = {"step": 0, "epoch": 0}
        ckpt_dir = sl.get("ckpt_dir") if isinstance(sl, dict) else None
        if ckpt_dir:
            self.load(ckpt_dir)

        # hipGraph capture of the micro-step (Engine.hip_graph: true):
        # records one forward+backward after warmup and replays it with
        # static input buffers — removes per-kernel launch overhead on
        # launch-bound (small-model) configs. Constraints enforced in
        # _maybe_capture_graph (single process, no overlap hooks, no
        # dropout/dynamic loss scale: Philox seeds and the scale are
        # baked into the capture).
        self._graph = None
        self._graph_want = bool(configs.get("Engine", {})
                                .get("hip_graph", False))
        self._graph_steps_seen = 0

        from paddlefleetx_amd.utils.profiler import ProfilerGuard
        self.profiler = ProfilerGuard(configs.get("Profiler"))

    # ------------------------------------------------------------------
    def _sync_params(self):
        """Broadcast params from dp-rank-0 (reference strategy.py:43 sync_params_buffers).
        Also broadcasts mp-REPLICATED params (no `is_mp` mark: LN, biases,
        position tables) over the mp group so all TP ranks start from one
        copy regardless of init-stream discipline."""
        mp_g = self.hcg.get_model_parallel_group()
        if mp_g.world_size > 1 and dist.is_initialized():
            from paddlefleetx_amd.models.moe.moe_layer import MoELayer
            expert_params = set()
            for m in self.module.model.modules():
                if isinstance(m, MoELayer):
                    for p in m.experts.parameters():
                        expert_params.add(id(p))
            for p in self.module.model.parameters():
                if getattr(p, "is_mp", False) or id(p) in expert_params:
                    continue
                dist.broadcast(p.data, src=mp_g.ranks[0], group=mp_g.group)
        from paddlefleetx_amd.parallel.zero3 import Stage3AdamW
        if isinstance(self.optimizer, Stage3AdamW):
            # stage 3: shards live per rank; sync shards over dp only
            dp = self.hcg.get_data_parallel_group()
            if dp.world_size > 1 and dist.is_initialized():
                for u, st in zip(self.module.model.units,
                                 self.optimizer.state):
                    dist.broadcast(u.shard, src=dp.ranks[0], group=dp.group)
                    st["master"].copy_(u.shard.float())
            return
        dp = self.hcg.get_data_parallel_group()
        sd = self.hcg.get_sharding_parallel_group()
        for g in (dp, sd):
            if g.world_size <= 1 or not dist.is_initialized():
                continue
            if isinstance(self.optimizer, FusedAdamW):
                for b in self.optimizer.buckets:
                    dist.broadcast(b.model_flat, src=g.ranks[0], group=g.group)
                    b.master.copy_(b.model_shard.float()
                                   if self.optimizer.sharding_group is not None
                                   else b.model_flat.float())
            else:
                for p in self.module.model.parameters():
                    dist.broadcast(p.data, src=g.ranks[0], group=g.group)

    # ------------------------------------------------------------------
    def fit(self, train_data_loader=None, valid_data_loader=None, epoch=None):
        epochs = epoch if epoch is not None else self.num_train_epochs
        start_epoch = self._load_recovery["epoch"]
        try:
            for ep in range(start_epoch, epochs):
                done = self._train_one_epoch(ep, train_data_loader,
                                             valid_data_loader)
                if done:
                    break
        finally:
            self.profiler.stop_and_summary()

    def _train_one_epoch(self, epoch: int, train_loader, valid_loader) -> bool:
        self.module.model.train()
        skip_until = self._load_recovery["step"] if \
            epoch == self._load_recovery["epoch"] else 0
        t_start = time.time()
        interval_cost = 0.0
        for step, batch in enumerate(train_loader):
            if step < skip_until:
                continue
            loss = self._fit_impl(batch)
            self.profiler.step()
            self.module.global_step += 1
            gstep = self.module.global_step
            interval_cost = time.time() - t_start
            if gstep % self.logging_freq == 0:
                self.module.training_step_end({
                    "epoch": epoch, "batch": step,
                    "loss": float(loss),
                    "train_cost": interval_cost / self.logging_freq,
                    "lr": self.lr_scheduler.get_lr(),
                    "found_inf": float(getattr(self, "_found_inf", 0.0)),
                })
                t_start = time.time()
            if self.eval_freq and gstep % self.eval_freq == 0 and valid_loader:
                self._evaluate_impl(epoch, valid_loader)
                self.module.model.train()
            if self.save_steps and gstep % self.save_steps == 0:
                if self.device.type == "cuda":
                    torch.cuda.synchronize()
                self.save(epoch, gstep)
            if self.max_steps and gstep >= self.max_steps:
                return True
        return False

    # ------------------------------------------------------------------
    def _fit_impl(self, batch) -> torch.Tensor:
        batch = self.module.pretreating_batch(batch)
        batch = tuple(t.to(self.device, non_blocking=True) if torch.is_tensor(t)
                      else t for t in batch)
        if self.is_pipeline:
            loss = self.module.model.forward_backward_pipeline(
                batch, self.module.loss_fn, self.accumulate_steps,
                scale=self.loss_scale)
        else:
            if self._graph_want and self._graph is None:
                self._maybe_capture_graph(batch)
            loss = self._model_forward_backward(batch)
        self._optim_update_params()
        return loss

    def _model_forward_backward(self, batch) -> torch.Tensor:
        micros = _split_micro(batch, self.accumulate_steps)
        total = None
        for i, mb in enumerate(micros):
            loss = self._graph_step(mb) if self._graph is not None \
                else None
            if loss is None:
                loss = self.module.training_step(mb)
                scaled = loss * (self.loss_scale / self.accumulate_steps)
                if self._overlap_reduce and i == len(micros) - 1:
                    self.optimizer.begin_overlap_reduce()
                self.module.backward(scaled)
            # accumulate ON DEVICE: a float() here would host-sync every
            # micro-batch and stall the launch pipeline. clone() because
            # under graph replay `loss` is the static capture tensor that
            # the NEXT replay overwrites
            d = loss.detach().clone()
            total = d if total is None else total + d
        return total / self.accumulate_steps

    def _maybe_capture_graph(self, batch):
        """Capture ONE micro-batch forward+backward into a hipGraph after
        two warmup steps. Grad accumulation works across replays because
        every parameter gradient lives in a pre-allocated bucket view
        (optims/optimizer.py) that the recorded kernels add into."""
        import torch.distributed as tdist
        if self.device.type != "cuda" or self.is_pipeline:
            return
        if tdist.is_initialized() and tdist.get_world_size() > 1:
            return  # collectives inside capture not supported here
        if getattr(self, "_overlap_reduce", False) or self.loss_scale != 1.0:
            return
        mcfg = self.configs.get("Model", {})
        if float(mcfg.get("hidden_dropout_prob", 0) or 0) > 0 or                 float(mcfg.get("attention_probs_dropout_prob", 0) or 0) > 0:
            return  # RNG seeds would be frozen into the capture
        self._graph_steps_seen += 1
        if self._graph_steps_seen <= 2:
            return  # allocator/TunableOp warmup on the eager path
        micros = _split_micro(batch, self.accumulate_steps)
        static = tuple(t.clone() if torch.is_tensor(t) else t
                       for t in micros[0])
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self.optimizer.zero_grad()
                loss = self.module.training_step(static)
                (loss * (1.0 / self.accumulate_steps)).backward()
        torch.cuda.current_stream().wait_stream(s)
        self.optimizer.zero_grad()
        g = torch.cuda.CUDAGraph()
        try:
            with torch.cuda.graph(g):
                loss = self.module.training_step(static)
                (loss * (1.0 / self.accumulate_steps)).backward()
        except Exception as e:
            logger.warning(f"hipGraph capture failed, staying eager: {e}")
            self.optimizer.zero_grad()
            self._graph_want = False
            return
        self.optimizer.zero_grad()
        self._graph = {"obj": g, "static": static, "loss": loss}
        logger.info("hipGraph captured: replaying the micro-step "
                    f"(acc={self.accumulate_steps})")

    def _graph_step(self, mb):
        st = self._graph["static"]
        for dst, srct in zip(st, mb):
            if torch.is_tensor(dst):
                dst.copy_(srct, non_blocking=True)
        self._graph["obj"].replay()
        return self._graph["loss"]

    def _optim_update_params(self):
        model = self.module.model
        # SP: LayerNorm/bias grads over mp group (sequence_parallel_utils.py:166-185)
        if getattr(model, "sequence_parallel", False) or any(
                getattr(p, "sequence_parallel", False)
                for p in model.parameters()):
            self._allreduce_sp_main_grads()
        self._found_inf = 0.0
        if isinstance(self.optimizer, FusedAdamW):
            # ZeRO reduce-scatter over sharding + DP allreduce on fused buffers
            dp = self.hcg.get_data_parallel_group()
            n_replicas = dp.world_size * self.sharding_degree
            if getattr(self, "_overlap_reduce", False):
                self.optimizer.finish_overlap_reduce()  # issued in backward
            else:
                self.optimizer.reduce_gradients(dp, avg_factor=1.0)
            inv = 1.0 / (n_replicas * self.loss_scale)
            self.optimizer.scale_grads(inv)
            if self.loss_scale != 1.0:
                self._found_inf = self._check_found_inf()
                if self._found_inf:
                    self.loss_scale = max(1.0, self.loss_scale / 2)
                    self._good_steps = 0
                    self.optimizer.zero_grad()
                    self.lr_scheduler.step()
                    return
                self._good_steps += 1
                if self._good_steps >= self._scale_growth_interval:
                    self.loss_scale *= 2
                    self._good_steps = 0
            if self.grad_clip_norm:
                self.optimizer.clip_grads(
                    self.grad_clip_norm,
                    mp_group=self.hcg.get_model_parallel_group(),
                    pp_group=self.hcg.get_pipe_parallel_group())
            self.lr_scheduler.step()
            self.optimizer.step(lr=self.lr_scheduler.get_lr())
            self.optimizer.zero_grad()
        else:
            # plain torch optimizer path (incl. Stage3AdamW which mimics it)
            if hasattr(self.optimizer, "reduce_and_step"):
                self.lr_scheduler.step()
                self.optimizer.reduce_and_step(
                    lr=self.lr_scheduler.get_lr(),
                    grad_clip=self.grad_clip_norm,
                    loss_scale=self.loss_scale,
                    dp_group=self.hcg.get_data_parallel_group())
            else:
                self._allreduce_plain_grads()
                if self.grad_clip_norm:
                    torch.nn.utils.clip_grad_norm_(
                        self.module.model.parameters(), self.grad_clip_norm)
                self.lr_scheduler.step()
                for g in self.optimizer.param_groups:
                    g["lr"] = self.lr_scheduler.get_lr()
                self.optimizer.step()
                self.optimizer.zero_grad(set_to_none=True)

    def _check_found_inf(self) -> float:
        found = 0.0 if self.optimizer.check_finite() else 1.0
        if dist.is_initialized():
            t = torch.tensor(found, device=self.device)
            dist.all_reduce(t, op=dist.ReduceOp.MAX)
            found = float(t)
        return found

    def _allreduce_sp_main_grads(self):
        g = self.hcg.get_model_parallel_group()
        if g.world_size == 1:
            return
        for p in self.module.model.parameters():
            if getattr(p, "sequence_parallel", False):
                buf = getattr(p, "main_grad", None)
                if buf is None:
                    buf = p.grad
                if buf is not None:
                    dist.all_reduce(buf, group=g.group)

    def _allreduce_plain_grads(self):
        dp = self.hcg.get_data_parallel_group()
        if dp.world_size == 1:
            return
        for p in self.module.model.parameters():
            if p.grad is not None:
                dist.all_reduce(p.grad, group=dp.group)
                p.grad.div_(dp.world_size)

    # ------------------------------------------------------------------
    @torch.no_grad()
    def _evaluate_impl(self, epoch: int, loader):
        self.module.model.eval()
        t0 = time.time()
        for i, batch in enumerate(loader):
            if i >= self.eval_iters:
                break
            batch = tuple(t.to(self.device) if torch.is_tensor(t) else t
                          for t in batch)
            if self.is_pipeline:
                # stage-local validation_step can't see the whole model;
                # run the forward-only pipeline schedule instead
                # (reference eager_engine.py:655 eval_batch)
                loss = self.module.model.eval_pipeline(
                    batch, self.module.loss_fn, self.accumulate_steps)
            else:
                loss = self.module.validation_step(batch)
            self.module.validation_step_end({
                "epoch": epoch, "batch": i, "loss": float(loss),
                "eval_cost": (time.time() - t0) / (i + 1)})
        if hasattr(self.module, "validation_epoch_end"):
            try:
                self.module.validation_epoch_end()
            except TypeError:
                pass  # modules with a log_dict-taking signature

    def export(self, output_dir: str = "./exported_model"):
        """Export the trained model for inference (reference
        eager_engine.py:832 + utils/export.py)."""
        from paddlefleetx_amd.utils.export import export_inference_model
        extra = {}
        if "Generation" in self.configs:
            extra["generation"] = dict(self.configs["Generation"])
        return export_inference_model(self.module.model,
                                      dict(self.configs["Model"]),
                                      output_dir, extra=extra)

    def inference(self, data, model_dir: str = "./exported_model"):
        """Run the exported model (reference eager_engine.py:852)."""
        from paddlefleetx_amd.core.inference_engine import InferenceEngine
        if not hasattr(self, "_infer_engine"):
            mp = self.hcg.get_model_parallel_world_size()
            self._infer_engine = InferenceEngine(model_dir, mp_degree=mp)
        return self._infer_engine.predict(data)

    def compress_model(self):
        """Apply the Compress config section (reference eager_engine.py
        :757-774 via utils/compression_helper.py)."""
        ccfg = self.configs.get("Compress", {})
        if not ccfg:
            return
        from paddlefleetx_amd.utils.compression_helper import (prune_model,
                                                               qat_model,
                                                               quant_model)
        if "Prune" in ccfg:
            p = ccfg["Prune"] or {}
            prune_model(self.module.model,
                        ratio=float(p.get("ratio", 0.125)),
                        structured=p.get("criterion", "l1_norm") != "unstructured",
                        include=p.get("include"))
        if "Quantization" in ccfg:
            q = ccfg["Quantization"] or {}
            if q.get("enable_qat") or ccfg.get("enable_qat"):
                # training-time fake quant (reference quant_model
                # compression_helper.py:210 QAT path)
                qat_model(self.module.model, include=q.get("include"),
                          bits=int(q.get("weight_bits", 8)))
            else:
                quant_model(self.module.model, include=q.get("include"))

    def evaluate(self, valid_data_loader=None, epoch: int = 0):
        self._evaluate_impl(epoch, valid_data_loader)

    @torch.no_grad()
    def predict(self, data_loader):
        self.module.model.eval()
        outs = []
        for batch in data_loader:
            batch = tuple(t.to(self.device) if torch.is_tensor(t) else t
                          for t in batch)
            outs.append(self.module(*batch))
        return outs

    # ------------------------------------------------------------------
    # checkpoint: epoch_X_step_Y/mp_XX_sharding_XX_pp_XX/{model.pdparams,
    #             model_state.pdopt, meta_state.pdopt}  (eager_engine.py:717-755)
    def _ckpt_subdir(self) -> str:
        return "mp_{:02d}_sharding_{:02d}_pp_{:02d}".format(
            self.hcg.get_model_parallel_rank(),
            self.hcg.get_sharding_parallel_rank(),
            self.hcg.get_pipe_parallel_rank())

    def save(self, epoch: int = 0, step: int = 0):
        if self.hcg.get_data_parallel_rank() != 0:
            return  # only dp rank 0 saves (io.py:44-46)
        out = os.path.join(self.output_dir, f"epoch_{epoch}_step_{step}",
                           self._ckpt_subdir())
        os.makedirs(out, exist_ok=True)
        torch.save(self.module.model.state_dict(),
                   os.path.join(out, "model.pdparams"))
        if self.optimizer is not None:
            torch.save(self.optimizer.state_dict(),
                       os.path.join(out, "model_state.pdopt"))
        meta = {"epoch": epoch, "step": step,
                "global_step": self.module.global_step,
                "lr_scheduler": self.lr_scheduler.state_dict(),
                "loss_scale": self.loss_scale,
                "good_steps": getattr(self, "_good_steps", 0),
                "cpu_rng_state": torch.get_rng_state()}
        from paddlefleetx_amd.parallel.rng import get_rng_tracker
        meta["rng_tracker"] = get_rng_tracker().state_dict()
        if torch.cuda.is_available():
            meta["cuda_rng_state"] = torch.cuda.get_rng_state()
        torch.save(meta, os.path.join(out, "meta_state.pdopt"))
        logger.info(f"saved checkpoint to {out}")

    def load(self, ckpt_dir: str):
        path = os.path.join(ckpt_dir, self._ckpt_subdir())
        if not os.path.isdir(path):
            path = ckpt_dir
        model_path = os.path.join(path, "model.pdparams")
        sd = torch.load(model_path, map_location=self.device,
                        weights_only=False)
        missing, unexpected = self.module.model.load_state_dict(sd, strict=False)
        if missing or unexpected:
            logger.warning(f"ckpt load: missing={missing} unexpected={unexpected}")
        if isinstance(self.optimizer, FusedAdamW):
            # re-pointing: state_dict load above replaced p.data tensors; refuse
            for b in self.optimizer.buckets:
                for p, off in zip(b.params, b.offsets):
                    b.model_flat[off:off + p.numel()].copy_(p.data.reshape(-1))
                    p.data = b.model_flat[off:off + p.numel()].view(p.shape)
                b.master.copy_(
                    b.model_flat[b.shard_lo:b.shard_hi].to(torch.float32))
        else:
            from paddlefleetx_amd.parallel.zero3 import Stage3AdamW
            if isinstance(self.optimizer, Stage3AdamW):
                # shards were refreshed by the wrapper's load_state_dict;
                # re-derive the fp32 masters (the optimizer checkpoint, if
                # present, overwrites them exactly below)
                for u, st in zip(self.optimizer.w.units,
                                 self.optimizer.state):
                    st["master"].copy_(u.shard.to(st["master"].device)
                                       .float())
        opt_path = os.path.join(path, "model_state.pdopt")
        if self.optimizer is not None and os.path.exists(opt_path):
            self.optimizer.load_state_dict(
                torch.load(opt_path, map_location=self.device,
                           weights_only=False))
        meta_path = os.path.join(path, "meta_state.pdopt")
        if os.path.exists(meta_path):
            meta = torch.load(meta_path, map_location="cpu", weights_only=False)
            self._load_recovery = {"step": meta["step"], "epoch": meta["epoch"]}
            self.module.global_step = meta.get("global_step", meta["step"])
            if "lr_scheduler" in meta:
                self.lr_scheduler.load_state_dict(meta["lr_scheduler"])
            if "loss_scale" in meta and self.loss_scale != 1.0:
                # resume the dynamic fp16 scale where it left off (a
                # reset to the 32768 default would burn found_inf skips)
                self.loss_scale = float(meta["loss_scale"])
                self._good_steps = int(meta.get("good_steps", 0))
            if "cpu_rng_state" in meta:
                torch.set_rng_state(meta["cpu_rng_state"])
            if "rng_tracker" in meta:
                # resume the mp-dropout streams where they left off
                # (the reference restores only the global CUDA state)
                from paddlefleetx_amd.parallel.rng import get_rng_tracker
                get_rng_tracker().load_state_dict(meta["rng_tracker"])
            if "cuda_rng_state" in meta and torch.cuda.is_available():
                torch.cuda.set_rng_state(meta["cuda_rng_state"])
        logger.info(f"loaded checkpoint from {path}")
