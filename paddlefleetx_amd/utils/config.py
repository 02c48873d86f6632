"""YAML config system: `_base_` inheritance, dotted CLI overrides, derivation passes.

Mirrors the reference surface (ppfleetx/utils/config.py:398 `get_config`,
:263 `_base_` inheritance, :333 dotted overrides, :33 process_dist_config,
:104 process_global_configs, :151 process_engine_config) with a fresh
implementation on plain dict/AttrDict.
"""

from __future__ import annotations

import copy
import os
from typing import Any, Dict, List, Optional

import yaml

__all__ = ["AttrDict", "get_config", "parse_args_overrides", "print_config"]

BASE_KEY = "_base_"


class AttrDict(dict):
    """dict with attribute access, recursively applied."""

    def __getattr__(self, key: str) -> Any:
        try:
            return self[key]
        except KeyError as e:
            raise AttributeError(key) from e

    def __setattr__(self, key: str, value: Any) -> None:
        self[key] = value

    def __deepcopy__(self, memo):
        return AttrDict({k: copy.deepcopy(v, memo) for k, v in self.items()})

    def setdefault_path(self, path: str, value: Any) -> Any:
        node = self
        keys = path.split(".")
        for k in keys[:-1]:
            node = node.setdefault(k, AttrDict())
        return node.setdefault(keys[-1], value)


def _to_attrdict(obj: Any) -> Any:
    if isinstance(obj, dict):
        return AttrDict({k: _to_attrdict(v) for k, v in obj.items()})
    if isinstance(obj, list):
        return [_to_attrdict(v) for v in obj]
    return obj


def _merge(base: Dict, new: Dict) -> Dict:
    """Recursively merge `new` on top of `base` (new wins)."""
    out = dict(base)
    for k, v in new.items():
        if k in out and isinstance(out[k], dict) and isinstance(v, dict):
            out[k] = _merge(out[k], v)
        else:
            out[k] = v
    return out


def parse_yaml_with_base(fname: str, _seen: Optional[set] = None) -> Dict:
    """Load a YAML file honoring `_base_:` (str or list) relative includes."""
    fname = os.path.abspath(fname)
    _seen = _seen or set()
    if fname in _seen:
        raise ValueError(f"circular _base_ include: {fname}")
    _seen.add(fname)
    with open(fname, "r") as f:
        cfg = yaml.safe_load(f) or {}
    if BASE_KEY in cfg:
        bases = cfg.pop(BASE_KEY)
        if isinstance(bases, str):
            bases = [bases]
        merged: Dict = {}
        for b in bases:
            bpath = os.path.join(os.path.dirname(fname), b)
            merged = _merge(merged, parse_yaml_with_base(bpath, _seen))
        cfg = _merge(merged, cfg)
    return cfg


def _parse_scalar(v: str) -> Any:
    """Parse an override value string with YAML semantics (1e-4 -> float etc.)."""
    try:
        out = yaml.safe_load(v)
    except yaml.YAMLError:
        return v
    if isinstance(out, str):
        # YAML 1.1 misses bare scientific notation like 1e-3
        try:
            return float(out)
        except ValueError:
            return out
    return out


def apply_override(cfg: Dict, key: str, value: Any) -> None:
    node = cfg
    keys = key.split(".")
    for k in keys[:-1]:
        if k not in node or not isinstance(node[k], dict):
            node[k] = AttrDict()
        node = node[k]
    node[keys[-1]] = value


def parse_args_overrides(overrides: Optional[List[str]]) -> List:
    """Parse `-o a.b.c=v` style overrides into (key, parsed_value) pairs."""
    out = []
    for item in overrides or []:
        if "=" not in item:
            raise ValueError(f"override must be key=value, got: {item}")
        k, v = item.split("=", 1)
        out.append((k.strip(), _parse_scalar(v.strip())))
    return out


# ---------------------------------------------------------------------------
# Derivation passes (reference: config.py:33-189)
# ---------------------------------------------------------------------------

def _env_world_size() -> int:
    return int(os.environ.get("WORLD_SIZE", os.environ.get("PADDLE_TRAINERS_NUM", "1")))


def _setdef(d: dict, key: str, default: Any) -> Any:
    """setdefault that also replaces an explicit None (empty YAML value)."""
    if d.get(key, None) is None:
        d[key] = default
    return d[key]


def process_dist_config(cfg: AttrDict) -> None:
    """Fill Distributed section; derive dp_degree from world size.

    Reference: ppfleetx/utils/config.py:33-101.
    """
    dist = _setdef(cfg, "Distributed", AttrDict())
    nranks = dist.get("world_size") or _env_world_size()
    mp = int(_setdef(dist, "mp_degree", 1))
    pp = int(_setdef(dist, "pp_degree", 1))
    cp = int(_setdef(dist, "cp_degree", 1))
    sharding = _setdef(dist, "sharding", AttrDict())
    if not isinstance(sharding, dict):
        sharding = AttrDict()
        dist["sharding"] = sharding
    sd = int(_setdef(sharding, "sharding_degree", 1))
    _setdef(sharding, "sharding_stage", 1)
    _setdef(sharding, "reduce_overlap", False)
    _setdef(sharding, "broadcast_overlap", False)
    _setdef(sharding, "offload", False)
    other = mp * pp * sd * cp
    assert nranks % other == 0, (
        f"world_size {nranks} not divisible by mp*pp*sharding*cp = {other}")
    dp = _setdef(dist, "dp_degree", nranks // other)
    assert dp * other == nranks, (
        f"dp_degree {dp} x mp {mp} x pp {pp} x sharding {sd} x cp {cp} != world {nranks}")
    dist["world_size"] = nranks
    # MoE legality (reference comm_groups.py:133-137): expert parallel needs pp==1, sharding==1
    if cfg.get("Model", {}).get("moe_configs", None):
        assert pp == 1 and sd == 1, "MoE expert parallel requires pp_degree==1 and sharding_degree==1"
    # pipeline knobs
    pipeline = _setdef(dist, "pipeline", AttrDict())
    _setdef(pipeline, "schedule", "1F1B")
    _setdef(pipeline, "virtual_pp_degree", 1)
    # default OFF on MI355X: each pp p2p channel has a dedicated xGMI
    # link, so splitting the boundary tensor across mp ranks (reference
    # default True for PCIe-bound clusters) mostly adds an allgather of
    # latency; the knob is implemented (parallel/pp.py) for parity
    _setdef(pipeline, "enable_partial_send_recv", False)
    if cfg.get("Model", {}).get("sequence_parallel", False):
        # SP shards activations along seq; partial send recv is incompatible
        # (reference config.py:112-119)
        pipeline["enable_partial_send_recv"] = False


def process_global_configs(cfg: AttrDict) -> None:
    """Batch-size math: global = local x dp x sharding (reference config.py:104-148)."""
    g = _setdef(cfg, "Global", AttrDict())
    dist = cfg["Distributed"]
    dp = dist["dp_degree"]
    sd = dist["sharding"]["sharding_degree"]
    data_world = dp * sd

    gbs = g.get("global_batch_size", None)
    lbs = g.get("local_batch_size", None)
    mbs = g.get("micro_batch_size", None)

    if gbs is None and lbs is None:
        raise ValueError("Global.global_batch_size or Global.local_batch_size required")
    if gbs is None:
        gbs = lbs * data_world
    elif lbs is None:
        assert gbs % data_world == 0, (
            f"global_batch_size {gbs} not divisible by dp*sharding {data_world}")
        lbs = gbs // data_world
    if mbs is None:
        mbs = lbs
    assert lbs % mbs == 0, f"local_batch_size {lbs} not divisible by micro_batch_size {mbs}"
    g["global_batch_size"], g["local_batch_size"], g["micro_batch_size"] = gbs, lbs, mbs
    _setdef(g, "seed", 1024)
    _setdef(g, "device", "gpu")
    _setdef(g, "max_steps", None)
    _setdef(g, "logging_freq", 10)
    _setdef(g, "eval_freq", None)
    _setdef(g, "save_steps", None)
    _setdef(g, "output_dir", "./output")


def process_engine_config(cfg: AttrDict) -> None:
    """accumulate_steps = local/micro (reference config.py:151-189)."""
    e = _setdef(cfg, "Engine", AttrDict())
    g = cfg["Global"]
    e["accumulate_steps"] = g["local_batch_size"] // g["micro_batch_size"]
    if g.get("max_steps") is not None:
        # an explicit Global.max_steps (CLI override or yaml) wins over
        # the Engine default so `-o Global.max_steps=N` always bounds
        # the run
        e["max_steps"] = g["max_steps"]
    else:
        _setdef(e, "max_steps", None)
    _setdef(e, "logging_freq", g.get("logging_freq", 10))
    _setdef(e, "eval_freq", g.get("eval_freq"))
    _setdef(e, "eval_iters", 10)
    _setdef(e, "test_iters", None)
    _setdef(e, "num_train_epochs", 1)
    _setdef(e, "save_load", AttrDict())
    _setdef(e["save_load"], "save_steps", g.get("save_steps"))
    _setdef(e["save_load"], "save_epoch", 1)
    _setdef(e["save_load"], "output_dir", g.get("output_dir", "./output"))
    _setdef(e["save_load"], "ckpt_dir", None)
    mp = _setdef(e, "mix_precision", AttrDict())
    _setdef(mp, "enable", True)
    _setdef(mp, "dtype", "bfloat16")
    _setdef(mp, "level", "O2")
    _setdef(mp, "scale_loss", 32768.0)
    _setdef(mp, "custom_black_list", [])
    _setdef(mp, "custom_white_list", [])


def process_model_configs(cfg: AttrDict) -> None:
    """Model-family defaults (reference models/language_model/utils.py:55-179)."""
    m = cfg.get("Model", None)
    if m is None:
        return
    if "hidden_size" in m:
        _setdef(m, "ffn_hidden_size", 4 * m["hidden_size"])
    if "use_recompute" in m:
        _setdef(m, "recompute_granularity", "full")
    _setdef(m, "sequence_parallel", False)
    _setdef(m, "fused_attn", True)
    _setdef(m, "fused_softmax_with_triangular", True)
    # vocab padding to multiple of 128*mp (reference language_module.py:62-74)
    if "vocab_size" in m:
        mp_deg = cfg.get("Distributed", {}).get("mp_degree", 1)
        mult = m.get("vocab_size_divisible_unit", 128) * mp_deg
        vs = m["vocab_size"]
        m["padded_vocab_size"] = ((vs + mult - 1) // mult) * mult


def process_optimizer_configs(cfg: AttrDict) -> None:
    o = _setdef(cfg, "Optimizer", AttrDict())
    _setdef(o, "name", "FusedAdamW")
    _setdef(o, "weight_decay", 0.01)
    _setdef(o, "beta1", 0.9)
    _setdef(o, "beta2", 0.95)
    _setdef(o, "epsilon", 1e-8)
    _setdef(o, "multi_precision", True)
    _setdef(o, "tensor_fusion", True)
    lr = _setdef(o, "lr", AttrDict())
    _setdef(lr, "name", "CosineAnnealingWithWarmupDecay")
    _setdef(lr, "max_lr", 1e-4)
    _setdef(lr, "min_lr", 1e-5)
    _setdef(lr, "warmup_rate", 0.01)
    _setdef(lr, "decay_steps", 360000)


def get_config(fname: str, overrides: Optional[List[str]] = None,
               show: bool = False) -> AttrDict:
    """Load + merge + override + derive. Reference: config.py:398-415."""
    cfg = _to_attrdict(parse_yaml_with_base(fname))
    for k, v in parse_args_overrides(overrides):
        apply_override(cfg, k, v)
    process_dist_config(cfg)
    process_global_configs(cfg)
    process_engine_config(cfg)
    process_model_configs(cfg)
    process_optimizer_configs(cfg)
    if show:
        print_config(cfg)
    return cfg


def print_config(cfg: Dict, indent: int = 0) -> None:
    from paddlefleetx_amd.utils.log import logger
    if indent == 0:
        logger.info("----------- Configuration -----------")
    for k, v in cfg.items():
        if isinstance(v, dict):
            logger.info("  " * indent + f"{k}:")
            print_config(v, indent + 1)
        else:
            logger.info("  " * indent + f"{k}: {v}")
    if indent == 0:
        logger.info("-------------------------------------")
