"""Semi-auto-parallel annotation surface: ProcessMesh + shard_tensor.

Reference: the GPT auto model annotates every parallel weight with
`auto.shard_tensor(weight, mesh[idx], [None, mesh.mp])`
(ppfleetx/models/language_model/gpt/auto/auto_model.py:92-713) and the
mesh helper `process_mesh_config` (auto/auto_utils.py:24-108); paddle's
graph compiler then derives the placement. MI355X-native: annotations
are recorded ON the parameter (`p._dist_attr`), the mesh maps logical
axes to the HybridTopology groups, and `materialize_annotations`
verifies/derives the concrete sharding the eager TP/PP layers already
implement — the AutoEngine planner (core/auto_engine.py) consumes the
annotated model to pick degrees, so annotated models run through the
SAME runtime instead of a separate compiled program.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Tuple

import torch
import torch.nn as nn

__all__ = ["ProcessMesh", "shard_tensor", "get_dist_attr",
           "collect_annotations", "validate_against_topology"]


class ProcessMesh:
    """Logical device mesh, e.g. ProcessMesh([2, 2, 2],
    dim_names=("dp", "mp", "pp")). Axis names map to HybridTopology
    groups (reference auto_utils.py:24-108 process_mesh_config)."""

    def __init__(self, shape: Sequence[int],
                 dim_names: Sequence[str] = ("dp", "mp", "pp")):
        assert len(shape) == len(dim_names)
        self.shape = tuple(int(s) for s in shape)
        self.dim_names = tuple(dim_names)

    def size(self, name: str) -> int:
        return self.shape[self.dim_names.index(name)]

    def __repr__(self):
        dims = ", ".join(f"{n}={s}"
                         for n, s in zip(self.dim_names, self.shape))
        return f"ProcessMesh({dims})"


def shard_tensor(p: torch.Tensor, mesh: ProcessMesh,
                 dims_mapping: Sequence[Optional[str]]) -> torch.Tensor:
    """Annotate: dims_mapping[i] names the mesh axis tensor dim i is
    split over (None = replicated). Mirrors auto.shard_tensor
    (auto_model.py:92)."""
    assert len(dims_mapping) == p.dim(), (len(dims_mapping), p.dim())
    for name in dims_mapping:
        if name is not None:
            assert name in mesh.dim_names, name
    p._dist_attr = {"mesh": mesh, "dims_mapping": tuple(dims_mapping)}
    return p


def get_dist_attr(p: torch.Tensor) -> Optional[Dict]:
    return getattr(p, "_dist_attr", None)


def collect_annotations(model: nn.Module) -> Dict[str, Dict]:
    return {n: get_dist_attr(p) for n, p in model.named_parameters()
            if get_dist_attr(p) is not None}


def validate_against_topology(model: nn.Module) -> List[str]:
    """Check every annotation against the sharding the eager layers
    actually implement (partition_dim tags from parallel/tp.py): the
    compiled-program reference trusts annotations; here they must AGREE
    with the constructed layout. Returns a list of mismatch messages
    (empty = consistent)."""
    from paddlefleetx_amd.parallel.env import get_hcg
    mp_size = get_hcg().get_model_parallel_world_size()
    problems = []
    for name, p in model.named_parameters():
        attr = get_dist_attr(p)
        if attr is None:
            continue
        mapped = [i for i, d in enumerate(attr["dims_mapping"])
                  if d == "mp"]
        is_mp = bool(getattr(p, "is_mp", False)) and mp_size > 1
        if mp_size > 1:
            if is_mp:
                pdim = getattr(p, "partition_dim", None)
                if mapped != [pdim]:
                    problems.append(
                        f"{name}: annotated mp dims {mapped} but layer "
                        f"shards dim {pdim}")
            elif mapped:
                problems.append(
                    f"{name}: annotated mp-sharded on {mapped} but the "
                    "layer replicates it")
        # at mp==1 any annotation is legal (degenerate mesh)
    return problems


def annotate_gpt(model: nn.Module, mesh: ProcessMesh) -> int:
    """Attach the reference auto-model's annotation scheme to a GPT
    network built from our parallel layers (auto_model.py:92-713):
    vocab/column weights sharded on dim 0, row weights on dim 1."""
    count = 0
    for name, p in model.named_parameters():
        if p.dim() == 2 and getattr(p, "is_mp", False):
            dims = [None, None]
            dims[getattr(p, "partition_dim", 0)] = "mp"
            shard_tensor(p, mesh, dims)
            count += 1
        elif p.dim() == 1 and getattr(p, "is_mp", False):
            shard_tensor(p, mesh, ["mp"])
            count += 1
        else:
            shard_tensor(p, mesh, [None] * p.dim())
            count += 1
    return count
