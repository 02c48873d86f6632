"""4D hybrid-parallel topology over torch.distributed (RCCL on ROCm).

Replaces the reference's Hybrid Communicate Group
(ppfleetx/distributed/apis/comm_groups.py:27-153, paddle fleet HCG) with a
native mesh: rank grid [pp, dp, sharding, mp] with mp innermost so TP
collectives stay on adjacent GPUs of the xGMI mesh.

MoE fuses dp x mp into the expert-parallel group
(comm_groups.py:125-153 HybridCommGroupForMoE).
"""

from __future__ import annotations

import os
from typing import List, Optional

import torch
import torch.distributed as dist

__all__ = ["HybridTopology"]


class CommGroupInfo:
    """One parallel axis: this rank's group / rank / world / peer list."""

    def __init__(self, group, ranks: List[int], axis: str):
        self.group = group
        self.ranks = ranks
        self.axis = axis
        self.world_size = len(ranks)
        myrank = dist.get_rank() if dist.is_initialized() else 0
        self.rank = ranks.index(myrank) if myrank in ranks else -1

    def __repr__(self):
        return f"CommGroup({self.axis}, ranks={self.ranks}, rank={self.rank})"


class _SingleGroup:
    """Degenerate axis of size 1 (no collective needed)."""

    def __init__(self, axis: str):
        self.group = None
        self.axis = axis
        self.ranks = [dist.get_rank() if dist.is_initialized() else 0]
        self.world_size = 1
        self.rank = 0


class HybridTopology:
    """Rank grid [pp, dp, sharding, cp, mp] (mp fastest-varying).

    rank = (((pp_i * dp + dp_i) * sd + sd_i) * cp + cp_i) * mp + mp_i

    cp is the context-parallel (Ulysses sequence) axis — the MI355X-native
    long-context generalization of the reference's DAP all-to-all axis swap
    (protein_folding/dap.py:244-379); cp=1 reproduces the reference's 4D
    topology exactly.
    """

    AXES = ("pp", "dp", "sharding", "cp", "mp")

    def __init__(self, dp: int = 1, mp: int = 1, pp: int = 1, sharding: int = 1,
                 cp: int = 1, moe_expert_parallel: bool = False):
        self.dp_degree, self.mp_degree = dp, mp
        self.pp_degree, self.sharding_degree = pp, sharding
        self.cp_degree = cp
        world = dist.get_world_size() if dist.is_initialized() else 1
        assert dp * mp * pp * sharding * cp == world, (
            f"dp{dp}*mp{mp}*pp{pp}*sharding{sharding}*cp{cp} != world {world}")
        self.world_size = world
        self.global_rank = dist.get_rank() if dist.is_initialized() else 0

        r = self.global_rank
        self.mp_rank = r % mp
        self.cp_rank = (r // mp) % cp
        self.sharding_rank = (r // (mp * cp)) % sharding
        self.dp_rank = (r // (mp * cp * sharding)) % dp
        self.pp_rank = r // (mp * cp * sharding * dp)

        self._groups = {}
        if world == 1 or not dist.is_initialized():
            for ax in self.AXES + ("data_world", "mp_sharding"):
                self._groups[ax] = _SingleGroup(ax)
            self.ep_group = _SingleGroup("ep")
            return

        # Build groups for each axis: vary that axis, fix the others.
        # Every rank executes every new_group call in identical order.
        def build(axis_sizes, varying):
            """varying: index into (pp, dp, sd, cp, mp) grid dims to vary."""
            pp_, dp_, sd_, cp_, mp_ = axis_sizes
            groups = []
            import itertools
            dims = [range(pp_), range(dp_), range(sd_), range(cp_),
                    range(mp_)]
            fixed_dims = [d for i, d in enumerate(dims) if i not in varying]
            for fixed in itertools.product(*fixed_dims):
                ranks = []
                vary_dims = [dims[i] for i in varying]
                for vv in itertools.product(*vary_dims):
                    coord = [0, 0, 0, 0, 0]
                    fi, vi = 0, 0
                    for i in range(5):
                        if i in varying:
                            coord[i] = vv[vi]; vi += 1
                        else:
                            coord[i] = fixed[fi]; fi += 1
                    rank = (((coord[0] * dp_ + coord[1]) * sd_ + coord[2])
                            * cp_ + coord[3]) * mp_ + coord[4]
                    ranks.append(rank)
                groups.append(ranks)
            return groups

        sizes = (pp, dp, sharding, cp, mp)
        axis_to_vary = {"pp": (0,), "dp": (1,), "sharding": (2,), "cp": (3,),
                        "mp": (4,),
                        "data_world": (1, 2),  # dp x sharding: batch sampler replicas
                        "mp_sharding": (2, 4)}
        myrank = self.global_rank
        for ax, varying in axis_to_vary.items():
            deg = 1
            for i in varying:
                deg *= sizes[i]
            if deg == 1:
                self._groups[ax] = _SingleGroup(ax)
                continue
            mine = None
            for ranks in build(sizes, varying):
                g = dist.new_group(ranks=ranks)
                if myrank in ranks:
                    mine = CommGroupInfo(g, ranks, ax)
            assert mine is not None
            self._groups[ax] = mine

        # Expert-parallel group = dp x mp fused (reference comm_groups.py:125-153)
        if moe_expert_parallel:
            assert pp == 1 and sharding == 1, "MoE EP requires pp==1, sharding==1"
            mine = None
            for ranks in build(sizes, (1, 4)):
                g = dist.new_group(ranks=ranks)
                if myrank in ranks:
                    mine = CommGroupInfo(g, ranks, "ep")
            self.ep_group = mine
        else:
            self.ep_group = self._groups["dp"] if dp > 1 else _SingleGroup("ep")

    # --- reference-parity accessors (env.py:101-108 get_hcg surface) ---
    def get_data_parallel_group(self): return self._groups["dp"]
    def get_model_parallel_group(self): return self._groups["mp"]
    def get_pipe_parallel_group(self): return self._groups["pp"]
    def get_sharding_parallel_group(self): return self._groups["sharding"]
    def get_data_world_group(self): return self._groups["data_world"]
    def get_expert_parallel_group(self): return self.ep_group

    def get_context_parallel_group(self): return self._groups["cp"]

    def get_data_parallel_rank(self): return self.dp_rank
    def get_model_parallel_rank(self): return self.mp_rank
    def get_pipe_parallel_rank(self): return self.pp_rank
    def get_sharding_parallel_rank(self): return self.sharding_rank
    def get_context_parallel_rank(self): return self.cp_rank

    def get_data_parallel_world_size(self): return self.dp_degree
    def get_model_parallel_world_size(self): return self.mp_degree
    def get_pipe_parallel_world_size(self): return self.pp_degree
    def get_sharding_parallel_world_size(self): return self.sharding_degree
    def get_context_parallel_world_size(self): return self.cp_degree

    def get_data_world_size(self):
        """dp x sharding: number of data-loader replicas (env.py:158-166)."""
        return self.dp_degree * self.sharding_degree

    def get_data_world_rank(self):
        return self.dp_rank * self.sharding_degree + self.sharding_rank

    # pipeline neighbors (global ranks)
    def pp_prev_rank(self) -> Optional[int]:
        if self.pp_rank == 0:
            return None
        return self._pp_global_rank(self.pp_rank - 1)

    def pp_next_rank(self) -> Optional[int]:
        if self.pp_rank == self.pp_degree - 1:
            return None
        return self._pp_global_rank(self.pp_rank + 1)

    def _pp_global_rank(self, pp_i: int) -> int:
        mp, sd, dp = self.mp_degree, self.sharding_degree, self.dp_degree
        return (((pp_i * dp + self.dp_rank) * sd + self.sharding_rank)
                * self.cp_degree + self.cp_rank) * mp + self.mp_rank

    def is_first_stage(self) -> bool:
        return self.pp_rank == 0

    def is_last_stage(self) -> bool:
        return self.pp_rank == self.pp_degree - 1

    def __repr__(self):
        return (f"HybridTopology(world={self.world_size}, dp={self.dp_degree}, "
                f"mp={self.mp_degree}, pp={self.pp_degree}, "
                f"sharding={self.sharding_degree}, rank={self.global_rank} -> "
                f"[pp{self.pp_rank} dp{self.dp_rank} sd{self.sharding_rank} mp{self.mp_rank}])")
