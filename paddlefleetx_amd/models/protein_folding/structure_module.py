"""Structure module: invariant point attention + iterative backbone
updates + side-chain torsion head (AlphaFold2 Algorithm 20-23).

The reference repo stops at the Evoformer trunk; this completes the
folding head so the all-atom geometry (geometry.py / all_atom.py) is
exercised end to end: single [*, N, Cs] + pair [*, N, N, Cz] ->
per-residue backbone rigids + 7 torsion sin/cos, trained with FAPE and
the torsion-angle loss.
"""

from __future__ import annotations

import math
from typing import Dict, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from paddlefleetx_amd.models.protein_folding.geometry import Rigid


class InvariantPointAttention(nn.Module):
    """IPA (AF2 Algorithm 22): scalar qk + point qk in the global frame
    + pair bias, attention over residues."""

    def __init__(self, c_s: int, c_z: int, c_hidden: int = 16,
                 num_heads: int = 12, num_qk_points: int = 4,
                 num_v_points: int = 8):
        super().__init__()
        self.h = num_heads
        self.c = c_hidden
        self.pq = num_qk_points
        self.pv = num_v_points
        hc = num_heads * c_hidden
        self.q = nn.Linear(c_s, hc, bias=False)
        self.kv = nn.Linear(c_s, 2 * hc, bias=False)
        self.q_pts = nn.Linear(c_s, num_heads * num_qk_points * 3)
        self.kv_pts = nn.Linear(
            c_s, num_heads * (num_qk_points + num_v_points) * 3)
        self.pair_bias = nn.Linear(c_z, num_heads, bias=False)
        self.head_weights = nn.Parameter(torch.zeros(num_heads))
        out_in = num_heads * (c_z + c_hidden + num_v_points * 4)
        self.out = nn.Linear(out_in, c_s)

    def forward(self, s: torch.Tensor, z: torch.Tensor, frames: Rigid,
                mask: Optional[torch.Tensor] = None) -> torch.Tensor:
        assert s.dim() == 3, "expects [B, N, c_s]"
        *lead, N, _ = s.shape
        h, c, pq, pv = self.h, self.c, self.pq, self.pv
        q = self.q(s).view(*lead, N, h, c)
        k, v = self.kv(s).view(*lead, N, h, 2 * c).split(c, dim=-1)

        # points: predicted in the local frame, attended in the global
        q_pts = self.q_pts(s).view(*lead, N, h * pq, 3)
        q_pts = frames[..., None].apply(q_pts).view(*lead, N, h, pq, 3)
        kv_pts = self.kv_pts(s).view(*lead, N, h * (pq + pv), 3)
        kv_pts = frames[..., None].apply(kv_pts).view(*lead, N, h, pq + pv, 3)
        k_pts, v_pts = kv_pts.split([pq, pv], dim=-2)

        # [*, h, N, N] scalar logits
        att = torch.einsum("...ihc,...jhc->...hij", q, k) \
            * math.sqrt(1.0 / (3 * c))
        att = att + math.sqrt(1.0 / 3) * \
            self.pair_bias(z).permute(*range(len(lead)), 3, 1, 2)
        # point distances
        d2 = ((q_pts[..., :, None, :, :, :] -
               k_pts[..., None, :, :, :, :]) ** 2).sum(-1)  # [*, i, j, h, pq]
        gamma = F.softplus(self.head_weights)
        wc = math.sqrt(2.0 / (9 * pq)) / 2.0
        att_pts = (d2.sum(-1) * gamma * wc * math.sqrt(1.0 / 3))
        att = att - att_pts.permute(*range(len(lead)), 3, 1, 2)
        if mask is not None:
            att = att.masked_fill(~mask[..., None, None, :].bool(), -1e9)
        att = att.softmax(dim=-1)

        # outputs: scalar, point (back to local frame + norm), pair
        o_sc = torch.einsum("...hij,...jhc->...ihc", att, v)
        o_pt = torch.einsum("...hij,...jhpx->...ihpx", att, v_pts)
        o_pt_local = frames[..., None, None].invert_apply(o_pt)
        o_pt_norm = torch.linalg.norm(o_pt_local, dim=-1, keepdim=True)
        o_pair = torch.einsum("...hij,...ijz->...ihz", att, z)
        o = torch.cat([o_sc.flatten(-2),
                       o_pt_local.flatten(-3),
                       o_pt_norm.flatten(-3),
                       o_pair.flatten(-2)], dim=-1)
        return self.out(o)


class AngleResnet(nn.Module):
    """Torsion head: 2 residual blocks -> 7 (sin, cos) pairs."""

    def __init__(self, c_s: int, c_hidden: int = 128, num_angles: int = 7):
        super().__init__()
        self.lin_in = nn.Linear(c_s * 2, c_hidden)
        self.blocks = nn.ModuleList([
            nn.Sequential(nn.ReLU(), nn.Linear(c_hidden, c_hidden),
                          nn.ReLU(), nn.Linear(c_hidden, c_hidden))
            for _ in range(2)])
        self.out = nn.Linear(c_hidden, num_angles * 2)
        self.num_angles = num_angles

    def forward(self, s, s_init):
        x = self.lin_in(torch.cat([F.relu(s), F.relu(s_init)], dim=-1))
        for blk in self.blocks:
            x = x + blk(x)
        ang = self.out(F.relu(x))
        return ang.view(*ang.shape[:-1], self.num_angles, 2)


class StructureModule(nn.Module):
    """Iterative backbone refinement (AF2 Algorithm 20): starts from the
    identity ("black-hole") frames; each of `num_layers` shared-weight
    iterations runs IPA, a transition, and a 6-DoF frame update
    (quaternion-vector + translation, geometry.Rigid.pre_compose)."""

    def __init__(self, c_s: int = 128, c_z: int = 64, num_layers: int = 8,
                 dropout: float = 0.0, position_scale: float = 10.0):
        super().__init__()
        self.num_layers = num_layers
        self.position_scale = position_scale
        self.norm_s = nn.LayerNorm(c_s)
        self.norm_z = nn.LayerNorm(c_z)
        self.init_proj = nn.Linear(c_s, c_s)
        self.ipa = InvariantPointAttention(c_s, c_z)
        self.norm_ipa = nn.LayerNorm(c_s)
        self.transition = nn.Sequential(
            nn.Linear(c_s, c_s), nn.ReLU(), nn.Linear(c_s, c_s), nn.ReLU(),
            nn.Linear(c_s, c_s))
        self.norm_trans = nn.LayerNorm(c_s)
        self.frame_update = nn.Linear(c_s, 6)
        nn.init.zeros_(self.frame_update.weight)
        nn.init.zeros_(self.frame_update.bias)
        self.angle_head = AngleResnet(c_s)
        self.dropout = dropout

    def forward(self, s: torch.Tensor, z: torch.Tensor,
                mask: Optional[torch.Tensor] = None) -> Dict[str, object]:
        *lead, N, _ = s.shape
        s_init = self.norm_s(s)
        z = self.norm_z(z)
        s = self.init_proj(s_init)
        frames = Rigid.identity((*lead, N), device=s.device, dtype=s.dtype)
        traj = []
        for _ in range(self.num_layers):
            s = s + self.ipa(s, z, frames.stop_rot_gradient(), mask)
            s = self.norm_ipa(F.dropout(s, self.dropout, self.training))
            s = self.norm_trans(s + self.transition(s))
            frames = frames.pre_compose(self.frame_update(s))
            traj.append(frames.scale_translation(self.position_scale))
        angles = self.angle_head(s, s_init)
        return {
            "frames": traj[-1],
            "traj": traj,
            "angles_sin_cos": angles,
            "single": s,
        }
