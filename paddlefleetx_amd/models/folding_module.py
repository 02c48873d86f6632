"""Protein-folding training module: Evoformer trunk + structure module.

Completes the reference's protein_folding package (which the reference
drives through its own runner) as a first-class BasicModule so
tools/train.py runs it from a YAML config: msa/pair trunk ->
single-representation head -> structure module -> FAPE + torsion losses
against target atom37 coordinates.
"""

from __future__ import annotations

from typing import Dict

import torch
import torch.nn as nn

from paddlefleetx_amd.core.module import BasicModule
from paddlefleetx_amd.models.protein_folding import all_atom
from paddlefleetx_amd.models.protein_folding.evoformer import EvoformerStack
from paddlefleetx_amd.models.protein_folding.structure_module import (
    StructureModule)


class FoldingModel(nn.Module):
    """Evoformer + structure module (heads sized by the config)."""

    def __init__(self, msa_dim: int = 64, pair_dim: int = 64,
                 single_dim: int = 128, num_evoformer_blocks: int = 4,
                 num_structure_layers: int = 4, num_heads: int = 4,
                 head_dim: int = 16, msa_vocab: int = 23):
        super().__init__()
        self.msa_embed = nn.Embedding(msa_vocab, msa_dim)
        self.pair_embed = nn.Linear(2 * msa_dim, pair_dim)
        self.evoformer = EvoformerStack(num_evoformer_blocks, msa_dim,
                                        pair_dim, num_heads, head_dim)
        self.single_head = nn.Linear(msa_dim, single_dim)
        self.structure = StructureModule(single_dim, pair_dim,
                                         num_structure_layers)

    def forward(self, msa_tokens: torch.Tensor) -> Dict[str, object]:
        """msa_tokens [B, S, N] int -> frames + torsion angles."""
        msa = self.msa_embed(msa_tokens)               # [B, S, N, Cm]
        q = msa[:, 0]                                  # query sequence
        pair = self.pair_embed(torch.cat([
            q[:, :, None, :].expand(-1, -1, q.shape[1], -1),
            q[:, None, :, :].expand(-1, q.shape[1], -1, -1)], dim=-1))
        msa, pair = self.evoformer(msa, pair)
        single = self.single_head(msa[:, 0])           # [B, N, Cs]
        out = self.structure(single, pair)
        return out


class FoldingCriterion(nn.Module):
    """FAPE on the backbone frames + torsion-angle loss
    (all_atom.frame_aligned_point_error / torsion_angle_loss)."""

    def forward(self, out: Dict, aatype, atom37_pos, atom37_mask):
        target_frames, frame_mask = all_atom.backbone_frames(
            atom37_pos, atom37_mask)
        ca = atom37_pos[..., 1, :]  # CA
        ca_mask = atom37_mask[..., 1]
        fape = 0.0
        for frames in out["traj"]:
            fape = fape + all_atom.frame_aligned_point_error(
                frames, target_frames, frame_mask, frames.trans, ca, ca_mask)
        fape = fape.mean() / len(out["traj"])
        tors = all_atom.atom37_to_torsion_angles(aatype, atom37_pos,
                                                 atom37_mask)
        t_loss = all_atom.torsion_angle_loss(
            out["angles_sin_cos"], tors["torsion_angles_sin_cos"],
            tors["alt_torsion_angles_sin_cos"], tors["torsion_angles_mask"])
        return fape + t_loss


class FoldingModule(BasicModule):
    def __init__(self, configs):
        super().__init__(configs)

    def get_model(self):
        mcfg = dict(self.configs["Model"])
        kw = {k: mcfg[k] for k in
              ("msa_dim", "pair_dim", "single_dim", "num_evoformer_blocks",
               "num_structure_layers", "num_heads", "head_dim")
              if k in mcfg}
        return FoldingModel(**kw)

    def get_loss_fn(self):
        return FoldingCriterion()

    def training_step(self, batch):
        msa_tokens, aatype, atom37_pos, atom37_mask = batch
        out = self(msa_tokens)
        return self.loss_fn(out, aatype, atom37_pos, atom37_mask)

    def validation_step(self, batch):
        with torch.no_grad():
            return self.training_step(batch)
