"""Rigid-body geometry for the folding model — all-atom math.

Covers the reference's r3.py (Vecs/Rots/Rigids algebra,
ppfleetx/models/protein_folding/r3.py:44-518) and quat_affine.py
(quaternion affines, quat_affine.py:69-386) with an idiomatic torch
design: rotations are plain [..., 3, 3] tensors and translations
[..., 3] tensors held in a `Rigid` value class, instead of the
reference's namedtuple-of-9-scalars layout (which existed to dodge
Paddle's slicing overhead). All ops broadcast and differentiate.
"""

from __future__ import annotations

from typing import Tuple

import torch

# quaternion multiply coefficients: QUAT_MULTIPLY[i,j,k] is the (i,j)->k
# term of the Hamilton product (reference quat_affine.py:342-384)
_QUAT_MULTIPLY = torch.zeros(4, 4, 4)
_QUAT_MULTIPLY[:, :, 0] = torch.tensor([[1, 0, 0, 0], [0, -1, 0, 0],
                                        [0, 0, -1, 0], [0, 0, 0, -1]],
                                       dtype=torch.float32)
_QUAT_MULTIPLY[:, :, 1] = torch.tensor([[0, 1, 0, 0], [1, 0, 0, 0],
                                        [0, 0, 0, 1], [0, 0, -1, 0]],
                                       dtype=torch.float32)
_QUAT_MULTIPLY[:, :, 2] = torch.tensor([[0, 0, 1, 0], [0, 0, 0, -1],
                                        [1, 0, 0, 0], [0, 1, 0, 0]],
                                       dtype=torch.float32)
_QUAT_MULTIPLY[:, :, 3] = torch.tensor([[0, 0, 0, 1], [0, 0, 1, 0],
                                        [0, -1, 0, 0], [1, 0, 0, 0]],
                                       dtype=torch.float32)
_QUAT_MULTIPLY_BY_VEC = _QUAT_MULTIPLY[:, 1:]


def quat_to_rot(quat: torch.Tensor) -> torch.Tensor:
    """Normalized quaternion [..., 4] (w, x, y, z) -> rotation [..., 3, 3].

    Reference quat_affine.py:116-129.
    """
    w, x, y, z = quat.unbind(-1)
    two = 2.0
    rot = torch.stack([
        1 - two * (y * y + z * z), two * (x * y - w * z), two * (x * z + w * y),
        two * (x * y + w * z), 1 - two * (x * x + z * z), two * (y * z - w * x),
        two * (x * z - w * y), two * (y * z + w * x), 1 - two * (x * x + y * y),
    ], dim=-1)
    return rot.view(*quat.shape[:-1], 3, 3)


def rot_to_quat(rot: torch.Tensor) -> torch.Tensor:
    """Rotation [..., 3, 3] -> quaternion [..., 4] via the symmetric
    K-matrix eigen decomposition (robust for all rotations; reference
    quat_affine.py:69-113)."""
    xx, xy, xz = rot[..., 0, 0], rot[..., 0, 1], rot[..., 0, 2]
    yx, yy, yz = rot[..., 1, 0], rot[..., 1, 1], rot[..., 1, 2]
    zx, zy, zz = rot[..., 2, 0], rot[..., 2, 1], rot[..., 2, 2]
    k = torch.stack([
        torch.stack([xx + yy + zz, zy - yz, xz - zx, yx - xy], dim=-1),
        torch.stack([zy - yz, xx - yy - zz, xy + yx, xz + zx], dim=-1),
        torch.stack([xz - zx, xy + yx, yy - xx - zz, yz + zy], dim=-1),
        torch.stack([yx - xy, xz + zx, yz + zy, zz - xx - yy], dim=-1),
    ], dim=-2) / 3.0
    _, vecs = torch.linalg.eigh(k.float())
    quat = vecs[..., -1]  # eigenvector of the largest eigenvalue
    # canonical sign: w >= 0
    quat = quat * torch.where(quat[..., :1] < 0, -1.0, 1.0)
    return quat.to(rot.dtype)


def quat_multiply(q1: torch.Tensor, q2: torch.Tensor) -> torch.Tensor:
    m = _QUAT_MULTIPLY.to(device=q1.device, dtype=q1.dtype)
    return torch.einsum("...i,ijk,...j->...k", q1, m, q2)


def quat_multiply_by_vec(quat: torch.Tensor, vec: torch.Tensor) -> torch.Tensor:
    """quat * (0, v) for the pre_compose update (quat_affine.py:131-137)."""
    m = _QUAT_MULTIPLY_BY_VEC.to(device=quat.device, dtype=quat.dtype)
    return torch.einsum("...i,ijk,...j->...k", quat, m, vec)


def rots_from_two_vecs(e0: torch.Tensor, e1: torch.Tensor) -> torch.Tensor:
    """Gram-Schmidt rotation from two unnormalized vectors [..., 3]:
    e0 -> x axis, e1's orthogonal part -> y, cross -> z
    (reference r3.py:380-406)."""
    e0 = e0 / torch.linalg.norm(e0, dim=-1, keepdim=True).clamp_min(1e-8)
    c = (e1 * e0).sum(-1, keepdim=True)
    e1 = e1 - c * e0
    e1 = e1 / torch.linalg.norm(e1, dim=-1, keepdim=True).clamp_min(1e-8)
    e2 = torch.cross(e0, e1, dim=-1)
    return torch.stack([e0, e1, e2], dim=-1)  # columns are the axes


class Rigid:
    """A rigid transform: rot [..., 3, 3] + trans [..., 3].

    Equivalent surface to the reference's Rigids (r3.py:193-368) and
    QuatAffine (quat_affine.py:190-341).
    """

    __slots__ = ("rot", "trans")

    def __init__(self, rot: torch.Tensor, trans: torch.Tensor):
        self.rot = rot
        self.trans = trans

    # --- constructors ---
    @staticmethod
    def identity(shape, device=None, dtype=torch.float32) -> "Rigid":
        rot = torch.eye(3, device=device, dtype=dtype).expand(*shape, 3, 3)
        trans = torch.zeros(*shape, 3, device=device, dtype=dtype)
        return Rigid(rot.contiguous(), trans)

    @staticmethod
    def from_3_points(p_neg_x: torch.Tensor, origin: torch.Tensor,
                      p_xy: torch.Tensor) -> "Rigid":
        """Frame from three points (N, CA, C for the backbone):
        origin at `origin`, x axis toward p_xy... matches the reference's
        rigids_from_3_points (r3.py:231-275): e0 = p_xy - origin,
        e1 = p_neg_x - origin."""
        e0 = p_xy - origin
        e1 = p_neg_x - origin
        rot = rots_from_two_vecs(e0, e1)
        return Rigid(rot, origin)

    @staticmethod
    def from_tensor_4x4(m: torch.Tensor) -> "Rigid":
        return Rigid(m[..., :3, :3], m[..., :3, 3])

    @staticmethod
    def from_tensor_flat12(m: torch.Tensor) -> "Rigid":
        """[..., 12] = 9 rot (row major) + 3 trans (r3.py:315-319)."""
        return Rigid(m[..., :9].view(*m.shape[:-1], 3, 3), m[..., 9:])

    @staticmethod
    def from_quat_and_trans(quat: torch.Tensor, trans: torch.Tensor,
                            normalize: bool = True) -> "Rigid":
        if normalize:
            quat = quat / torch.linalg.norm(quat, dim=-1,
                                            keepdim=True).clamp_min(1e-12)
        return Rigid(quat_to_rot(quat), trans)

    # --- algebra ---
    def compose(self, other: "Rigid") -> "Rigid":
        """self ∘ other (apply other first; r3.py:322-327)."""
        rot = torch.matmul(self.rot, other.rot)
        trans = self.apply(other.trans)
        return Rigid(rot, trans)

    def invert(self) -> "Rigid":
        inv_rot = self.rot.transpose(-1, -2)
        inv_trans = -torch.einsum("...ij,...j->...i", inv_rot, self.trans)
        return Rigid(inv_rot, inv_trans)

    def apply(self, point: torch.Tensor) -> torch.Tensor:
        """Rotate+translate points [..., 3] (r3.py:334-337)."""
        return torch.einsum("...ij,...j->...i", self.rot, point) + self.trans

    def invert_apply(self, point: torch.Tensor) -> torch.Tensor:
        return torch.einsum("...ji,...j->...i", self.rot,
                            point - self.trans)

    def pre_compose(self, update: torch.Tensor) -> "Rigid":
        """QuatAffine.pre_compose (quat_affine.py:259-280): update
        [..., 6] = 3 quaternion vector components + 3 translation."""
        quat = rot_to_quat(self.rot)
        vec, t_upd = update[..., :3], update[..., 3:]
        new_quat = quat + quat_multiply_by_vec(quat, vec)
        new_quat = new_quat / torch.linalg.norm(
            new_quat, dim=-1, keepdim=True).clamp_min(1e-12)
        new_rot = quat_to_rot(new_quat)
        new_trans = self.apply(t_upd)
        return Rigid(new_rot, new_trans)

    def scale_translation(self, s: float) -> "Rigid":
        return Rigid(self.rot, self.trans * s)

    def stop_rot_gradient(self) -> "Rigid":
        return Rigid(self.rot.detach(), self.trans)

    # --- conversions / tensor protocol ---
    def to_tensor_4x4(self) -> torch.Tensor:
        m = torch.zeros(*self.trans.shape[:-1], 4, 4,
                        device=self.trans.device, dtype=self.trans.dtype)
        m[..., :3, :3] = self.rot
        m[..., :3, 3] = self.trans
        m[..., 3, 3] = 1.0
        return m

    def to_tensor_flat12(self) -> torch.Tensor:
        return torch.cat([self.rot.reshape(*self.rot.shape[:-2], 9),
                          self.trans], dim=-1)

    def __getitem__(self, idx) -> "Rigid":
        """Index the BATCH dims (the trailing 3x3 / 3 stay intact), so
        `frames[..., None]` unsqueezes a broadcast dim like on a [*, 3]
        tensor of translations."""
        if not isinstance(idx, tuple):
            idx = (idx,)
        return Rigid(self.rot[idx + (slice(None), slice(None))],
                     self.trans[idx + (slice(None),)])

    @property
    def shape(self):
        return self.trans.shape[:-1]

    def map(self, fn) -> "Rigid":
        return Rigid(fn(self.rot), fn(self.trans))

    def detach(self) -> "Rigid":
        return Rigid(self.rot.detach(), self.trans.detach())
