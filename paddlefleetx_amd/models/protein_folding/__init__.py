from paddlefleetx_amd.models.protein_folding.evoformer import (
    EvoformerIteration, EvoformerStack, GatedAttention, MSAColumnAttention,
    MSARowAttentionWithPairBias, OuterProductMean, Transition,
    TriangleAttention, TriangleMultiplication)

from paddlefleetx_amd.models.protein_folding.geometry import (  # noqa: F401
    Rigid, quat_multiply, quat_to_rot, rot_to_quat, rots_from_two_vecs)
from paddlefleetx_amd.models.protein_folding.structure_module import (  # noqa: F401
    AngleResnet, InvariantPointAttention, StructureModule)
from paddlefleetx_amd.models.protein_folding.template import (  # noqa: F401
    SingleTemplateEmbedding, TemplateEmbedding, TemplatePair,
    dgram_from_positions)

__all__ = [
    "GatedAttention", "MSARowAttentionWithPairBias", "MSAColumnAttention",
    "Transition", "OuterProductMean", "TriangleMultiplication",
    "TriangleAttention", "EvoformerIteration", "EvoformerStack",
    "Rigid", "rot_to_quat", "quat_to_rot", "quat_multiply",
    "rots_from_two_vecs", "InvariantPointAttention", "StructureModule",
    "AngleResnet", "TemplatePair", "SingleTemplateEmbedding",
    "TemplateEmbedding", "dgram_from_positions",
]
