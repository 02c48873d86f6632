from paddlefleetx_amd.models.protein_folding.evoformer import (
    EvoformerIteration, EvoformerStack, GatedAttention, MSAColumnAttention,
    MSARowAttentionWithPairBias, OuterProductMean, Transition,
    TriangleAttention, TriangleMultiplication)

__all__ = [
    "GatedAttention", "MSARowAttentionWithPairBias", "MSAColumnAttention",
    "Transition", "OuterProductMean", "TriangleMultiplication",
    "TriangleAttention", "EvoformerIteration", "EvoformerStack",
]
