from paddlefleetx_amd.models.vit.vit import (ViT, ViT_6B_patch14_224,
                                             ViT_G_patch14_224,
                                             ViT_base_patch16_224,
                                             ViT_base_patch16_384,
                                             ViT_base_patch32_224,
                                             ViT_base_patch32_384,
                                             ViT_g_patch14_224,
                                             ViT_huge_patch14_224,
                                             ViT_huge_patch14_384,
                                             ViT_large_patch16_224,
                                             ViT_large_patch16_384,
                                             ViT_large_patch32_224,
                                             ViT_tiny_patch16_224,
                                             build_vit)

__all__ = [
    "ViT", "build_vit", "ViT_tiny_patch16_224", "ViT_base_patch16_224",
    "ViT_base_patch16_384", "ViT_base_patch32_224", "ViT_base_patch32_384",
    "ViT_large_patch16_224", "ViT_large_patch16_384", "ViT_large_patch32_224",
    "ViT_huge_patch14_224", "ViT_huge_patch14_384", "ViT_g_patch14_224",
    "ViT_G_patch14_224", "ViT_6B_patch14_224",
]
