"""Module factory (reference ppfleetx/models/__init__.py:31-35 build_module)."""

from __future__ import annotations

from paddlefleetx_amd.utils.log import logger


def build_module(config):
    # vision configs name the module under Model.module (vit base.yaml),
    # language configs under Model.name
    name = config["Model"].get("module") or config["Model"]["name"]
    from paddlefleetx_amd.models.language_module import (GPTFinetuneModule,
                                                         GPTGenerationModule,
                                                         GPTModule)
    table = {
        "GPTModule": GPTModule,
        "GPTGenerationModule": GPTGenerationModule,
        "GPTFinetuneModule": GPTFinetuneModule,
    }
    # late registrations to avoid importing every family eagerly
    if name == "MoEModule":
        from paddlefleetx_amd.models.moe_module import MoEModule
        table["MoEModule"] = MoEModule
    if name == "GPTEvalModule":
        from paddlefleetx_amd.models.eval_module import GPTEvalModule
        table["GPTEvalModule"] = GPTEvalModule
    if name == "ViTModule" or name == "GeneralClsModule":
        from paddlefleetx_amd.models.vit_module import GeneralClsModule
        table["ViTModule"] = GeneralClsModule
        table["GeneralClsModule"] = GeneralClsModule
    if name == "ImagenModule":
        from paddlefleetx_amd.models.imagen_module import ImagenModule
        table["ImagenModule"] = ImagenModule
    if name == "MOCOModule":
        from paddlefleetx_amd.models.moco import MOCOModule
        table["MOCOModule"] = MOCOModule
    if name == "FoldingModule":
        from paddlefleetx_amd.models.folding_module import FoldingModule
        table["FoldingModule"] = FoldingModule
    if name in ("ErnieModule", "ErnieSeqClsModule"):
        from paddlefleetx_amd.models.ernie_module import (ErnieModule,
                                                          ErnieSeqClsModule)
        table["ErnieModule"] = ErnieModule
        table["ErnieSeqClsModule"] = ErnieSeqClsModule
    if name not in table:
        raise ValueError(f"unknown module {name}")
    logger.info(f"building module {name}")
    import torch
    if torch.cuda.is_available():
        # construct parameters directly on the GPU: CPU-side init of a 6.7B
        # model costs ~100 s; on-device init is seconds
        with torch.device("cuda"):
            return table[name](config)
    return table[name](config)
