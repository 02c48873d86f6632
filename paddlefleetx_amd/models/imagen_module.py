"""Imagen training module.

Reference: ppfleetx ImagenModule (multimodal configs) — batch =
(images, text_ids, text_mask); loss = diffusion eps-MSE.
"""

from __future__ import annotations

import torch

from paddlefleetx_amd.core.module import BasicModule
from paddlefleetx_amd.utils.log import logger


class ImagenModule(BasicModule):
    def get_model(self):
        mcfg = dict(self.configs["Model"])
        for k in ("name", "module"):
            mcfg.pop(k, None)
        unet_name = mcfg.pop("unet_name", None)
        unet = None
        if unet_name:
            from paddlefleetx_amd.models import imagen as I
            ukw = dict(mcfg.pop("unet_kwargs", {}))
            ukw.setdefault("text_embed_dim", mcfg.get("text_embed_dim", 512))
            unet = getattr(I, unet_name)(**ukw)
        from paddlefleetx_amd.models.imagen import ImagenModel
        return ImagenModel(unet=unet, **mcfg)

    def get_loss_fn(self):
        return None  # loss computed inside ImagenModel.forward

    def training_step(self, batch):
        images, text_ids, text_mask = batch[:3]
        return self.model(images, text_ids=text_ids, text_mask=text_mask)

    def validation_step(self, batch):
        return self.training_step(batch)

    def training_step_end(self, log_dict):
        logger.train("[train] epoch: %d, batch: %d, loss: %.9f, "
                     "avg_batch_cost: %.5f sec"
                     % (log_dict["epoch"], log_dict["batch"],
                        log_dict["loss"], log_dict["train_cost"]))
