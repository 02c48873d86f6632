from paddlefleetx_amd.models.imagen.modeling import (ImagenCascade,
                                                     ImagenCriterion,
                                                     ImagenModel)
from paddlefleetx_amd.models.imagen.unet import (BaseUnet64, SRUnet256,
                                                 SRUnet1024, Unet,
                                                 Unet64_397M)

__all__ = ["Unet", "Unet64_397M", "BaseUnet64", "SRUnet256", "SRUnet1024",
           "ImagenModel", "ImagenCriterion", "ImagenCascade"]
