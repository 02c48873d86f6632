from paddlefleetx_amd.models.gpt.model import (GPTForPretraining, GPTModel,
                                               GPTPretrainingCriterion)

__all__ = ["GPTModel", "GPTForPretraining", "GPTPretrainingCriterion"]
