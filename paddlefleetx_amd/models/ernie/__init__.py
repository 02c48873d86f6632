from paddlefleetx_amd.models.ernie.model import (ErnieEmbeddings,
                                                 ErnieEncoderLayer,
                                                 ErnieForPretraining,
                                                 ErnieForSequenceClassification,
                                                 ErnieModel, ErniePooler,
                                                 ErniePretrainingCriterion,
                                                 ErniePretrainingHeads)

__all__ = [
    "ErnieModel", "ErnieEmbeddings", "ErniePooler", "ErnieEncoderLayer",
    "ErnieForPretraining", "ErniePretrainingHeads", "ErniePretrainingCriterion",
    "ErnieForSequenceClassification",
]
