"""Lightning-style module contract (reference core/module/basic_module.py:29-86)."""

from __future__ import annotations

from typing import Any, Optional

import torch.nn as nn


class BasicModule(nn.Module):
    """Subclasses implement get_model/get_loss_fn + the step hooks."""

    def __init__(self, configs=None):
        super().__init__()
        self.configs = configs
        self.global_step = 0
        self.model: Optional[nn.Module] = self.get_model()
        self.loss_fn = self.get_loss_fn()

    # --- construction hooks ---
    def get_model(self) -> nn.Module:
        raise NotImplementedError

    def get_loss_fn(self):
        return None

    # --- data hooks ---
    def pretreating_batch(self, batch):
        return batch

    # --- step hooks ---
    def forward(self, *args, **kwargs):
        return self.model(*args, **kwargs)

    def training_step(self, batch) -> Any:
        raise NotImplementedError

    def training_step_end(self, log_dict):
        pass

    def validation_step(self, batch) -> Any:
        raise NotImplementedError

    def validation_step_end(self, log_dict):
        pass

    def test_step(self, batch) -> Any:
        return self.validation_step(batch)

    def test_step_end(self, log_dict):
        return self.validation_step_end(log_dict)

    def backward(self, loss):
        loss.backward()

    def input_spec(self):
        return None
