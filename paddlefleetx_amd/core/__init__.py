from paddlefleetx_amd.core.engine import BasicEngine, EagerEngine
from paddlefleetx_amd.core.module import BasicModule

__all__ = ["BasicEngine", "EagerEngine", "BasicModule"]
