from paddlefleetx_amd.models.moe.gate import (GShardGate, NaiveGate,
                                              SwitchGate, build_gate)
from paddlefleetx_amd.models.moe.moe_layer import ExpertLayer, MoELayer

__all__ = ["NaiveGate", "GShardGate", "SwitchGate", "build_gate",
           "MoELayer", "ExpertLayer"]
