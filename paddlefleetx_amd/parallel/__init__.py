from paddlefleetx_amd.parallel.env import (get_hcg, init_dist_env, set_seed,
                                           get_data_world_rank, get_data_world_size)
from paddlefleetx_amd.parallel.topology import HybridTopology

__all__ = ["get_hcg", "init_dist_env", "set_seed", "HybridTopology",
           "get_data_world_rank", "get_data_world_size"]
