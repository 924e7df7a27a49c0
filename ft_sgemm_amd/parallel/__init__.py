from .distributed import (block_row_sgemm, init_from_env, local_shard,
                          replicated_weak_scaling_step)

__all__ = [
    "block_row_sgemm", "init_from_env", "local_shard",
    "replicated_weak_scaling_step",
]
