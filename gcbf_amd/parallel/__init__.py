from .ddp import (GradSynchronizer, all_reduce_scalar, broadcast_modules,
                  cleanup_distributed, env_world, init_distributed)

__all__ = ["GradSynchronizer", "all_reduce_scalar", "broadcast_modules",
           "cleanup_distributed", "env_world", "init_distributed"]
