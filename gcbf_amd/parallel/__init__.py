from .ddp import (BucketedGradSynchronizer, GradSynchronizer,
                  all_reduce_scalar, broadcast_modules, cleanup_distributed,
                  env_world, init_distributed, make_grad_synchronizer)

__all__ = ["BucketedGradSynchronizer", "GradSynchronizer",
           "all_reduce_scalar", "broadcast_modules", "cleanup_distributed",
           "env_world", "init_distributed", "make_grad_synchronizer"]
