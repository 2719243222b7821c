from .ddp import (all_reduce_grads, barrier, get_rank, get_world_size,  # noqa: F401
                  init_distributed, is_distributed, oversubscribed)
