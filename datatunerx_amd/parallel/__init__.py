from .ddp import (GradSynchronizer, init_distributed, is_main,
                  sync_scalar_mean)
