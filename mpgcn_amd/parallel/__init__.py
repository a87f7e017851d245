from mpgcn_amd.parallel.ddp import (DistContext, GradAllReducer, init_distributed, rank_watchdog)

__all__ = ["DistContext", "GradAllReducer", "init_distributed", "rank_watchdog"]
