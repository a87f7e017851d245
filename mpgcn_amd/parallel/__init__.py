from mpgcn_amd.parallel.ddp import DistContext, GradAllReducer, init_distributed

__all__ = ["DistContext", "GradAllReducer", "init_distributed"]
