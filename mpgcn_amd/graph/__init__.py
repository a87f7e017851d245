from mpgcn_amd.graph.supports import (
    get_support_K,
    build_supports,
    random_walk_normalize,
    symmetric_normalize,
    chebyshev_polynomials,
)
from mpgcn_amd.graph.dynamic import construct_dynamic_graphs

__all__ = [
    "get_support_K",
    "build_supports",
    "random_walk_normalize",
    "symmetric_normalize",
    "chebyshev_polynomials",
    "construct_dynamic_graphs",
]
