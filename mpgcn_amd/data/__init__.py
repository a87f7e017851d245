from mpgcn_amd.data.container import DataInput, DataGenerator, ODBatchIterator
from mpgcn_amd.data.synthetic import synthetic_od, synthetic_adjacency

__all__ = [
    "DataInput",
    "DataGenerator",
    "ODBatchIterator",
    "synthetic_od",
    "synthetic_adjacency",
]
