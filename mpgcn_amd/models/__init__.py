from mpgcn_amd.models.bdgcn import BDGCN
from mpgcn_amd.models.gcn1d import GCN
from mpgcn_amd.models.mpgcn import MPGCN, TemporalEncoder

__all__ = ["BDGCN", "GCN", "MPGCN", "TemporalEncoder"]
