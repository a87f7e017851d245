from mpgcn_amd.train import metrics
from mpgcn_amd.train.trainer import ModelTrainer

__all__ = ["ModelTrainer", "metrics"]
