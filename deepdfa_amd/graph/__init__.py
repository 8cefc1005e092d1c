from .batch import BatchedCFG, batch_graphs
from .synthetic import synthetic_cfg, synthetic_cfg_batch

__all__ = ["BatchedCFG", "batch_graphs", "synthetic_cfg", "synthetic_cfg_batch"]
