from .datamodule import BigVulDatasetLineVDDataModule
from .dataset import BigVulDatasetLineVD, collate_graphs
from .dclass import BigVulDataset, synthetic_bigvul_df
from .features import FeatureSpec, parse_limits

__all__ = [
    "BigVulDataset",
    "BigVulDatasetLineVD",
    "BigVulDatasetLineVDDataModule",
    "collate_graphs",
    "synthetic_bigvul_df",
    "FeatureSpec",
    "parse_limits",
]
