from .absdf import SUBKEYS, build_vocab, get_dataflow_features, node_feature_indices, to_hash
from .cpg import parse_joern_json, synthetic_cpg
from .dbize import cpg_to_tables, dbize

__all__ = [
    "SUBKEYS",
    "build_vocab",
    "get_dataflow_features",
    "node_feature_indices",
    "to_hash",
    "parse_joern_json",
    "synthetic_cpg",
    "cpg_to_tables",
    "dbize",
]
