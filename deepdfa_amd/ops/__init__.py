from ._ext import has_ext, load_ext
from .flowgnn import attn_pool, embed4, gru_cell, segment_max, spmm_sum

__all__ = [
    "has_ext",
    "load_ext",
    "embed4",
    "spmm_sum",
    "gru_cell",
    "attn_pool",
    "segment_max",
]
