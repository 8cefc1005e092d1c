from .base_module import BaseModule
from .flow_gnn import FlowGNNGGNNModule, GatedGraphConv, GlobalAttentionPooling

__all__ = ["BaseModule", "FlowGNNGGNNModule", "GatedGraphConv", "GlobalAttentionPooling"]
