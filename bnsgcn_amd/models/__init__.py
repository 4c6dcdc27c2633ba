from .context import GraphContext
from .layers import GCNLayer, SAGELayer, GATLayer
from .models import GNNBase, GCN, GraphSAGE, GAT, create_model
from .sync_bn import SyncBatchNorm

__all__ = ["GraphContext", "GCNLayer", "SAGELayer", "GATLayer",
           "GNNBase", "GCN", "GraphSAGE", "GAT", "create_model",
           "SyncBatchNorm"]
