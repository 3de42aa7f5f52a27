from .mlp import MLP, init_param
from .gnn import (AttentionalAggregation, CBFGNNLayer, ControllerGNNLayer,
                  CBFNetLayer, MACBFControllerLayer)

__all__ = ["MLP", "init_param", "AttentionalAggregation", "CBFGNNLayer",
           "ControllerGNNLayer", "CBFNetLayer", "MACBFControllerLayer"]
