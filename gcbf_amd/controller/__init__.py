from .base import MultiAgentController
from .gnn_controller import GNNController
from .macbf_controller import MACBFController
from .nominal import NominalController

__all__ = ["MultiAgentController", "GNNController", "MACBFController",
           "NominalController"]
