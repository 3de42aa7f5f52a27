from .trainer import Trainer
from .summary import SummaryWriter, NullWriter

__all__ = ["Trainer", "SummaryWriter", "NullWriter"]
