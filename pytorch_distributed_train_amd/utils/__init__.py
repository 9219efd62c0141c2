from .checkpoint import save_checkpoint, load_checkpoint
from .metrics import RankLogger, StepTimer
from .amp import GradScaler

__all__ = ["save_checkpoint", "load_checkpoint", "RankLogger", "StepTimer",
           "GradScaler"]
