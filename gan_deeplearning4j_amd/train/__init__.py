from .gan_trainer import GanTrainer, ReferenceProtocolTrainer
from .metrics import MetricsLogger

__all__ = ["GanTrainer", "ReferenceProtocolTrainer", "MetricsLogger"]
