"""MI355X-native distributed training scaffold.

A from-scratch rebuild of the FRL-Distributed-ML-Scaffold capability
surface designed for MI355X: PyTorch-ROCm autograd +
hand-written CDNA4 HIP kernels (csrc/) + RCCL over xGMI.
"""

from .config import TrainConfig, load_config, config_hash
from .runtime import init_distributed, DistContext
from .trainer import Trainer

__version__ = "0.1.0"
__all__ = ["TrainConfig", "load_config", "config_hash", "init_distributed", "DistContext", "Trainer"]
