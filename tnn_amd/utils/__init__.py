"""Cross-cutting utilities (reference include/common/, utils/, logging/,
profiling/)."""

from .logging import get_logger
from .config import TConfig
from .checkpoint import save_model, load_model, save_checkpoint, load_checkpoint
from .profiler import Profiler, GlobalProfiler, EventType
from .env import EnvLoader, env_get
from .graphstep import GraphedInference
from . import hwinfo

__all__ = ["get_logger", "TConfig", "save_model", "load_model",
           "save_checkpoint", "load_checkpoint",
           "Profiler", "GlobalProfiler", "EventType", "EnvLoader", "env_get",
           "GraphedInference",
           "hwinfo"]
