"""Distributed training (reference include/distributed/, include/partitioner/).

MI355X-native re-design: the reference's TCP/RoCE coordinator/worker
processes become **one torch.distributed rank per GPU** (backend "nccl" ==
RCCL over xGMI on ROCm; "gloo" for CPU tests). Pipeline stages map to
ranks; activations/grads move rank→rank as device-buffer P2P send/recv
(neighbor traffic maps 1:1 onto xGMI point-to-point links); the
Message/CommandType control plane becomes a tiny object broadcast channel.
"""

from .comm import (init_distributed, is_initialized, rank, world_size,
                   Communicator)
from .partitioner import (NaivePipelinePartitioner, WeightedPipelinePartitioner,
                          NaiveDataPartitioner, partition_model)
from .pipeline import PipelineEngine
from .train import (train_pipeline_epoch, validate_pipeline_epoch,
                    train_pipeline_model, save_stage_checkpoint,
                    load_stage_checkpoint)
from .ddp import DataParallelEngine

__all__ = ["init_distributed", "is_initialized", "rank", "world_size",
           "Communicator", "NaivePipelinePartitioner",
           "WeightedPipelinePartitioner", "NaiveDataPartitioner",
           "partition_model", "PipelineEngine", "DataParallelEngine",
           "train_pipeline_epoch", "validate_pipeline_epoch",
           "train_pipeline_model", "save_stage_checkpoint",
           "load_stage_checkpoint"]
