from .comm import CommContext, destroy, get_comm, init_distributed
from .pipeline import PartitionPlan, PipelineEngine

__all__ = [
    "CommContext", "init_distributed", "get_comm", "destroy",
    "PartitionPlan", "PipelineEngine",
]
