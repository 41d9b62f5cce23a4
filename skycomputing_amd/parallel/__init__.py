from .comm import CommContext, destroy, get_comm, init_distributed
from .pipeline import PartitionPlan, PipelineEngine
from .static_exec import GraphedPipelineStep

__all__ = [
    "CommContext", "init_distributed", "get_comm", "destroy",
    "PartitionPlan", "PipelineEngine", "GraphedPipelineStep",
]
