from .comm import CommContext, destroy, get_comm, init_distributed
from .interleaved import InterleavedPipelineEngine, build_interleaved_plan
from .interleaved_graph import GraphedInterleavedStep
from .pipeline import PartitionPlan, PipelineEngine
from .static_exec import GraphedPipelineStep

__all__ = [
    "CommContext", "init_distributed", "get_comm", "destroy",
    "PartitionPlan", "PipelineEngine", "GraphedPipelineStep",
    "InterleavedPipelineEngine", "build_interleaved_plan",
    "GraphedInterleavedStep",
]
