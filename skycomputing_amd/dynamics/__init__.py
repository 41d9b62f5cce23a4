from .allocator import AllocationError, Allocator, bert_boundary_payloads
from .benchmarker import DeviceBenchmarker, ModelBenchmarker, default_bert_probe_cfg
from .estimator import Estimator
from .parameter_server import ParameterServer
from .worker import Worker, WorkerManager

__all__ = [
    "Allocator", "AllocationError", "bert_boundary_payloads",
    "DeviceBenchmarker", "ModelBenchmarker", "default_bert_probe_cfg",
    "Estimator", "ParameterServer", "Worker", "WorkerManager",
]
