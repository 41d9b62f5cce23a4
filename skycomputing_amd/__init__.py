"""skycomputing_amd — MI355X-native load-balanced pipeline-parallel training.

A from-scratch rebuild of the capabilities of hpcaitech/SkyComputing
(scaelum) designed for AMD Instinct MI355X (gfx950, CDNA4):

  * process-per-GPU SPMD over RCCL/xGMI instead of host-orchestrated
    TensorPipe RPC;
  * hand-written HIP/CDNA4 kernels for the BERT hot path (fully-fused
    MFMA attention fwd+bwd, fused LayerNorm(+residual+dropout), bias-GELU,
    masked softmax, fused embedding, multi-tensor SGD, MFMA GEMM with
    fused epilogues) instead of eager CUDA ops + optional apex;
  * benchmark-driven even/dynamic/optimal layer allocation with an EXACT
    subset-DP solver replacing the reference's 300s/20%-gap MIP;
  * device-resident activations, bf16 compute, 288 GB HBM3E sizing.

See SURVEY.md for the reference analysis this build follows.
"""

__version__ = "0.2.0"

from . import builder, config, dataset, dynamics, models, ops, parallel, registry, runner
from .logger import Logger
from .optim import FusedSGD
from .stimulator import Stimulator
from .timer import DeviceTimer, DistributedTimer

__all__ = [
    "builder", "config", "dataset", "dynamics", "models", "ops", "parallel",
    "registry", "runner", "Logger", "FusedSGD", "Stimulator", "DeviceTimer",
    "DistributedTimer", "__version__",
]
