"""BASELINE config 4: 96-layer BERT with seeded synthetic per-GPU
heterogeneity (Stimulator) to exercise the load balancer on a homogeneous
8xMI355X node — the optimal-vs-even headline capability.

Run (compare ALLOCATE_TYPE=even vs this config, or use
experiment/speedup_bench.py for the wall-clock A/B):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
    --master-addr 127.0.0.1 experiment/launch.py -c experiment/configs/bert96_stimulate_8gpu.py
"""

base = "../config.py"

model_config = dict(kind="bert", num_encoder_layers=96, num_class=3)
allocator_config = dict(
    mode="optimal",
    benchmark=dict(batch_size=32, seq_len=128, iterations=5),
    stimulate=True,
)
