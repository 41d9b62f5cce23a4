"""BASELINE config 3 (paper headline): 160-layer BERT (H=1024, A=16),
dynamic allocation, 8xMI355X.

Run: python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
  --master-addr 127.0.0.1 experiment/launch.py -c experiment/configs/bert160_dynamic_8gpu.py
"""

base = "../config.py"

model_config = dict(kind="bert", num_encoder_layers=160, num_class=3)
allocator_config = dict(
    mode="dynamic",
    benchmark=dict(batch_size=32, seq_len=128, iterations=5),
    stimulate=False,
)
