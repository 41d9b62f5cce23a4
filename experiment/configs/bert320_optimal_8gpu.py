"""BASELINE config 5: 320-layer BERT-large, optimal allocation, 8xMI355X —
deep-stack stress of the 288 GB HBM per stage.

Run: python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
  --master-addr 127.0.0.1 experiment/launch.py -c experiment/configs/bert320_optimal_8gpu.py
"""

base = "../config.py"

model_config = dict(kind="bert", num_encoder_layers=320, num_class=3)
allocator_config = dict(
    mode="optimal",
    benchmark=dict(batch_size=32, seq_len=128, iterations=5),
    stimulate=False,
)
