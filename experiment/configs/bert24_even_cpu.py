"""BASELINE config 1: BERT-large 24-layer, even allocation, 2-rank CPU
plumbing check (reference shipped this as its smoke configuration).

Run: python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 experiment/launch.py -c experiment/configs/bert24_even_cpu.py
"""

base = "../config.py"

model_config = dict(kind="bert", num_encoder_layers=24, num_class=3)
allocator_config = dict(
    mode="even",
    benchmark=dict(batch_size=32, seq_len=128, iterations=5),
    stimulate=False,
)
