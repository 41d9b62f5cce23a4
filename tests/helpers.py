"""Shared helpers for multi-process (gloo) tests."""

from __future__ import annotations

import os

import torch.multiprocessing as mp


def tiny_bert_cfg(num_encoder_layers: int = 2, dropout: float = 0.0) -> list[dict]:
    from skycomputing_amd.models import bert_pipeline_config

    return bert_pipeline_config(
        num_encoder_layers,
        dict(
            hidden_size=64,
            num_attention_heads=4,
            intermediate_size=128,
            vocab_size=500,
            max_position_embeddings=64,
            hidden_dropout_prob=dropout,
            attention_probs_dropout_prob=dropout,
        ),
    )


def run_multiprocess(fn, world_size: int, port: int, *args, timeout: float = 180.0,
                     retries: int = 2):
    """Spawn `world_size` ranks running fn(rank, world_size, *args); raise on
    failure. The rendezvous port is jittered by pid to avoid TIME_WAIT
    collisions; transient rendezvous/teardown failures are retried once on
    a fresh port (loaded CI machines occasionally drop a gloo connect)."""
    last = None
    for attempt in range(retries + 1):
        p_eff = 20000 + (port + os.getpid() * 7 + attempt * 131) % 40000
        try:
            _run_once(fn, world_size, p_eff, args, timeout)
            return
        except (RuntimeError, TimeoutError) as e:
            last = e
    raise last


def _run_once(fn, world_size, port, args, timeout):
    ctx = mp.get_context("spawn")
    procs = []
    for rank in range(world_size):
        env = {
            "RANK": str(rank),
            "WORLD_SIZE": str(world_size),
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
            "LOCAL_RANK": str(rank),
        }
        p = ctx.Process(target=_entry, args=(fn, rank, world_size, env, args))
        p.start()
        procs.append(p)
    for p in procs:
        p.join(timeout)
    try:
        for rank, p in enumerate(procs):
            if p.is_alive():
                raise TimeoutError(f"rank {rank} timed out")
            if p.exitcode != 0:
                raise RuntimeError(f"rank {rank} exited with {p.exitcode}")
    finally:
        for p in procs:
            if p.is_alive():
                p.terminate()


def _entry(fn, rank, world_size, env, args):
    os.environ.update(env)
    fn(rank, world_size, *args)
