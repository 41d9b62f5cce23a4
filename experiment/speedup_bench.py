#!/usr/bin/env python3
"""Headline experiment: optimal/dynamic vs even allocation under injected
heterogeneity (the reference's 55%-faster claim, README.md:5 / arXiv
2202.11836, measured there on a 64-node cluster).

Runs the SAME pipeline workload twice (even, then optimal/dynamic) with
per-rank compute slowdowns injected (GPU busy-spin or CPU sleep scaled by
measured stage time) and reports wall-clock sec/iter for each and the
speedup. SPMD: launch under torch.distributed.run with one rank per GPU:

    torchrun --standalone --nproc-per-node 8 experiment/speedup_bench.py \
        --layers 160 --slowdowns 0,1,0.3,2,0.1,1.5,0.6,0.9

On CPU (gloo) the same script demonstrates the capability at small scale
(tests/test_speedup_e2e.py runs it with 3 ranks).
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--layers", type=int, default=24)
    p.add_argument("--batch", type=int, default=32)
    p.add_argument("--seq", type=int, default=128)
    p.add_argument("--hidden", type=int, default=1024)
    p.add_argument("--heads", type=int, default=16)
    p.add_argument("--steps", type=int, default=8)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--microbatches", type=int, default=0)
    p.add_argument("--schedule", default="gpipe")
    p.add_argument("--slowdowns", default="", help="per-rank factors, csv")
    p.add_argument("--stimulate", action="store_true")
    p.add_argument("--modes", default="even,optimal")
    p.add_argument("--json-out", default="")
    return p.parse_args()


def main():
    args = parse_args()
    from skycomputing_amd.dataset import SyntheticGlueDataset
    from skycomputing_amd.dynamics import (
        Allocator, DeviceBenchmarker, ModelBenchmarker, WorkerManager,
    )
    from skycomputing_amd.models import bert_pipeline_config
    from skycomputing_amd.optim import FusedSGD
    from skycomputing_amd.parallel import PartitionPlan, PipelineEngine, init_distributed
    from skycomputing_amd.stimulator import Stimulator

    comm = init_distributed()
    world, rank = comm.world_size, comm.rank
    use_cuda = torch.cuda.is_available()
    dtype = torch.bfloat16 if use_cuda else torch.float32
    torch.manual_seed(99 + rank)

    slowdowns = [float(s) for s in args.slowdowns.split(",") if s]
    slowdowns = (slowdowns * world)[:world] if slowdowns else [0.0] * world
    M = args.microbatches or (1 if world == 1 else min(8, args.batch))

    bc = dict(hidden_size=args.hidden, num_attention_heads=args.heads,
              intermediate_size=4 * args.hidden, vocab_size=30522,
              max_position_embeddings=512,
              hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0)
    layer_cfgs = bert_pipeline_config(args.layers, bc)

    wm = WorkerManager.from_world(world, [dict(slowdown=s) for s in slowdowns])
    stim = Stimulator(world) if args.stimulate else None
    db = DeviceBenchmarker(comm, batch_size=args.batch // M, seq_len=args.seq,
                           hidden=args.hidden, iterations=3, stimulator=stim)
    dev_results = db.benchmark(wm)
    mb = ModelBenchmarker(layer_cfgs, batch_size=args.batch // M, seq_len=args.seq)

    ds = SyntheticGlueDataset(size=args.batch * 2, max_seq_length=args.seq, seed=3)
    loader = torch.utils.data.DataLoader(ds, batch_size=args.batch, drop_last=True)
    batches = list(loader)

    results = {}
    for mode in args.modes.split(","):
        if rank == 0:
            workers = [dict(rank=r, **dev_results[r]) for r in range(world)]
            mres = mb.benchmark()
            alloc = Allocator(mres["flops"], mres["mem"], workers)
            plan = alloc.allocate(mode)
            plan_d = plan.to_dict()
        else:
            plan_d = None
        plan = PartitionPlan.from_dict(comm.broadcast_object(plan_d, src=0))
        engine = PipelineEngine(
            comm, layer_cfgs, plan,
            loss_fn=lambda lg, lb: torch.nn.functional.cross_entropy(lg.float(), lb),
            dtype=dtype,
            stage_kwargs=dict(record_forward_time=True, slowdown=slowdowns[rank]),
        )
        opt = FusedSGD(engine.parameters(), lr=1e-3)

        def step(i):
            data, labels = batches[i % len(batches)]
            opt.zero_grad(set_to_none=True)
            engine.run_iteration(data, labels, num_microbatches=M, schedule=args.schedule)
            opt.step()

        for i in range(args.warmup):
            step(i)
        comm.barrier()
        if use_cuda:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for i in range(args.steps):
            step(i)
        comm.barrier()
        if use_cuda:
            torch.cuda.synchronize()
        el = max(comm.all_gather_object(time.perf_counter() - t0)) / args.steps
        results[mode] = el
        if rank == 0:
            print(f"[speedup_bench] {mode:8s}: {el*1e3:9.1f} ms/iter  plan="
                  + ",".join(f"r{r}:{b-a}" for r, (a, b) in zip(plan.stage_ranks, plan.ranges)),
                  flush=True)
        del engine, opt
        if use_cuda:
            torch.cuda.empty_cache()

    if rank == 0:
        out = {"results_ms": {k: v * 1e3 for k, v in results.items()},
               "slowdowns": slowdowns, "world": world, "layers": args.layers,
               "microbatches": M, "schedule": args.schedule}
        if "even" in results:
            for m, v in results.items():
                if m != "even":
                    out[f"speedup_{m}_vs_even"] = results["even"] / v
        line = json.dumps(out)
        print(line, flush=True)
        if args.json_out:
            with open(args.json_out, "w") as f:
                f.write(line + "\n")


if __name__ == "__main__":
    main()
