#!/usr/bin/env python3
"""Experiment driver CLI.

Capability parity with the reference launcher
(reference: experiment/launch.py:20-236), SPMD redesign: every rank runs
this same program under ``torch.distributed.run`` (or plain python for one
GPU) — there is no host/worker asymmetry, no RPC server loop, no HOST file
(rendezvous comes from MASTER_ADDR/PORT env, 127.0.0.1 on one node).

    torchrun --standalone --nproc-per-node 8 experiment/launch.py \
        -c experiment/config.py

Flow: load config -> init RCCL world -> self-benchmark every rank ->
allocate layer ranges (even/dynamic/optimal) -> build the stage ->
Runner.train with hooks.
"""

from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("-c", "--config", required=True)
    p.add_argument("--allocate", default=None, help="override allocator mode")
    p.add_argument("--max-iter", type=int, default=None)
    return p.parse_args()


def build_layer_cfgs(model_config: dict) -> list[dict]:
    kind = model_config.get("kind", "bert")
    if kind == "bert":
        from skycomputing_amd.models import bert_pipeline_config

        return bert_pipeline_config(
            model_config["num_encoder_layers"],
            dict(model_config.get("bert_config", {})),
            num_class=model_config.get("num_class", 3),
        )
    if kind == "resnet":
        from skycomputing_amd.models import resnet_pipeline_config

        return resnet_pipeline_config(
            model_config.get("depth", 18), model_config.get("num_class", 10)
        )
    if kind == "layers":
        return list(model_config["layers"])
    raise ValueError(f"unknown model kind {kind!r}")


def main():
    args = parse_args()
    from skycomputing_amd.builder import build_dataloader_from_cfg, build_hook
    from skycomputing_amd.config import load_config
    from skycomputing_amd.dynamics import (
        Allocator, DeviceBenchmarker, ModelBenchmarker, WorkerManager,
    )
    from skycomputing_amd.logger import Logger
    from skycomputing_amd.optim import FusedSGD
    from skycomputing_amd.parallel import PartitionPlan, PipelineEngine, init_distributed
    from skycomputing_amd.runner import Runner
    from skycomputing_amd.stimulator import Stimulator

    cfg = load_config(args.config)
    comm = init_distributed()
    rank, world = comm.rank, comm.world_size
    use_cuda = torch.cuda.is_available()
    dtype = torch.bfloat16 if use_cuda else torch.float32

    log_dir = cfg.get("logging_config", {}).get("log_dir", "./logs")
    logger = Logger(log_file=os.path.join(log_dir, f"rank{rank}.log"), rank=rank)

    layer_cfgs = build_layer_cfgs(cfg.model_config)
    L = len(layer_cfgs)
    logger.info(f"model: {L} pipeline layers; world={world}")

    # worker pool (cycle the config list over ranks)
    wcfgs = list(cfg.get("worker_config", [{}])) or [{}]
    extra = [dict(wcfgs[r % len(wcfgs)]) for r in range(world)]
    wm = WorkerManager.from_world(world, extra)

    acfg = cfg.get("allocator_config", {"mode": "even"})
    mode = args.allocate or acfg.get("mode", "even")
    virtual = int(acfg.get("virtual_stages", 1)) if world > 1 else 1
    if mode == "even" and not acfg.get("stimulate"):
        if virtual > 1:
            from skycomputing_amd.parallel.interleaved import build_interleaved_plan

            plan = build_interleaved_plan(L, world, virtual)
        else:
            base, rem = divmod(L, world)
            bounds = [0]
            for i in range(world):
                bounds.append(bounds[-1] + base + (1 if i < rem else 0))
            plan = PartitionPlan(
                stage_ranks=list(range(world)),
                ranges=[(bounds[i], bounds[i + 1]) for i in range(world)],
            )
    else:
        bench = acfg.get("benchmark", {})
        stim = Stimulator(world) if acfg.get("stimulate") else None
        db = DeviceBenchmarker(
            comm,
            batch_size=bench.get("batch_size", 32),
            seq_len=bench.get("seq_len", 128),
            hidden=bench.get("hidden", 1024),
            iterations=bench.get("iterations", 5),
            stimulator=stim,
        )
        dev_results = db.benchmark(wm)
        if rank == 0:
            mb = ModelBenchmarker(
                layer_cfgs,
                batch_size=bench.get("batch_size", 32),
                seq_len=bench.get("seq_len", 128),
            )
            model_results = mb.benchmark()
            workers = [dict(rank=r, **dev_results[r]) for r in range(world)]
            alloc = Allocator(model_results["flops"], model_results["mem"], workers)
            if virtual > 1:
                plan = alloc.interleaved_allocate(virtual)
            else:
                plan = alloc.allocate(mode)
            logger.info(
                f"allocation ({mode}): "
                + ", ".join(f"r{r}:[{a},{b})" for r, (a, b) in zip(plan.stage_ranks, plan.ranges))
            )
            plan_d = plan.to_dict()
        else:
            plan_d = None
        plan = PartitionPlan.from_dict(comm.broadcast_object(plan_d, src=0))
        if virtual == 1:
            for r, rng in zip(plan.stage_ranks, plan.ranges):
                wm.assign_model_to_worker(r, rng)

    sd = float(extra[rank].get("slowdown") or 0.0)
    if virtual > 1:
        from skycomputing_amd.parallel.interleaved import InterleavedPipelineEngine

        engine = InterleavedPipelineEngine(
            comm, layer_cfgs, plan,
            loss_fn=lambda logits, labels: torch.nn.functional.cross_entropy(
                logits.float(), labels
            ),
            dtype=dtype,
            stage_kwargs=dict(record_forward_time=True, slowdown=sd,
                              mem_limit=extra[rank].get("mem_limit")),
        )
    else:
        engine = PipelineEngine(
            comm, layer_cfgs, plan,
            loss_fn=lambda logits, labels: torch.nn.functional.cross_entropy(
                logits.float(), labels
            ),
            dtype=dtype,
            stage_kwargs=dict(record_forward_time=True, slowdown=sd,
                              mem_limit=extra[rank].get("mem_limit")),
        )

    tcfg = cfg.train_config
    opt_cfg = tcfg.get("optimizer", {})
    opt = FusedSGD(
        engine.parameters(),
        lr=opt_cfg.get("lr", 1e-3),
        momentum=opt_cfg.get("momentum", 0.0),
        weight_decay=opt_cfg.get("weight_decay", 0.0),
    )

    M = tcfg.get("num_microbatches", 0) or (1 if world == 1 else min(8, cfg.data_config["batch_size"]))
    runner = Runner(
        engine, opt, comm,
        max_epoch=tcfg.get("max_epoch", 1),
        max_iter=args.max_iter if args.max_iter is not None else tcfg.get("max_iter"),
        num_microbatches=M,
        schedule=tcfg.get("schedule", "gpipe"),
        logger=logger,
        log_interval=tcfg.get("log_interval", 1),
    )
    for hook_cfg in tcfg.get("hooks", []):
        runner.register_hook(build_hook(dict(hook_cfg)))

    loader = build_dataloader_from_cfg(
        cfg.data_config["batch_size"], dict(cfg.data_config["dataset"])
    )
    runner.train(loader)
    logger.info(f"done: {runner.iter} iterations, last loss {runner.last_loss}")
    comm.barrier()
    from skycomputing_amd.parallel import destroy

    destroy()


if __name__ == "__main__":
    main()
