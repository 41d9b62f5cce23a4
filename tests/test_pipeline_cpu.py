"""Multi-process (gloo, CPU) pipeline integration tests.

These exercise the full SPMD path — comm substrate, partition plan, stage
build, forward/backward over P2P, local optimizer step — the way the driver
and the 8xMI355X runs do, just on CPU with the gloo backend.
"""

from __future__ import annotations

import numpy as np
import pytest
import torch

from .helpers import run_multiprocess, tiny_bert_cfg


def _make_batch(bsz=8, seq=16, vocab=500, seed=0):
    g = torch.Generator().manual_seed(seed)
    ids = torch.randint(0, vocab, (bsz, seq), generator=g)
    mask = torch.ones(bsz, seq, dtype=torch.long)
    tids = torch.zeros(bsz, seq, dtype=torch.long)
    labels = torch.randint(0, 3, (bsz,), generator=g)
    return (ids, tids, mask), labels


def _single_process_reference(layer_cfgs, batch, labels, lr, steps, num_microbatches=1):
    """Ground truth: same model, same init seed, trained locally."""
    torch.manual_seed(1234)
    from skycomputing_amd.builder import build_module_from_cfg

    stage = build_module_from_cfg(layer_cfgs, record_forward_time=False)
    opt = torch.optim.SGD(stage.parameters(), lr=lr)
    loss_fn = torch.nn.CrossEntropyLoss()
    losses = []
    for _ in range(steps):
        opt.zero_grad()
        if num_microbatches == 1:
            out = stage(*batch)
            loss = loss_fn(out, labels)
            loss.backward()
            losses.append(float(loss.detach()))
        else:
            M = num_microbatches
            tot = 0.0
            mb_in = [torch.chunk(b, M) for b in batch]
            mb_lab = torch.chunk(labels, M)
            for m in range(M):
                out = stage(*(x[m] for x in mb_in))
                loss = loss_fn(out, mb_lab[m])
                (loss / M).backward()
                tot += float(loss.detach())
            losses.append(tot / M)
        opt.step()
    return losses


def _pipeline_worker(rank, world_size, layer_cfgs, batch, labels, lr, steps, M, schedule, out_dir):
    torch.manual_seed(1234)  # same init on every rank; slices differ per stage
    from skycomputing_amd.optim import FusedSGD
    from skycomputing_amd.parallel import PartitionPlan, PipelineEngine, init_distributed

    comm = init_distributed(backend="gloo", timeout_s=60)
    # NOTE: to match the single-process init exactly, each rank builds the
    # FULL model under the same seed, then keeps only its slice.
    from skycomputing_amd.builder import build_module_from_cfg

    full = build_module_from_cfg(layer_cfgs, record_forward_time=False)
    L = len(layer_cfgs)
    cut = L // 2
    plan = PartitionPlan(stage_ranks=[0, 1], ranges=[(0, cut), (cut, L)])
    engine = PipelineEngine(comm, layer_cfgs, plan, loss_fn=torch.nn.CrossEntropyLoss(),
                            stage_kwargs=dict(record_forward_time=False))
    # overwrite stage weights with the reference slice init
    start, end = plan.ranges[engine.stage_idx]
    engine.stage.load_layer_state_dicts(
        [
            {k: v.detach().clone() for k, v in full.module[i].state_dict().items()}
            for i in range(start, end)
        ]
    )
    opt = FusedSGD(engine.parameters(), lr=lr)
    losses = []
    for _ in range(steps):
        opt.zero_grad()
        loss = engine.run_iteration(batch, labels, num_microbatches=M, schedule=schedule)
        opt.step()
        losses.append(loss)
    if rank == 0:
        np.save(f"{out_dir}/losses.npy", np.array(losses, dtype=np.float64))
    comm.barrier()
    from skycomputing_amd.parallel import destroy

    destroy()


@pytest.mark.parametrize("schedule,M", [("sequential", 1), ("gpipe", 4)])
def test_two_stage_pipeline_matches_local(tmp_path, schedule, M):
    layer_cfgs = tiny_bert_cfg(2)
    batch, labels = _make_batch()
    lr, steps = 0.05, 3
    ref = _single_process_reference(layer_cfgs, batch, labels, lr, steps, M)
    port = 29600 + (5 if schedule == "gpipe" else 0)
    run_multiprocess(
        _pipeline_worker, 2, port, layer_cfgs, batch, labels, lr, steps, M, schedule, str(tmp_path)
    )
    got = np.load(f"{tmp_path}/losses.npy")
    assert np.allclose(got, np.array(ref), rtol=1e-4, atol=1e-5), (got, ref)
    # loss should decrease over steps
    assert got[-1] < got[0]


def _overlap_worker(rank, world_size, layer_cfgs, batch, labels, lr, steps,
                    M, overlap, out_dir):
    import os as _os

    if not overlap:
        _os.environ["SKY_NO_OVERLAP"] = "1"
    torch.manual_seed(1234)
    from skycomputing_amd.builder import build_module_from_cfg
    from skycomputing_amd.optim import FusedSGD
    from skycomputing_amd.parallel import PartitionPlan, PipelineEngine, destroy, init_distributed

    comm = init_distributed(backend="gloo", timeout_s=60)
    full = build_module_from_cfg(layer_cfgs, record_forward_time=False)
    L = len(layer_cfgs)
    plan = PartitionPlan(stage_ranks=[0, 1, 2],
                         ranges=[(0, L // 3), (L // 3, 2 * L // 3), (2 * L // 3, L)])
    engine = PipelineEngine(comm, layer_cfgs, plan,
                            loss_fn=torch.nn.CrossEntropyLoss(),
                            stage_kwargs=dict(record_forward_time=False))
    start, end = plan.ranges[engine.stage_idx]
    engine.stage.load_layer_state_dicts(
        [{k: v.detach().clone() for k, v in full.module[i].state_dict().items()}
         for i in range(start, end)])
    opt = FusedSGD(engine.parameters(), lr=lr)
    losses = []
    for _ in range(steps):
        opt.zero_grad()
        losses.append(engine.run_iteration(batch, labels, num_microbatches=M))
        opt.step()
    if rank == 0:
        np.save(f"{out_dir}/losses_{'ov' if overlap else 'bl'}.npy",
                np.array(losses, dtype=np.float64))
    comm.barrier()
    destroy()


def test_gpipe_overlap_matches_blocking(tmp_path):
    """The overlapped GPipe transport (pre-posted irecvs + isends) must be
    loss-identical to the blocking hops on a 3-stage pipeline (middle rank
    exercises both directions)."""
    layer_cfgs = tiny_bert_cfg(3)
    batch, labels = _make_batch()
    for overlap in (True, False):
        run_multiprocess(_overlap_worker, 3, 29660 + int(overlap), layer_cfgs,
                         batch, labels, 0.05, 3, 4, overlap, str(tmp_path))
    ov = np.load(f"{tmp_path}/losses_ov.npy")
    bl = np.load(f"{tmp_path}/losses_bl.npy")
    assert np.allclose(ov, bl, rtol=1e-6, atol=1e-7), (ov, bl)


def _uneven_plan_worker(rank, world_size, layer_cfgs, batch, labels, out_dir):
    torch.manual_seed(7)
    from skycomputing_amd.parallel import PartitionPlan, PipelineEngine, init_distributed

    comm = init_distributed(backend="gloo", timeout_s=60)
    L = len(layer_cfgs)
    # stage order deliberately NOT rank order; rank 2 idle
    plan = PartitionPlan(stage_ranks=[1, 0], ranges=[(0, 2), (2, L)])
    engine = PipelineEngine(
        comm, layer_cfgs, plan, loss_fn=torch.nn.CrossEntropyLoss(),
        stage_kwargs=dict(record_forward_time=False),
    )
    loss = engine.run_iteration(batch, labels, num_microbatches=1, schedule="sequential")
    assert loss is not None and np.isfinite(loss)
    comm.barrier()
    from skycomputing_amd.parallel import destroy

    destroy()


def test_reordered_stages_and_idle_rank():
    layer_cfgs = tiny_bert_cfg(1)
    batch, labels = _make_batch()
    run_multiprocess(_uneven_plan_worker, 3, 29640, layer_cfgs, batch, labels, ".")


def _1f1b_worker(rank, world_size, layer_cfgs, batch, labels, lr, steps, M, out_dir):
    _pipeline_worker(rank, world_size, layer_cfgs, batch, labels, lr, steps, M, "1f1b", out_dir)


def test_1f1b_matches_local_three_stages(tmp_path):
    """1F1B schedule (fused crossing P2P) must produce the same losses as
    single-process training."""
    layer_cfgs = tiny_bert_cfg(3)
    batch, labels = _make_batch()
    lr, steps = 0.05, 3
    ref = _single_process_reference(layer_cfgs, batch, labels, lr, steps, 4)
    run_multiprocess(
        _1f1b_3s_entry, 3, 29860, layer_cfgs, batch, labels, lr, steps, 4, str(tmp_path)
    )
    got = np.load(f"{tmp_path}/losses.npy")
    assert np.allclose(got, np.array(ref), rtol=1e-4, atol=1e-5), (got, ref)


def _1f1b_3s_entry(rank, world_size, layer_cfgs, batch, labels, lr, steps, M, out_dir):
    torch.manual_seed(1234)
    from skycomputing_amd.optim import FusedSGD
    from skycomputing_amd.parallel import PartitionPlan, PipelineEngine, destroy, init_distributed
    from skycomputing_amd.builder import build_module_from_cfg

    comm = init_distributed(backend="gloo", timeout_s=60)
    full = build_module_from_cfg(layer_cfgs, record_forward_time=False)
    L = len(layer_cfgs)
    cuts = [0, L // 3, 2 * L // 3, L]
    plan = PartitionPlan(stage_ranks=[0, 1, 2],
                         ranges=[(cuts[i], cuts[i + 1]) for i in range(3)])
    engine = PipelineEngine(comm, layer_cfgs, plan, loss_fn=torch.nn.CrossEntropyLoss(),
                            stage_kwargs=dict(record_forward_time=False))
    start, end = plan.ranges[engine.stage_idx]
    engine.stage.load_layer_state_dicts(
        [{k: v.detach().clone() for k, v in full.module[i].state_dict().items()}
         for i in range(start, end)]
    )
    opt = FusedSGD(engine.parameters(), lr=lr)
    losses = []
    for _ in range(steps):
        opt.zero_grad()
        loss = engine.run_iteration(batch, labels, num_microbatches=M, schedule="1f1b")
        opt.step()
        losses.append(loss)
    if rank == 0:
        np.save(f"{out_dir}/losses.npy", np.array(losses, dtype=np.float64))
    comm.barrier()
    destroy()


def _val_worker(rank, world_size, layer_cfgs, out_dir):
    torch.manual_seed(21)
    from skycomputing_amd.dataset import SyntheticGlueDataset
    from skycomputing_amd.optim import FusedSGD
    from skycomputing_amd.parallel import PartitionPlan, PipelineEngine, destroy, init_distributed
    from skycomputing_amd.runner import Runner

    comm = init_distributed(backend="gloo", timeout_s=60)
    L = len(layer_cfgs)
    plan = PartitionPlan(stage_ranks=[0, 1], ranges=[(0, L // 2), (L // 2, L)])
    engine = PipelineEngine(comm, layer_cfgs, plan,
                            loss_fn=torch.nn.CrossEntropyLoss(),
                            stage_kwargs=dict(record_forward_time=False))
    opt = FusedSGD(engine.parameters(), lr=0.01)
    runner = Runner(engine, opt, comm, max_epoch=1, max_iter=2, log_interval=10)
    ds = SyntheticGlueDataset(size=32, max_seq_length=16, vocab_size=500, seed=9)
    loader = torch.utils.data.DataLoader(ds, batch_size=8, drop_last=True)
    runner.train(loader)
    acc = runner.val(loader, max_batches=2)
    assert acc is not None and 0.0 <= acc <= 1.0
    comm.barrier()
    destroy()


def test_runner_val_pipeline():
    run_multiprocess(_val_worker, 2, 29950, tiny_bert_cfg(1), ".")


def _four_stage_worker(rank, world_size, layer_cfgs, batch, labels, lr, steps, M, out_dir):
    torch.manual_seed(1234)
    from skycomputing_amd.builder import build_module_from_cfg
    from skycomputing_amd.optim import FusedSGD
    from skycomputing_amd.parallel import PartitionPlan, PipelineEngine, init_distributed

    comm = init_distributed(backend="gloo", timeout_s=60)
    full = build_module_from_cfg(layer_cfgs, record_forward_time=False)
    L = len(layer_cfgs)
    base, rem = divmod(L, world_size)
    bounds = [0]
    for i in range(world_size):
        bounds.append(bounds[-1] + base + (1 if i < rem else 0))
    plan = PartitionPlan(
        stage_ranks=list(range(world_size)),
        ranges=[(bounds[i], bounds[i + 1]) for i in range(world_size)],
    )
    engine = PipelineEngine(comm, layer_cfgs, plan, loss_fn=torch.nn.CrossEntropyLoss(),
                            stage_kwargs=dict(record_forward_time=False))
    start, end = plan.ranges[engine.stage_idx]
    engine.stage.load_layer_state_dicts(
        [
            {k: v.detach().clone() for k, v in full.module[i].state_dict().items()}
            for i in range(start, end)
        ]
    )
    opt = FusedSGD(engine.parameters(), lr=lr)
    losses = []
    for _ in range(steps):
        opt.zero_grad()
        loss = engine.run_iteration(batch, labels, num_microbatches=M, schedule="gpipe")
        opt.step()
        losses.append(loss)
    if rank == 0:
        np.save(f"{out_dir}/losses4.npy", np.array(losses, dtype=np.float64))
    comm.barrier()
    from skycomputing_amd.parallel import destroy

    destroy()


def test_four_stage_gpipe_m8_matches_local(tmp_path):
    """The driver's SCALE shape: even split over 4 stages, gpipe, M=8 —
    per-step losses must match the single-process reference exactly."""
    layer_cfgs = tiny_bert_cfg(2)
    batch, labels = _make_batch()
    lr, steps, M = 0.05, 3, 8
    ref = _single_process_reference(layer_cfgs, batch, labels, lr, steps, M)
    run_multiprocess(
        _four_stage_worker, 4, 29720, layer_cfgs, batch, labels, lr, steps, M, str(tmp_path)
    )
    got = np.load(f"{tmp_path}/losses4.npy")
    assert np.allclose(got, np.array(ref), rtol=1e-4, atol=1e-5), (got, ref)
    assert got[-1] < got[0]


def _isend_worker(rank, world_size, out_dir):
    import torch as T

    from skycomputing_amd.parallel import init_distributed

    comm = init_distributed(backend="gloo", timeout_s=60)
    peer = 1 - rank
    t1 = T.arange(8, dtype=T.float32)
    t2 = T.full((3, 2), float(rank), dtype=T.float32)
    # handshake (blocking) then non-blocking steady state
    if rank == 0:
        comm.isend_tensors([t1, t2], peer, "u", blocking=True)
        works = comm.isend_tensors([t1 * 2, t2 * 2], peer, "u")
        for w in works:
            w.wait()
    else:
        a = comm.recv_tensors(peer, "u")
        b = comm.recv_tensors(peer, "u")
        assert T.equal(a[0], T.arange(8, dtype=T.float32))
        assert T.equal(b[0], T.arange(8, dtype=T.float32) * 2)
        assert a[1].shape == (3, 2) and float(b[1][0, 0]) == 0.0
    comm.barrier()
    from skycomputing_amd.parallel import destroy

    destroy()


def test_isend_tensors_transport():
    """isend_tensors: blocking handshake first, then meta-less
    non-blocking sends with buffer-retaining work handles."""
    run_multiprocess(_isend_worker, 2, 29810, ".")


def _plan_worker(rank, world_size, layer_cfgs, batch, labels, lr, steps, M,
                 schedule, ranges, out_dir):
    torch.manual_seed(1234)
    from skycomputing_amd.builder import build_module_from_cfg
    from skycomputing_amd.optim import FusedSGD
    from skycomputing_amd.parallel import PartitionPlan, PipelineEngine, init_distributed

    comm = init_distributed(backend="gloo", timeout_s=60)
    full = build_module_from_cfg(layer_cfgs, record_forward_time=False)
    plan = PartitionPlan(stage_ranks=list(range(len(ranges))), ranges=list(ranges))
    engine = PipelineEngine(comm, layer_cfgs, plan, loss_fn=torch.nn.CrossEntropyLoss(),
                            stage_kwargs=dict(record_forward_time=False))
    if engine.stage_idx is not None:
        start, end = plan.ranges[engine.stage_idx]
        engine.stage.load_layer_state_dicts(
            [
                {k: v.detach().clone() for k, v in full.module[i].state_dict().items()}
                for i in range(start, end)
            ]
        )
    opt = FusedSGD(engine.parameters(), lr=lr)
    losses = []
    for _ in range(steps):
        opt.zero_grad()
        losses.append(engine.run_iteration(batch, labels, num_microbatches=M,
                                           schedule=schedule))
        opt.step()
    if rank == 0:
        np.save(f"{out_dir}/losses_p.npy", np.array(losses, dtype=np.float64))
    comm.barrier()
    from skycomputing_amd.parallel import destroy

    destroy()


@pytest.mark.parametrize("case", ["lopsided_gpipe", "thin_middle_1f1b"])
def test_adversarial_partitions_match_local(tmp_path, case):
    """Extreme partitions the allocator can legitimately produce under
    heterogeneity: a 1-layer first stage with everything else on stage 2,
    and a 3-stage plan whose middle stage holds a single layer (1F1B)."""
    layer_cfgs = tiny_bert_cfg(2)  # 9 layers
    L = len(layer_cfgs)
    batch, labels = _make_batch()
    lr, steps = 0.05, 2
    if case == "lopsided_gpipe":
        world, M, schedule = 2, 8, "gpipe"
        ranges = [(0, 1), (1, L)]
        port = 29850
    else:
        world, M, schedule = 3, 4, "1f1b"
        ranges = [(0, 4), (4, 5), (5, L)]
        port = 29860
    ref = _single_process_reference(layer_cfgs, batch, labels, lr, steps, M)
    run_multiprocess(_plan_worker, world, port, layer_cfgs, batch, labels,
                     lr, steps, M, schedule, ranges, str(tmp_path))
    got = np.load(f"{tmp_path}/losses_p.npy")
    assert np.allclose(got, np.array(ref), rtol=1e-4, atol=1e-5), (got, ref)


def _resume_worker(rank, world_size, layer_cfgs, out_dir):
    torch.manual_seed(0)
    from skycomputing_amd.builder import build_dataloader_from_cfg
    from skycomputing_amd.optim import FusedSGD
    from skycomputing_amd.parallel import PartitionPlan, PipelineEngine, init_distributed
    from skycomputing_amd.runner import CheckpointHook, Runner

    comm = init_distributed(backend="gloo", timeout_s=60)
    loader = build_dataloader_from_cfg(
        4, dict(type="SyntheticGlueDataset", size=16, max_seq_length=8,
                vocab_size=500, num_class=3, seed=3))
    plan = PartitionPlan(stage_ranks=[0], ranges=[(0, len(layer_cfgs))])

    def make_runner(max_epoch, hook):
        eng = PipelineEngine(comm, layer_cfgs, plan,
                             loss_fn=torch.nn.CrossEntropyLoss(),
                             stage_kwargs=dict(record_forward_time=False))
        opt = FusedSGD(eng.parameters(), lr=0.01)
        r = Runner(eng, opt, comm, max_epoch=max_epoch)
        r.register_hook(hook)
        return r

    # phase 1: train 2 epochs, checkpoint each
    r1 = make_runner(2, CheckpointHook(save_path=f"{out_dir}/ck", save_interval=1))
    r1.train(loader)
    assert r1.epoch == 1 and r1.iter == 8  # 2 epochs x 4 batches

    # phase 2: resume from epoch_1 with counters -> trains ONLY epoch 2
    r2 = make_runner(2, CheckpointHook(
        save_path=f"{out_dir}/ck2", save_interval=1,
        load_from=f"{out_dir}/ck/epoch_1.pth", resume_counters=True))
    r2.train(loader)
    assert r2.epoch == 1
    assert r2.iter == 4 + 4  # resumed at iter 4, ran one more epoch
    comm.barrier()
    from skycomputing_amd.parallel import destroy

    destroy()


def test_runner_resume_counters(tmp_path):
    run_multiprocess(_resume_worker, 1, 29870, tiny_bert_cfg(1), str(tmp_path))


def _best_ckpt_worker(rank, world_size, layer_cfgs, out_dir):
    torch.manual_seed(2)
    from skycomputing_amd.builder import build_dataloader_from_cfg
    from skycomputing_amd.optim import FusedSGD
    from skycomputing_amd.parallel import PartitionPlan, PipelineEngine, init_distributed
    from skycomputing_amd.runner import CheckpointHook, Runner

    comm = init_distributed(backend="gloo", timeout_s=60)
    loader = build_dataloader_from_cfg(
        4, dict(type="SyntheticGlueDataset", size=8, max_seq_length=8,
                vocab_size=500, num_class=3, seed=5))
    plan = PartitionPlan(stage_ranks=[0], ranges=[(0, len(layer_cfgs))])
    eng = PipelineEngine(comm, layer_cfgs, plan,
                         loss_fn=torch.nn.CrossEntropyLoss(),
                         stage_kwargs=dict(record_forward_time=False))
    opt = FusedSGD(eng.parameters(), lr=0.01)
    r = Runner(eng, opt, comm, max_epoch=1)
    hook = CheckpointHook(save_path=f"{out_dir}/ck", save_best=True)
    r.register_hook(hook)
    r.train(loader)
    acc1 = r.val(loader, max_batches=2)
    assert acc1 is not None
    import os as _os

    assert _os.path.isfile(f"{out_dir}/ck/best.pth")
    # a worse epoch must not overwrite: simulate by forcing best very high
    hook._best_acc = 2.0
    mtime = _os.path.getmtime(f"{out_dir}/ck/best.pth")
    r.val(loader, max_batches=2)
    assert _os.path.getmtime(f"{out_dir}/ck/best.pth") == mtime
    comm.barrier()
    from skycomputing_amd.parallel import destroy

    destroy()


def test_best_checkpoint_tracking(tmp_path):
    run_multiprocess(_best_ckpt_worker, 1, 29900, tiny_bert_cfg(1), str(tmp_path))


def _four_stage_1f1b_worker(rank, world_size, layer_cfgs, batch, labels, lr,
                            steps, M, out_dir):
    torch.manual_seed(1234)
    from skycomputing_amd.builder import build_module_from_cfg
    from skycomputing_amd.optim import FusedSGD
    from skycomputing_amd.parallel import PartitionPlan, PipelineEngine, init_distributed

    comm = init_distributed(backend="gloo", timeout_s=60)
    full = build_module_from_cfg(layer_cfgs, record_forward_time=False)
    L = len(layer_cfgs)
    base, rem = divmod(L, world_size)
    bounds = [0]
    for i in range(world_size):
        bounds.append(bounds[-1] + base + (1 if i < rem else 0))
    plan = PartitionPlan(stage_ranks=list(range(world_size)),
                         ranges=[(bounds[i], bounds[i + 1]) for i in range(world_size)])
    engine = PipelineEngine(comm, layer_cfgs, plan, loss_fn=torch.nn.CrossEntropyLoss(),
                            stage_kwargs=dict(record_forward_time=False))
    start, end = plan.ranges[engine.stage_idx]
    engine.stage.load_layer_state_dicts(
        [
            {k: v.detach().clone() for k, v in full.module[i].state_dict().items()}
            for i in range(start, end)
        ]
    )
    opt = FusedSGD(engine.parameters(), lr=lr)
    losses = []
    for _ in range(steps):
        opt.zero_grad()
        losses.append(engine.run_iteration(batch, labels, num_microbatches=M,
                                           schedule="1f1b"))
        opt.step()
    if rank == 0:
        np.save(f"{out_dir}/losses_1f1b4.npy", np.array(losses, dtype=np.float64))
    comm.barrier()
    from skycomputing_amd.parallel import destroy

    destroy()


def test_four_stage_1f1b_matches_local(tmp_path):
    """1F1B at 4 stages with M=8 (deeper warmup/steady/cooldown phases
    than the 3-stage case) — losses must match the local reference."""
    layer_cfgs = tiny_bert_cfg(2)
    batch, labels = _make_batch()
    lr, steps, M = 0.05, 2, 8
    ref = _single_process_reference(layer_cfgs, batch, labels, lr, steps, M)
    run_multiprocess(_four_stage_1f1b_worker, 4, 29920, layer_cfgs, batch,
                     labels, lr, steps, M, str(tmp_path))
    got = np.load(f"{tmp_path}/losses_1f1b4.npy")
    assert np.allclose(got, np.array(ref), rtol=1e-4, atol=1e-5), (got, ref)
