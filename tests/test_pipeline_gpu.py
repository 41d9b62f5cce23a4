"""Single-GPU end-to-end tests on MI355X (bf16, HIP kernels, full model)."""

from __future__ import annotations

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _require_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from skycomputing_amd.ops import hiplib

    assert hiplib.available()


def test_bert_stage_trains_bf16():
    """Loss decreases over a few steps of the full fused-op bf16 path."""
    torch.manual_seed(0)
    from skycomputing_amd.builder import build_module_from_cfg
    from skycomputing_amd.models import bert_pipeline_config
    from skycomputing_amd.optim import FusedSGD

    cfgs = bert_pipeline_config(
        2, dict(hidden_size=512, num_attention_heads=8, intermediate_size=2048,
                vocab_size=5000, hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0)
    )
    stage = build_module_from_cfg(cfgs, device="cuda:0", dtype=torch.bfloat16)
    opt = FusedSGD(stage.parameters(), lr=1e-2)
    ids = torch.randint(0, 5000, (16, 64), device="cuda")
    tids = torch.zeros_like(ids)
    mask = torch.ones_like(ids)
    labels = torch.randint(0, 3, (16,), device="cuda")
    losses = []
    for _ in range(12):
        opt.zero_grad()
        logits = stage(ids, tids, mask)
        loss = torch.nn.functional.cross_entropy(logits.float(), labels)
        loss.backward()
        opt.step()
        losses.append(float(loss))
    assert all(torch.isfinite(torch.tensor(losses)))
    assert losses[-1] < losses[0] * 0.8, losses


def test_bf16_step_matches_fp32_reference():
    """One training step of a small BERT stage in bf16 (HIP kernels) tracks
    the fp32 eager reference within bf16 tolerance."""
    torch.manual_seed(1)
    from skycomputing_amd.builder import build_module_from_cfg

    cfgs_kwargs = dict(hidden_size=256, num_attention_heads=4, intermediate_size=1024,
                       vocab_size=2000, hidden_dropout_prob=0.0,
                       attention_probs_dropout_prob=0.0)
    from skycomputing_amd.models import bert_pipeline_config

    cfgs = bert_pipeline_config(1, cfgs_kwargs)
    torch.manual_seed(42)
    gpu = build_module_from_cfg(cfgs, device="cuda:0", dtype=torch.bfloat16)
    torch.manual_seed(42)
    cpu = build_module_from_cfg(cfgs, device="cpu")
    ids = torch.randint(0, 2000, (8, 32))
    tids = torch.zeros_like(ids)
    mask = torch.ones_like(ids)
    lg = gpu(ids.cuda(), tids.cuda(), mask.cuda())
    lc = cpu(ids, tids, mask)
    assert torch.allclose(lg.float().cpu(), lc, atol=0.1, rtol=0.1), (
        (lg.float().cpu() - lc).abs().max()
    )
    labels = torch.randint(0, 3, (8,))
    torch.nn.functional.cross_entropy(lg.float(), labels.cuda()).backward()
    torch.nn.functional.cross_entropy(lc.float(), labels).backward()
    # compare a representative param grad (embeddings LN weight)
    g_gpu = gpu.module[0].layer_norm.weight.grad.float().cpu()
    g_cpu = cpu.module[0].layer_norm.weight.grad
    assert torch.allclose(g_gpu, g_cpu, atol=0.05, rtol=0.1), (g_gpu - g_cpu).abs().max()


def test_device_benchmark_and_detect_mem():
    from skycomputing_amd.builder import build_module_from_cfg
    from skycomputing_amd.dynamics.benchmarker import default_bert_probe_cfg

    stage = build_module_from_cfg(
        default_bert_probe_cfg(512), device="cuda:0", dtype=torch.bfloat16
    )
    free = stage.detect_mem()
    assert free > 1 << 30  # at least 1 GB free on a 288 GB part

    from skycomputing_amd.ops import hiplib
    import ctypes

    f = ctypes.c_uint64()
    t = ctypes.c_uint64()
    rc = hiplib.lib().sky_detect_mem(ctypes.addressof(f), ctypes.addressof(t))
    assert rc == 0 and t.value > 200 * (1 << 30)  # 288 GB part


def test_graft_smoke():
    import __graft_entry__

    __graft_entry__.smoke()


def test_graphed_train_step():
    """hipGraph-captured step: finite decreasing loss, dropout mask varies
    across replays (device RNG tick)."""
    torch.manual_seed(2)
    from skycomputing_amd.builder import build_module_from_cfg
    from skycomputing_amd.models import bert_pipeline_config
    from skycomputing_amd.optim import FusedSGD
    from skycomputing_amd.parallel.graph import GraphedTrainStep

    cfgs = bert_pipeline_config(
        2, dict(hidden_size=256, num_attention_heads=4, intermediate_size=1024,
                vocab_size=2000, hidden_dropout_prob=0.1,
                attention_probs_dropout_prob=0.1)
    )
    stage = build_module_from_cfg(cfgs, device="cuda:0", dtype=torch.bfloat16,
                                  record_forward_time=False)
    opt = FusedSGD(stage.parameters(), lr=1e-2)
    ids = torch.randint(0, 2000, (8, 32), device="cuda")
    inputs = [ids, torch.zeros_like(ids), torch.ones_like(ids)]
    labels = torch.randint(0, 3, (8,), device="cuda")
    g = GraphedTrainStep(
        stage, opt,
        lambda lg, lb: torch.nn.functional.cross_entropy(lg.float(), lb),
        inputs, labels,
    )
    losses = []
    for _ in range(10):
        g.step(inputs, labels)
        losses.append(g.loss_value())
    assert all(torch.isfinite(torch.tensor(losses)))
    assert losses[-1] < losses[0], losses
    # with dropout active, replayed losses on identical data must differ
    # (device RNG tick) once weights stop changing much; check variation
    assert len({round(l, 6) for l in losses}) > 1


def test_dropout_varies_across_ticks():
    from skycomputing_amd.ops import functions as F

    x = torch.ones(4096, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    y1 = F.DropoutFn.apply(x, 0.5)
    F.rng_tick()
    torch.cuda.synchronize()
    # same salt path can't be forced (new salt per call), so check the
    # underlying kernel directly with a fixed salt before/after tick
    from skycomputing_amd.ops import hiplib
    from skycomputing_amd.ops.hiplib import check, ptr

    lib = hiplib.require()
    out_a = torch.empty_like(x)
    out_b = torch.empty_like(x)
    st = F.rng_state()
    stream = torch.cuda.current_stream().cuda_stream
    check(lib.sky_dropout_fwd(stream, ptr(x), ptr(out_a), x.numel(), 0.5, 123,
                              st.data_ptr(), 1), "f")
    F.rng_tick()
    check(lib.sky_dropout_fwd(stream, ptr(x), ptr(out_b), x.numel(), 0.5, 123,
                              st.data_ptr(), 1), "f")
    torch.cuda.synchronize()
    assert not torch.equal(out_a, out_b)


def test_graphed_pipeline_step_single_stage():
    """GraphedPipelineStep (per-microbatch stage graphs, the multi-rank
    executor) validated at world=1: must reproduce the eager schedule's
    loss and train."""
    torch.manual_seed(3)
    from skycomputing_amd.builder import build_module_from_cfg  # noqa: F401
    from skycomputing_amd.models import bert_pipeline_config
    from skycomputing_amd.optim import FusedSGD
    from skycomputing_amd.parallel import (
        GraphedPipelineStep, PartitionPlan, PipelineEngine, init_distributed,
    )

    comm = init_distributed()
    cfgs = bert_pipeline_config(
        2, dict(hidden_size=256, num_attention_heads=4, intermediate_size=1024,
                vocab_size=2000, hidden_dropout_prob=0.0,
                attention_probs_dropout_prob=0.0)
    )
    plan = PartitionPlan(stage_ranks=[0], ranges=[(0, len(cfgs))])
    engine = PipelineEngine(
        comm, cfgs, plan,
        loss_fn=lambda lg, lb: torch.nn.functional.cross_entropy(lg.float(), lb),
        dtype=torch.bfloat16,
        stage_kwargs=dict(record_forward_time=False),
    )
    ids = torch.randint(0, 2000, (16, 32))
    inputs = (ids, torch.zeros_like(ids), torch.ones_like(ids))
    labels = torch.randint(0, 3, (16,))

    # lr=0: warmup/capture must not move weights; graphed loss == eager loss
    opt0 = FusedSGD(engine.parameters(), lr=0.0)
    eager_loss = engine.run_iteration(inputs, labels, num_microbatches=4, schedule="gpipe")
    g = GraphedPipelineStep(engine, opt0, 4, inputs, labels)
    graphed_loss = g.step(inputs, labels)
    assert abs(graphed_loss - eager_loss) < 0.05, (graphed_loss, eager_loss)
    # replay is stable
    graphed_loss2 = g.step(inputs, labels)
    assert abs(graphed_loss2 - graphed_loss) < 1e-4

    # now a real-lr executor must train
    torch.manual_seed(4)
    engine2 = PipelineEngine(
        comm, cfgs, plan,
        loss_fn=lambda lg, lb: torch.nn.functional.cross_entropy(lg.float(), lb),
        dtype=torch.bfloat16,
        stage_kwargs=dict(record_forward_time=False),
    )
    opt = FusedSGD(engine2.parameters(), lr=1e-2)
    g2 = GraphedPipelineStep(engine2, opt, 4, inputs, labels)
    losses = [g2.step(inputs, labels) for _ in range(10)]
    assert all(torch.isfinite(torch.tensor(losses)))
    assert losses[-1] < losses[0] * 0.9, losses


def test_interleaved_engine_single_rank_v2():
    """InterleavedPipelineEngine at world=1, v=2 (both chunks local, the
    mailbox/gradbox path) on bf16 HIP kernels: must match the one-chunk
    eager pipeline's loss and train."""
    torch.manual_seed(5)
    from skycomputing_amd.models import bert_pipeline_config
    from skycomputing_amd.optim import FusedSGD
    from skycomputing_amd.parallel import PartitionPlan, PipelineEngine, init_distributed
    from skycomputing_amd.parallel.interleaved import (
        InterleavedPipelineEngine, build_interleaved_plan,
    )

    comm = init_distributed()
    cfg = dict(hidden_size=256, num_attention_heads=4, intermediate_size=1024,
               vocab_size=2000, hidden_dropout_prob=0.0,
               attention_probs_dropout_prob=0.0)
    cfgs = bert_pipeline_config(2, cfg)
    ids = torch.randint(0, 2000, (16, 32))
    inputs = (ids, torch.zeros_like(ids), torch.ones_like(ids))
    labels = torch.randint(0, 3, (16,))
    lf = lambda lg, lb: torch.nn.functional.cross_entropy(lg.float(), lb.to(lg.device))  # noqa: E731

    torch.manual_seed(6)
    ref_engine = PipelineEngine(
        comm, cfgs, PartitionPlan(stage_ranks=[0], ranges=[(0, len(cfgs))]),
        loss_fn=lf, dtype=torch.bfloat16,
        stage_kwargs=dict(record_forward_time=False),
    )
    ref_loss = ref_engine.run_iteration(inputs, labels, num_microbatches=4,
                                        schedule="gpipe")

    torch.manual_seed(6)
    plan = build_interleaved_plan(len(cfgs), 1, 2)
    engine = InterleavedPipelineEngine(
        comm, cfgs, plan, loss_fn=lf, dtype=torch.bfloat16,
        stage_kwargs=dict(record_forward_time=False),
    )
    il_loss = engine.run_iteration(inputs, labels, num_microbatches=4)
    assert abs(il_loss - ref_loss) < 0.05, (il_loss, ref_loss)

    opt = FusedSGD(engine.parameters(), lr=1e-2)
    losses = []
    for _ in range(8):
        opt.zero_grad()
        losses.append(engine.run_iteration(inputs, labels, num_microbatches=4))
        opt.step()
    assert all(torch.isfinite(torch.tensor(losses)))
    assert losses[-1] < losses[0] * 0.95, losses


def test_graphed_interleaved_single_rank_v2():
    """GraphedInterleavedStep at world=1, v=2: the zero-copy local
    boundary (aliased pool leaves) and per-chunk graph replay must
    reproduce the eager interleaved loss and train."""
    torch.manual_seed(7)
    from skycomputing_amd.models import bert_pipeline_config
    from skycomputing_amd.optim import FusedSGD
    from skycomputing_amd.parallel import init_distributed
    from skycomputing_amd.parallel.interleaved import (
        InterleavedPipelineEngine, build_interleaved_plan,
    )
    from skycomputing_amd.parallel.interleaved_graph import GraphedInterleavedStep

    comm = init_distributed()
    cfg = dict(hidden_size=256, num_attention_heads=4, intermediate_size=1024,
               vocab_size=2000, hidden_dropout_prob=0.0,
               attention_probs_dropout_prob=0.0)
    cfgs = bert_pipeline_config(2, cfg)
    ids = torch.randint(0, 2000, (16, 32))
    inputs = (ids, torch.zeros_like(ids), torch.ones_like(ids))
    labels = torch.randint(0, 3, (16,))
    lf = lambda lg, lb: torch.nn.functional.cross_entropy(lg.float(), lb.to(lg.device))  # noqa: E731

    torch.manual_seed(8)
    plan = build_interleaved_plan(len(cfgs), 1, 2)
    engine = InterleavedPipelineEngine(
        comm, cfgs, plan, loss_fn=lf, dtype=torch.bfloat16,
        stage_kwargs=dict(record_forward_time=False),
    )
    opt0 = FusedSGD(engine.parameters(), lr=0.0)
    eager_loss = engine.run_iteration(inputs, labels, num_microbatches=4)
    g = GraphedInterleavedStep(engine, opt0, 4, inputs, labels)
    graphed_loss = g.step(inputs, labels)
    assert abs(graphed_loss - eager_loss) < 0.05, (graphed_loss, eager_loss)
    graphed_loss2 = g.step(inputs, labels)
    assert abs(graphed_loss2 - graphed_loss) < 1e-4

    torch.manual_seed(9)
    engine2 = InterleavedPipelineEngine(
        comm, cfgs, plan, loss_fn=lf, dtype=torch.bfloat16,
        stage_kwargs=dict(record_forward_time=False),
    )
    opt = FusedSGD(engine2.parameters(), lr=1e-2)
    g2 = GraphedInterleavedStep(engine2, opt, 4, inputs, labels)
    losses = [g2.step(inputs, labels) for _ in range(10)]
    assert all(torch.isfinite(torch.tensor(losses)))
    assert losses[-1] < losses[0] * 0.9, losses
