"""End-to-end CLI and subsystem integration tests (CPU, gloo)."""

from __future__ import annotations

import os
import subprocess
import sys

import numpy as np
import pytest
import torch

from .helpers import run_multiprocess, tiny_bert_cfg

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

TINY_CONFIG = """
model_config = dict(
    kind="bert",
    num_encoder_layers=2,
    bert_config=dict(hidden_size=64, num_attention_heads=4, intermediate_size=128,
                     vocab_size=500, max_position_embeddings=64,
                     hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0),
    num_class=3,
)
data_config = dict(
    batch_size=8,
    dataset=dict(layer_type="SyntheticGlueDataset", size=64, max_seq_length=16,
                 vocab_size=500, num_class=3, seed=1),
)
worker_config = [dict(slowdown=0.0), dict(slowdown=4.0)]
allocator_config = dict(
    mode="optimal",
    benchmark=dict(batch_size=4, seq_len=16, hidden=64, iterations=2),
    stimulate=False,
)
train_config = dict(
    max_epoch=1, max_iter=3,
    optimizer=dict(lr=0.01),
    num_microbatches=2, schedule="gpipe", log_interval=1,
    hooks=[dict(layer_type="TimerHook"), dict(layer_type="StopHook", root=".")],
)
logging_config = dict(log_dir="{logdir}")
"""


def test_launch_cli_two_ranks_optimal(tmp_path):
    """Full driver path under torch.distributed.run: benchmark -> optimal
    allocation (rank 1 slowed 4x) -> pipeline training with hooks."""
    cfg = tmp_path / "cfg.py"
    cfg.write_text(TINY_CONFIG.format(logdir=str(tmp_path / "logs")))
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    r = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run", "--standalone",
            "--local-addr", "127.0.0.1", "--nproc-per-node", "2",
            os.path.join(REPO, "experiment", "launch.py"),
            "-c", str(cfg),
        ],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=400,
    )
    assert r.returncode == 0, r.stdout[-3000:] + r.stderr[-3000:]
    log = (tmp_path / "logs" / "rank0.log").read_text()
    assert "allocation (optimal)" in log
    assert "done: 3 iterations" in log
    # slowed rank 1 must get the smaller slice
    import re

    sizes = {
        int(r): int(b) - int(a)
        for r, a, b in re.findall(r"r(\d+):\[(\d+),(\d+)\)", log)
    }
    assert sizes[1] < sizes[0], sizes


def _ckpt_worker(rank, world, layer_cfgs, out_dir):
    torch.manual_seed(100 + rank)
    from skycomputing_amd.dynamics import ParameterServer
    from skycomputing_amd.parallel import PartitionPlan, PipelineEngine, destroy, init_distributed

    comm = init_distributed(backend="gloo", timeout_s=60)
    L = len(layer_cfgs)
    plan = PartitionPlan(stage_ranks=[0, 1], ranges=[(0, L // 2), (L // 2, L)])
    engine = PipelineEngine(comm, layer_cfgs, plan,
                            loss_fn=torch.nn.CrossEntropyLoss(),
                            stage_kwargs=dict(record_forward_time=False))
    ps = ParameterServer(L)
    ps.gather_from_engine(engine, comm)
    path = os.path.join(out_dir, "epoch_1.pth")
    if rank == 0:
        ps.save_weights_to_file(path, meta={"epoch": 1})
    comm.barrier()

    # restore into a REPARTITIONED engine (checkpoints are partition-portable)
    plan2 = PartitionPlan(stage_ranks=[1, 0], ranges=[(0, 2), (2, L)])
    engine2 = PipelineEngine(comm, layer_cfgs, plan2,
                             loss_fn=torch.nn.CrossEntropyLoss(),
                             stage_kwargs=dict(record_forward_time=False))
    ps2 = ParameterServer(L)
    if rank == 0:
        meta = ps2.load_weights_from_file(path)
        assert meta["epoch"] == 1
    ps2.scatter_to_engine(engine2, comm)

    # verify: engine2's layers now match engine's saved weights
    ps3 = ParameterServer(L)
    ps3.gather_from_engine(engine2, comm)
    if rank == 0:
        for i in range(L):
            sd_a = ps.get_state_dict(i)
            sd_b = ps3.get_state_dict(i)
            for k in sd_a:
                assert torch.allclose(sd_a[k], sd_b[k]), (i, k)
    comm.barrier()
    destroy()


def test_checkpoint_partition_portable(tmp_path):
    run_multiprocess(_ckpt_worker, 2, 29720, tiny_bert_cfg(2), str(tmp_path))


def _idle_rank_scatter_worker(rank, world, layer_cfgs, out_dir):
    torch.manual_seed(200 + rank)
    from skycomputing_amd.dynamics import ParameterServer
    from skycomputing_amd.parallel import PartitionPlan, PipelineEngine, destroy, init_distributed

    comm = init_distributed(backend="gloo", timeout_s=60)
    L = len(layer_cfgs)
    # rank 2 owns NO stage: sharded scatter must skip it without deadlock
    plan = PartitionPlan(stage_ranks=[0, 1], ranges=[(0, L // 2), (L // 2, L)])
    engine = PipelineEngine(comm, layer_cfgs, plan,
                            loss_fn=torch.nn.CrossEntropyLoss(),
                            stage_kwargs=dict(record_forward_time=False))
    ps = ParameterServer(L)
    ps.gather_from_engine(engine, comm)
    ps2 = ParameterServer(L)
    if rank == 0:
        ps2._layers = [dict(sd) for sd in ps._layers]  # pretend-restored
    ps2.scatter_to_engine(engine, comm)
    # stage ranks must have loaded their slices; idle rank returned early
    if engine.stage_idx is not None:
        ps3 = ParameterServer(L)
        ps3.gather_from_engine(engine, comm)
    else:
        ps3 = ParameterServer(L)
        ps3.gather_from_engine(engine, comm)
    if rank == 0:
        for i in range(L):
            for k, v in ps.get_state_dict(i).items():
                assert torch.allclose(v, ps3.get_state_dict(i)[k]), (i, k)
    comm.barrier()
    destroy()


def test_sharded_scatter_with_idle_rank(tmp_path):
    """3 ranks, 2 stages: the per-rank sharded scatter must neither hang
    nor send anything to the stage-less rank."""
    run_multiprocess(_idle_rank_scatter_worker, 3, 29725, tiny_bert_cfg(2),
                     str(tmp_path))


def _stop_worker(rank, world, layer_cfgs, root):
    torch.manual_seed(5)
    from skycomputing_amd.optim import FusedSGD
    from skycomputing_amd.parallel import PartitionPlan, PipelineEngine, destroy, init_distributed
    from skycomputing_amd.runner import Runner, StopHook

    comm = init_distributed(backend="gloo", timeout_s=60)
    L = len(layer_cfgs)
    plan = PartitionPlan(stage_ranks=[0, 1], ranges=[(0, L // 2), (L // 2, L)])
    engine = PipelineEngine(comm, layer_cfgs, plan,
                            loss_fn=torch.nn.CrossEntropyLoss(),
                            stage_kwargs=dict(record_forward_time=False))
    opt = FusedSGD(engine.parameters(), lr=0.01)
    runner = Runner(engine, opt, comm, max_epoch=1, max_iter=50, log_interval=100)
    runner.register_hook(StopHook(root=root))

    def flip(r):
        if r.iter == 1 and rank == 0:
            StopHook.stop(root)

    from skycomputing_amd.runner import Hook

    class Flipper(Hook):
        def after_train_iter(self, r):
            flip(r)

    runner.hooks.insert(0, Flipper())

    from skycomputing_amd.dataset import SyntheticGlueDataset

    ds = SyntheticGlueDataset(size=64, max_seq_length=16, vocab_size=500, seed=2)
    loader = torch.utils.data.DataLoader(ds, batch_size=8, drop_last=True)
    runner.train(loader)
    assert runner.iter <= 3, runner.iter  # stopped early, together
    comm.barrier()
    destroy()


def test_stop_hook_cooperative(tmp_path):
    run_multiprocess(_stop_worker, 2, 29750, tiny_bert_cfg(1), str(tmp_path))


def _glue_train_worker(rank, world, layer_cfgs, data_dir):
    torch.manual_seed(11)
    from skycomputing_amd.dataset import GlueDataset
    from skycomputing_amd.optim import FusedSGD
    from skycomputing_amd.parallel import PartitionPlan, PipelineEngine, destroy, init_distributed
    from skycomputing_amd.runner import Runner

    comm = init_distributed(backend="gloo", timeout_s=60)
    L = len(layer_cfgs)
    plan = PartitionPlan(stage_ranks=[0, 1], ranges=[(0, L // 2), (L // 2, L)])
    engine = PipelineEngine(comm, layer_cfgs, plan,
                            loss_fn=torch.nn.CrossEntropyLoss(),
                            stage_kwargs=dict(record_forward_time=False))
    opt = FusedSGD(engine.parameters(), lr=0.05)
    runner = Runner(engine, opt, comm, max_epoch=2, max_iter=4, log_interval=100)
    ds = GlueDataset(data_dir, task="mnli", max_seq_length=16)
    loader = torch.utils.data.DataLoader(ds, batch_size=2, drop_last=True)
    runner.train(loader)
    if rank == world - 1:
        assert runner.last_loss is not None and runner.last_loss == runner.last_loss
    comm.barrier()
    destroy()


def test_tokenized_mnli_training_from_disk(tmp_path):
    """End-to-end: real (tiny) MNLI tsv on disk -> WordPiece tokenization ->
    pipeline training. Demonstrates synthetic data is a bench choice, not a
    capability limit (reference trains on tokenized MNLI,
    experiment/launch.py:20-236)."""
    pytest.importorskip("transformers")
    vocab = ["[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]",
             "the", "cat", "sat", "mat", "dog", "ran", "a", "on"]
    (tmp_path / "vocab.txt").write_text("\n".join(vocab) + "\n")
    rows = ["index\tsentence1\tsentence2\tgold_label"]
    labels = ["entailment", "neutral", "contradiction"]
    for i in range(8):
        rows.append(f"{i}\tthe cat sat on a mat\tthe dog ran\t{labels[i % 3]}")
    (tmp_path / "train.tsv").write_text("\n".join(rows) + "\n")
    run_multiprocess(_glue_train_worker, 2, 29770, tiny_bert_cfg(1), str(tmp_path))


def test_glue_mrpc_and_cola_readers(tmp_path):
    """MRPC (headered pair-sentence) and CoLA (headerless) processors
    (reference: scaelum/dataset/glue/processor.py:305-310)."""
    pytest.importorskip("transformers")
    vocab = ["[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]", "the", "cat", "sat"]
    mrpc = tmp_path / "mrpc"
    mrpc.mkdir()
    (mrpc / "vocab.txt").write_text("\n".join(vocab) + "\n")
    (mrpc / "train.tsv").write_text(
        "Quality\t#1 ID\t#2 ID\t#1 String\t#2 String\n"
        "1\t10\t11\tthe cat sat\tthe cat\n"
        "0\t12\t13\tthe cat\tsat\n")
    from skycomputing_amd.dataset import GlueDataset

    ds = GlueDataset(str(mrpc), task="mrpc", max_seq_length=12)
    assert len(ds) == 2
    (ids, mask, tids), label = ds[0]
    assert int(label) == 1 and int(tids.max()) == 1  # two segments
    cola = tmp_path / "cola"
    cola.mkdir()
    (cola / "vocab.txt").write_text("\n".join(vocab) + "\n")
    # CoLA: NO header row; cols = source, label, star, sentence
    (cola / "train.tsv").write_text(
        "gj04\t1\t\tthe cat sat\ngj04\t0\t*\tcat the\n")
    ds2 = GlueDataset(str(cola), task="cola", max_seq_length=12)
    assert len(ds2) == 2  # first line must NOT be dropped as a header
    assert int(ds2[0][1]) == 1 and int(ds2[1][1]) == 0


def test_resnet_pipeline_builds_and_runs():
    from skycomputing_amd.builder import build_module_from_cfg
    from skycomputing_amd.models import resnet_pipeline_config

    cfgs = resnet_pipeline_config(18, num_class=10)
    stage = build_module_from_cfg(cfgs, record_forward_time=False)
    x = torch.randn(2, 3, 32, 32)
    out = stage(x)
    assert out.shape == (2, 10)
    out.sum().backward()


def _replan_worker(rank, world, layer_cfgs, batch, labels, out_dir):
    """Train 2 iters on plan A, migrate weights through the ParameterServer
    to a different partition, continue training — loss sequence must be
    identical to training on a fixed plan (same math, different placement)."""
    torch.manual_seed(77)
    from skycomputing_amd.dynamics import ParameterServer
    from skycomputing_amd.optim import FusedSGD
    from skycomputing_amd.parallel import PartitionPlan, PipelineEngine, destroy, init_distributed
    from skycomputing_amd.builder import build_module_from_cfg

    comm = init_distributed(backend="gloo", timeout_s=60)
    L = len(layer_cfgs)
    full = build_module_from_cfg(layer_cfgs, record_forward_time=False)

    def fresh_engine(plan):
        e = PipelineEngine(comm, layer_cfgs, plan,
                           loss_fn=torch.nn.CrossEntropyLoss(),
                           stage_kwargs=dict(record_forward_time=False))
        return e

    def seed_from_full(e):
        s0, e0 = e.plan.ranges[e.stage_idx]
        e.stage.load_layer_state_dicts(
            [{k: v.detach().clone() for k, v in full.module[i].state_dict().items()}
             for i in range(s0, e0)]
        )

    lr = 0.05
    planA = PartitionPlan(stage_ranks=[0, 1], ranges=[(0, L // 2), (L // 2, L)])
    planB = PartitionPlan(stage_ranks=[1, 0], ranges=[(0, 2), (2, L)])

    # run: 2 iters on A -> migrate -> 2 iters on B
    eng = fresh_engine(planA)
    seed_from_full(eng)
    opt = FusedSGD(eng.parameters(), lr=lr)
    losses = []
    for _ in range(2):
        opt.zero_grad()
        losses.append(eng.run_iteration(batch, labels, 2, "gpipe"))
        opt.step()
    ps = ParameterServer(L)
    ps.gather_from_engine(eng, comm)
    eng2 = fresh_engine(planB)
    ps.scatter_to_engine(eng2, comm)
    opt2 = FusedSGD(eng2.parameters(), lr=lr)
    for _ in range(2):
        opt2.zero_grad()
        losses.append(eng2.run_iteration(batch, labels, 2, "gpipe"))
        opt2.step()

    # reference: 4 iters on the fixed plan A
    eng3 = fresh_engine(planA)
    seed_from_full(eng3)
    opt3 = FusedSGD(eng3.parameters(), lr=lr)
    ref = []
    for _ in range(4):
        opt3.zero_grad()
        ref.append(eng3.run_iteration(batch, labels, 2, "gpipe"))
        opt3.step()

    if rank == 0:
        import numpy as np

        assert np.allclose(losses, ref, rtol=1e-4, atol=1e-5), (losses, ref)
    comm.barrier()
    destroy()


def test_reallocation_mid_training(tmp_path):
    from .helpers import run_multiprocess, tiny_bert_cfg
    import torch as _t

    layer_cfgs = tiny_bert_cfg(2)
    g = _t.Generator().manual_seed(0)
    ids = _t.randint(0, 500, (8, 16), generator=g)
    batch = (ids, _t.zeros_like(ids), _t.ones_like(ids))
    labels = _t.randint(0, 3, (8,), generator=g)
    run_multiprocess(_replan_worker, 2, 29910, layer_cfgs, batch, labels, str(tmp_path))


def test_baseline_configs_load():
    """The five BASELINE.json configurations ship as ready-to-run config
    files; each must parse through the config loader with the expected
    layer count and allocation mode."""
    import glob

    from skycomputing_amd.config import load_config

    expect = {
        "bert24_even_cpu.py": (24, "even", False),
        "bert24_optimal_8gpu.py": (24, "optimal", False),
        "bert160_dynamic_8gpu.py": (160, "dynamic", False),
        "bert96_stimulate_8gpu.py": (96, "optimal", True),
        "bert320_optimal_8gpu.py": (320, "optimal", False),
    }
    files = {os.path.basename(f): f
             for f in glob.glob(os.path.join(REPO, "experiment/configs/*.py"))}
    assert set(files) == set(expect)
    for name, (layers, mode, stim) in expect.items():
        cfg = load_config(files[name])
        assert cfg.model_config["num_encoder_layers"] == layers
        assert cfg.allocator_config["mode"] == mode
        assert bool(cfg.allocator_config.get("stimulate")) == stim
        assert cfg.data_config["batch_size"] == 32
        assert cfg.train_config["max_iter"] == 30


def test_launch_cli_virtual_stages(tmp_path):
    """The canonical CLI with allocator_config.virtual_stages=2 routes
    through the interleaved engine + interleaved_allocate end to end."""
    cfg = tmp_path / "cfgv.py"
    cfg.write_text(TINY_CONFIG.format(logdir=str(tmp_path / "logs"))
                   .replace('stimulate=False,', 'stimulate=False,\n    virtual_stages=2,'))
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    r = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run", "--standalone",
            "--local-addr", "127.0.0.1", "--nproc-per-node", "2",
            os.path.join(REPO, "experiment", "launch.py"),
            "-c", str(cfg),
        ],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=400,
    )
    assert r.returncode == 0, r.stdout[-3000:] + r.stderr[-3000:]
    log = (tmp_path / "logs" / "rank0.log").read_text()
    assert "done: 3 iterations" in log
