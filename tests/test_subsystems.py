"""Unit tests for the smaller subsystems: workers, stimulator, timers,
datasets, optimizer, parameter server."""

from __future__ import annotations

import pytest
import torch

from skycomputing_amd.dataset import (
    GlueDataset, RandomTokenGenerator, SyntheticGlueDataset,
)
from skycomputing_amd.dynamics import ParameterServer, Worker, WorkerManager
from skycomputing_amd.optim import FusedSGD
from skycomputing_amd.stimulator import Stimulator
from skycomputing_amd.timer import DeviceTimer, DistributedTimer



def test_worker_roundtrip_and_manager():
    w = Worker(rank=2, extra_config=dict(slowdown=1.5))
    w.model_config = (3, 7)
    w.order = 0
    w2 = Worker.from_dict(w.to_dict())
    assert w2.rank == 2 and w2.model_config == (3, 7) and w2.extra_config["slowdown"] == 1.5

    wm = WorkerManager.from_world(3)
    wm.assign_model_to_worker(0, (4, 8), order=1)
    wm.assign_model_to_worker(1, (0, 4), order=0)
    wm.assign_model_to_worker(2, (8, 10), order=2)
    order = [w.rank for w in wm.pipeline_order()]
    assert order == [1, 0, 2]
    wm.remove_worker(1)
    assert len(wm) == 2
    with pytest.raises(KeyError):
        wm.get_worker_by_rank(1)
    with pytest.raises(ValueError):
        wm.add_worker(Worker(rank=0))


def test_stimulator_deterministic_ranges():
    s1, s2 = Stimulator(8, seed=42), Stimulator(8, seed=42)
    assert (s1.compute_factors == s2.compute_factors).all()
    assert all(1 <= f < 4 for f in s1.compute_factors)
    assert all(1 <= f < 3 for f in s1.memory_factors)
    assert all(1 <= f < 2 for f in s1.network_factors)
    res = {0: {"time": 1.0, "avai_mem": 100.0}}
    scaled = s1.scale_benchmark(res)
    assert scaled[0]["time"] == pytest.approx(s1.compute_factor(0))
    assert scaled[0]["avai_mem"] == pytest.approx(100.0 / s1.memory_factor(0))


def test_device_timer_cpu():
    import time

    t = DeviceTimer(use_cuda=False)
    t.start(); time.sleep(0.02); t.stop()
    t.start(); time.sleep(0.01); t.stop()
    assert 0.025 < t.elapsed() < 0.5
    assert 0.008 < t.last() < 0.3
    t.reset()
    assert t.elapsed() == 0.0


def test_distributed_timer_intervals():
    import time

    dt = DistributedTimer()
    dt.add_timestamp("x"); time.sleep(0.01); dt.add_timestamp("x")
    assert dt.get_prev_interval("x") > 0.005
    assert dt.get_prev_interval("missing") == 0.0
    dt.clean()
    assert dt.get_prev_interval("x") == 0.0


def test_synthetic_glue_schema_and_determinism():
    a = SyntheticGlueDataset(size=16, max_seq_length=8, vocab_size=100, seed=5)
    b = SyntheticGlueDataset(size=16, max_seq_length=8, vocab_size=100, seed=5)
    (ids, mask, tids), label = a[3]
    assert ids.shape == (8,) and mask.shape == (8,) and tids.shape == (8,)
    assert torch.equal(a[3][0][0], b[3][0][0])
    assert (a.input_ids < 100).all()
    # type ids only where attended
    assert ((a.token_type_ids == 1) <= (a.attention_mask == 1)).all()


def test_glue_dataset_tokenizes_from_disk(tmp_path):
    vocab = ["[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]",
             "the", "cat", "sat", "mat", "dog", "ran"]
    (tmp_path / "vocab.txt").write_text("\n".join(vocab) + "\n")
    rows = ["index\tsentence1\tsentence2\tgold_label",
            "0\tthe cat sat\tthe dog ran\tentailment",
            "1\tthe mat\tthe cat\tneutral"]
    (tmp_path / "train.tsv").write_text("\n".join(rows) + "\n")
    try:
        ds = GlueDataset(str(tmp_path), task="mnli", max_seq_length=16)
    except ImportError:
        pytest.skip("transformers unavailable")
    assert len(ds) == 2
    (ids, mask, tids), label = ds[0]
    assert ids.shape == (16,)
    assert ids[0] == 2  # [CLS]
    assert int(label) == 1  # entailment
    assert mask.sum() >= 7


def test_glue_dataset_single_sentence_tasks(tmp_path):
    """sst-2/cola have no second sentence — tokenization must pass
    text_pair=None, not a list of empty strings (regression test)."""
    vocab = ["[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]", "the", "cat", "sat"]
    (tmp_path / "vocab.txt").write_text("\n".join(vocab) + "\n")
    (tmp_path / "train.tsv").write_text(
        "sentence\tlabel\nthe cat sat\t0\nthe cat\t1\n")
    try:
        ds = GlueDataset(str(tmp_path), task="sst-2", max_seq_length=12)
    except ImportError:
        pytest.skip("transformers unavailable")
    assert len(ds) == 2
    (ids, mask, tids), label = ds[1]
    assert ids.shape == (12,)
    assert int(label) == 1
    assert int(tids.max()) == 0  # single segment


def test_random_token_generator():
    g = RandomTokenGenerator(batch_size=2, seq_len=4, vocab_size=10)
    ids, mask, tids = g.generate()
    assert ids.shape == (2, 4) and (ids < 10).all()


def test_fused_sgd_state_dict_roundtrip():
    p = [torch.randn(8, requires_grad=True)]
    opt = FusedSGD(p, lr=0.1, momentum=0.9, master_weights=True)
    p[0].grad = torch.randn(8)
    opt.step()
    sd = opt.state_dict()
    opt2 = FusedSGD([p[0]], lr=0.1, momentum=0.9, master_weights=True)
    opt2.load_state_dict(sd)
    assert torch.allclose(opt2.masters[0], opt.masters[0])
    assert torch.allclose(opt2.momentum_bufs[0], opt.momentum_bufs[0])


def test_fused_sgd_sync_masters_after_restore():
    # out-of-band weight restore + stale fp32 masters
    # must not revert params on the next step. lr=0 step must be a no-op
    # after sync_masters().
    p = [torch.randn(8, dtype=torch.bfloat16, requires_grad=True)]
    opt = FusedSGD(p, lr=0.0, momentum=0.9)  # masters auto-enabled for bf16
    assert opt.masters is not None
    restored = torch.randn(8, dtype=torch.bfloat16)
    with torch.no_grad():
        p[0].copy_(restored)  # simulates checkpoint restore
    opt.sync_masters()
    p[0].grad = torch.randn(8, dtype=torch.bfloat16)
    opt.step()
    assert torch.equal(p[0].detach(), restored)


def test_fused_sgd_bf16_momentum_bufs_are_fp32():
    # momentum buffers must be fp32 regardless of
    # param dtype (the HIP multi-tensor kernel reads them as float*).
    p = [torch.randn(8, dtype=torch.bfloat16, requires_grad=True)]
    opt = FusedSGD(p, lr=0.1, momentum=0.9, master_weights=False)
    assert opt.momentum_bufs[0].dtype == torch.float32
    p[0].grad = torch.randn(8, dtype=torch.bfloat16)
    opt.step()  # eager path must handle fp32 bufs + bf16 params
    assert torch.isfinite(p[0].float()).all()


def test_parameter_server_save_load(tmp_path):
    ps = ParameterServer(2)
    ps.update_weights({"w": torch.randn(3)}, 0)
    ps.update_weights({"w": torch.randn(3)}, 1)
    path = str(tmp_path / "epoch_1.pth")
    ps.save_weights_to_file(path, meta={"epoch": 1})
    ps2 = ParameterServer(2)
    meta = ps2.load_weights_from_file(path)
    assert meta["epoch"] == 1
    assert torch.allclose(ps2.get_state_dict(0)["w"], ps.get_state_dict(0)["w"])
    ps3 = ParameterServer(3)
    with pytest.raises(ValueError):
        ps3.load_weights_from_file(path)
    ps4 = ParameterServer(2)
    with pytest.raises(RuntimeError):
        ps4.save_weights_to_file(str(tmp_path / "x.pth"))


def test_config_registry_hooks_buildable():
    from skycomputing_amd.builder import build_hook
    from skycomputing_amd.runner import CheckpointHook, StopHook, TimerHook

    assert isinstance(build_hook(dict(layer_type="TimerHook")), TimerHook)
    assert isinstance(build_hook(dict(layer_type="StopHook", root=".")), StopHook)
    assert isinstance(
        build_hook(dict(layer_type="CheckpointHook", save_path="/tmp/x")), CheckpointHook
    )


def test_dataloader_generator_advances_and_wraps():
    """The reference's DataloaderGenerator always returned the FIRST batch
    (scaelum/dataset/data_generator.py:30-34); ours must advance through
    the loader and wrap around at exhaustion."""
    from skycomputing_amd.dataset.data_generator import DataloaderGenerator

    gen = DataloaderGenerator(dict(
        batch_size=4,
        dataset_cfg=dict(type="RandomMlpDataset", size=8, dim=6, num_class=3),
    ))
    b1 = gen.generate()
    b2 = gen.generate()
    assert b1.shape == (4, 6)
    assert not torch.allclose(b1, b2)  # advanced, not the first batch again
    b3 = gen.generate()  # loader exhausted (8/4 = 2 batches) -> wraps
    assert torch.allclose(b1, b3)


def test_estimator_benchmark_speed_cpu():
    from skycomputing_amd.dataset.data_generator import RandomTensorGenerator
    from skycomputing_amd.dynamics.estimator import Estimator

    model = torch.nn.Sequential(torch.nn.Linear(16, 16), torch.nn.ReLU())
    gen = RandomTensorGenerator(size=(4, 16))
    t = Estimator.benchmark_speed(model, gen, iterations=3, warmup=1,
                                  backward=True, device=torch.device("cpu"))
    assert t > 0


def test_logger_writes_per_rank_file(tmp_path):
    from skycomputing_amd.logger import Logger

    log = tmp_path / "run.log"
    lg = Logger(log_file=str(log), rank=3)
    lg.info("hello world")
    text = log.read_text()
    assert "hello world" in text


def test_hook_cadence_helpers():
    from skycomputing_amd.runner.hooks import Hook

    class R:  # 0-based counters: epoch 3 = the 4th epoch, iter 9 = the 10th
        epoch = 3
        iter = 9

    assert Hook.every_n_epochs(R, 2)
    assert not Hook.every_n_epochs(R, 3)
    assert Hook.every_n_iters(R, 5)
    assert not Hook.every_n_iters(R, 4)
    assert not Hook.every_n_iters(R, 0)  # disabled cadence never fires


def test_chrome_trace_export(tmp_path):
    import json
    import time

    dt = DistributedTimer()
    for _ in range(2):
        dt.add_timestamp("iter"); time.sleep(0.002); dt.add_timestamp("iter")
    path = str(tmp_path / "trace.json")
    dt.export_chrome_trace(path, rank=3)
    trace = json.loads(open(path).read())
    ev = trace["traceEvents"]
    assert len(ev) == 2
    assert all(e["ph"] == "X" and e["tid"] == 3 and e["dur"] > 0 for e in ev)


def test_metrics_hook_writes_jsonl(tmp_path):
    import json

    from skycomputing_amd.builder import build_hook
    from skycomputing_amd.runner import MetricsHook

    path = str(tmp_path / "metrics.jsonl")
    hook = build_hook(dict(layer_type="MetricsHook", path=path))
    assert isinstance(hook, MetricsHook)

    class R:  # minimal runner surface
        comm = None
        epoch = 0
        iter = 0
        last_loss = 1.25
        iter_times = [0.01]

    r = R()
    hook.before_run(r)
    hook.after_train_iter(r)
    r.iter = 1
    r.last_loss = 1.1
    hook.after_train_iter(r)
    hook.after_run(r)
    recs = [json.loads(l) for l in open(path)]
    assert len(recs) == 2
    assert recs[0]["loss"] == 1.25 and recs[1]["iter"] == 1
    assert recs[0]["iter_time_s"] == 0.01


def test_lr_schedule_hook_math():
    from skycomputing_amd.builder import build_hook
    from skycomputing_amd.runner import LRScheduleHook

    h = build_hook(dict(layer_type="LRScheduleHook", base_lr=1.0,
                        warmup_iters=10, total_iters=110, decay="linear",
                        min_lr=0.1))
    assert isinstance(h, LRScheduleHook)
    assert h.lr_at(0, 110) == pytest.approx(0.1)   # warmup ramp start
    assert h.lr_at(9, 110) == pytest.approx(1.0)
    assert h.lr_at(10, 110) == pytest.approx(1.0)
    mid = h.lr_at(60, 110)
    assert 0.1 < mid < 1.0
    assert h.lr_at(109, 110) == pytest.approx(0.1, abs=0.02)
    hc = LRScheduleHook(base_lr=2.0, warmup_iters=0, total_iters=100, decay="cosine")
    assert hc.lr_at(0, 100) == pytest.approx(2.0)
    assert hc.lr_at(99, 100) == pytest.approx(0.0, abs=0.002)

    class Opt:
        lr = 0.0

    class R:
        iter = 5
        max_iter = 110
        optimizer = Opt()

    h.before_train_iter(R)
    assert R.optimizer.lr == pytest.approx(0.6)


def test_checkpoint_meta_roundtrip(tmp_path):
    """Checkpoint meta carries epoch/iter for counter resume."""
    ps = ParameterServer(1)
    ps.update_weights({"w": torch.randn(2)}, 0)
    path = str(tmp_path / "epoch_3.pth")
    ps.save_weights_to_file(path, meta={"epoch": 3, "iter": 90})
    ps2 = ParameterServer(1)
    meta = ps2.load_weights_from_file(path)
    assert meta["epoch"] == 3 and meta["iter"] == 90


def test_glue_dataset_feature_cache(tmp_path):
    vocab = ["[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]", "the", "cat"]
    (tmp_path / "vocab.txt").write_text("\n".join(vocab) + "\n")
    (tmp_path / "train.tsv").write_text("sentence\tlabel\nthe cat\t1\n")
    try:
        ds1 = GlueDataset(str(tmp_path), task="sst-2", max_seq_length=8)
    except ImportError:
        pytest.skip("transformers unavailable")
    cache = tmp_path / "cached_sst-2_train_8.pt"
    assert cache.is_file()
    # second load comes from the cache even without the tokenizer inputs
    (tmp_path / "train.tsv").unlink()
    ds2 = GlueDataset(str(tmp_path), task="sst-2", max_seq_length=8)
    assert torch.equal(ds1.input_ids, ds2.input_ids)
    assert torch.equal(ds1.labels, ds2.labels)
