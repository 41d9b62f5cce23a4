"""Registry / config / builder / model / estimator unit tests."""

from __future__ import annotations

import torch
import torch.nn as nn

from skycomputing_amd.builder import (
    SequentialWrapper,
    build_layer,
    build_module_from_cfg,
)
from skycomputing_amd.config import Config, load_config
from skycomputing_amd.dynamics import Estimator, ModelBenchmarker
from skycomputing_amd.registry import Registry

from .helpers import tiny_bert_cfg


def test_registry_decorator_and_torch_fallback():
    reg = Registry("t")

    @reg.register_module
    class Foo:
        pass

    assert reg.get_module("Foo") is Foo
    assert reg.get_module("Linear") is nn.Linear
    assert "Foo" in reg and "Linear" in reg and "Nope" not in reg


def test_config_loader_with_base(tmp_path):
    (tmp_path / "base.py").write_text("a = 1\nb = dict(x=2)\n")
    (tmp_path / "child.py").write_text("base = 'base.py'\nb = dict(x=3)\nc = 4\n")
    cfg = load_config(str(tmp_path / "child.py"))
    assert cfg.a == 1 and cfg.b.x == 3 and cfg.c == 4
    assert isinstance(cfg, Config)


def test_sequential_wrapper_tuple_splat():
    class A(nn.Module):
        def forward(self, x):
            return x + 1, x * 2

    class B(nn.Module):
        def forward(self, y, z):
            return y + z

    seq = SequentialWrapper(A(), B())
    out = seq(torch.tensor(1.0))
    assert float(out) == 4.0


def test_build_layer_from_registry():
    layer = build_layer(dict(layer_type="Linear", in_features=4, out_features=2))
    assert isinstance(layer, nn.Linear)


def test_bert_stage_forward_shapes():
    cfgs = tiny_bert_cfg(2)
    stage = build_module_from_cfg(cfgs, record_forward_time=True)
    ids = torch.randint(0, 500, (4, 16))
    out = stage(ids, torch.zeros(4, 16, dtype=torch.long), torch.ones(4, 16, dtype=torch.long))
    assert out.shape == (4, 3)
    assert len(stage.forward_time) == 1
    out.sum().backward()
    assert stage.module[0].word_embeddings.weight.grad is not None


def test_stage_detect_mem_cpu():
    stage = build_module_from_cfg(tiny_bert_cfg(1))
    assert stage.detect_mem() > 0
    stage2 = build_module_from_cfg(tiny_bert_cfg(1), mem_limit=123)
    assert stage2.detect_mem() == 123


def test_stage_state_dict_roundtrip():
    cfgs = tiny_bert_cfg(1)
    torch.manual_seed(0)
    s1 = build_module_from_cfg(cfgs)
    torch.manual_seed(1)
    s2 = build_module_from_cfg(cfgs)
    dicts = s1.get_layer_state_dicts()
    s2.load_layer_state_dicts(dicts)
    for p1, p2 in zip(s1.parameters(), s2.parameters()):
        assert torch.allclose(p1, p2)


def test_model_benchmarker_analytic_costs():
    cfgs = tiny_bert_cfg(3)
    mb = ModelBenchmarker(cfgs, batch_size=4, seq_len=16)
    res = mb.benchmark()
    assert len(res["flops"]) == len(cfgs)
    assert all(f >= 0 for f in res["flops"])
    assert all(m > 0 for m in res["mem"])
    # encoder triplets have identical costs (construction is cached)
    assert res["flops"][1] == res["flops"][4]
    # Head flops dominated by 4 H^2 GEMMs; check magnitude
    layer = build_layer(cfgs[1])
    f = Estimator.layer_flops(layer, 4, 16)
    assert f > 2.0 * 4 * 16 * 64 * 64 * 4 * 0.9


def test_bert_160_layer_config_shape():
    from skycomputing_amd.models import bert_pipeline_config

    cfgs = bert_pipeline_config(160)
    assert len(cfgs) == 160 * 3 + 3
    assert cfgs[0]["layer_type"] == "BertEmbeddings"
    assert cfgs[-1]["layer_type"] == "BertTailForClassification"


def test_slowdown_injection_cpu():
    import time

    cfgs = tiny_bert_cfg(1)
    slow = build_module_from_cfg(cfgs, record_forward_time=True, slowdown=3.0)
    args = (
        torch.randint(0, 500, (2, 8)),
        torch.zeros(2, 8, dtype=torch.long),
        torch.ones(2, 8, dtype=torch.long),
    )
    slow(*args)  # warmup (lazy kernel init noise)
    # self-normalized: the injected sleep is 3x the stage's own measured
    # compute, so wall >= ~comp * (1 + 3) regardless of machine speed; one
    # re-measure tolerates a scheduling hiccup on a loaded machine
    wall = comp = 0.0
    for _attempt in range(2):
        slow.reset_timing()
        t0 = time.perf_counter(); slow(*args); wall = time.perf_counter() - t0
        comp = slow.total_forward_time()
        if comp > 0 and wall > comp * 2.5:
            break
    assert comp > 0
    assert wall > comp * 2.5, (wall, comp)
