"""CPU-side tests for GraphedTrainStep: the captured body must compute
exactly one eager optimizer step, shape signatures must gate replay, and
capture must fall back gracefully where hipGraphs are unavailable.
(The GPU equivalence tests live in tests/test_gpu_e2e.py.)"""
import copy

import pytest
import torch

from chinesener_amd.train.graph_step import GraphedTrainStep


class TinyModel(torch.nn.Module):
    def __init__(self):
        super().__init__()
        self.lin = torch.nn.Linear(4, 3)

    def forward(self, batch):
        x = batch["x"]
        loss = (self.lin(x) ** 2).mean()
        out = type("O", (), {})()
        out.loss = loss
        return out


class ConstSchedule:
    def apply(self, opt, step):
        for g in opt.param_groups:
            g["lr"] = 1e-2
        return 1e-2


def clip(model):
    torch.nn.utils.clip_grad_norm_(model.parameters(), 1.0)


def make_pair():
    torch.manual_seed(0)
    m1 = TinyModel()
    m2 = copy.deepcopy(m1)
    o1 = torch.optim.SGD(m1.parameters(), lr=1e-2)
    o2 = torch.optim.SGD(m2.parameters(), lr=1e-2)
    return m1, o1, m2, o2


def test_body_is_one_eager_step():
    m1, o1, m2, o2 = make_pair()
    g = GraphedTrainStep(m1, o1, ConstSchedule(), clip)
    batch = {"x": torch.randn(8, 4)}
    g.static = batch
    loss_body = g._body()

    # manual eager step on the clone
    o2.zero_grad()
    out = m2(batch)
    out.loss.backward()
    clip(m2)
    o2.step()
    assert torch.allclose(loss_body, out.loss)
    for a, b in zip(m1.parameters(), m2.parameters()):
        assert torch.equal(a, b)


def test_signature_gates_replay():
    m1, o1, _, _ = make_pair()
    g = GraphedTrainStep(m1, o1, ConstSchedule(), clip)
    b1 = {"x": torch.randn(8, 4)}
    b2 = {"x": torch.randn(6, 4)}          # different batch size
    b3 = {"x": torch.randn(8, 4).double()}  # different dtype
    g.sig = g._signature(b1)
    g.graph = object()  # pretend captured
    assert g.matches({"x": torch.randn(8, 4)})
    assert not g.matches(b2)
    assert not g.matches(b3)


@pytest.mark.skipif(torch.cuda.is_available(), reason="CPU-only fallback path")
def test_capture_falls_back_on_cpu():
    m1, o1, _, _ = make_pair()
    g = GraphedTrainStep(m1, o1, ConstSchedule(), clip)
    ok = g.try_capture({"x": torch.randn(8, 4)})
    assert ok is False and g.failed and g.graph is None
    # second attempt short-circuits
    assert g.try_capture({"x": torch.randn(8, 4)}) is False
