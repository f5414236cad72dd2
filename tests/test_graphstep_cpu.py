"""CPU-side behavior of the hipGraph train-step utilities.

The capture itself needs a GPU (tests/test_ops_gpu.py::test_graphed_train_step);
here we pin down the guards and the capturable-optimizer bookkeeping that must
work identically everywhere.
"""

import os

import torch

from deeprest_amd.engine.graphstep import GraphedTrainStep
from deeprest_amd.ops.adam import FusedAdam
from deeprest_amd.ops.pinball import _q_tensor


def test_graphstep_build_returns_none_without_gpu():
    model = torch.nn.Linear(4, 2)
    opt = FusedAdam(model.parameters())
    x = torch.randn(3, 4)
    y = torch.randn(3, 2)
    g = GraphedTrainStep.build(model, opt, torch.nn.functional.mse_loss, x, y)
    assert g is None  # no CUDA here -> caller stays eager


def test_graphstep_env_optout(monkeypatch):
    monkeypatch.setenv("DEEPREST_NO_GRAPH", "1")
    model = torch.nn.Linear(4, 2)
    opt = FusedAdam(model.parameters())
    g = GraphedTrainStep.build(model, opt, torch.nn.functional.mse_loss,
                               torch.randn(2, 4), torch.randn(2, 2))
    assert g is None


def test_capturable_adam_single_group_only():
    m1 = torch.nn.Linear(4, 2)
    m2 = torch.nn.Linear(4, 2)
    groups = [{"params": list(m1.parameters())}, {"params": list(m2.parameters())}]
    try:
        FusedAdam(groups, capturable=True)
        raise AssertionError("expected ValueError for multi-group capturable")
    except ValueError:
        pass


def test_capturable_adam_cpu_falls_back_to_eager_math():
    """capturable=True must not change CPU results (device table is GPU-only)."""
    torch.manual_seed(0)
    def make():
        torch.manual_seed(5)
        return torch.nn.Linear(6, 3)

    results = []
    for capturable in (False, True):
        model = make()
        opt = FusedAdam(model.parameters(), lr=1e-2, capturable=capturable)
        x = torch.randn(8, 6)
        y = torch.randn(8, 3)
        for _ in range(4):
            loss = torch.nn.functional.mse_loss(model(x), y)
            opt.zero_grad()
            loss.backward()
            opt.step()
        results.append(torch.cat([p.detach().reshape(-1) for p in model.parameters()]))
    assert torch.equal(results[0], results[1])


def test_quantile_tensor_cache_identity():
    q1 = _q_tensor((0.05, 0.5, 0.95), torch.device("cpu"))
    q2 = _q_tensor((0.05, 0.5, 0.95), torch.device("cpu"))
    assert q1 is q2  # cached: no rebuild per call
    q3 = _q_tensor((0.1, 0.9), torch.device("cpu"))
    assert q3.numel() == 2 and q3 is not q1
