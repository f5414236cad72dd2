"""engine/step.py TrainStep unit tests (the shared Trainer/bench step)."""

import torch

from deeprest_amd.engine.step import TrainStep


class _Tiny(torch.nn.Module):
    def __init__(self):
        super().__init__()
        self.lin = torch.nn.Linear(4, 2)

    def forward(self, x):
        return self.lin(x)

    def loss(self, out, y):
        return torch.nn.functional.mse_loss(out.float(), y)


def test_trainstep_runs_and_updates():
    torch.manual_seed(0)
    m = _Tiny()
    opt = torch.optim.SGD(m.parameters(), lr=0.1)
    step = TrainStep(m, opt, autocast_dtype=None)
    x, y = torch.randn(8, 4), torch.randn(8, 2)
    before = [p.detach().clone() for p in m.parameters()]
    losses = [float(step(x, y)) for _ in range(3)]
    assert losses[2] < losses[0]
    assert any(not torch.equal(a, b)
               for a, b in zip(before, m.parameters()))
    assert not step.graphed


def test_trainstep_dist_hook_called():
    calls = []

    class FakeDist:
        def all_reduce_gradients(self, model):
            calls.append(sum(p.grad.abs().sum().item()
                             for p in model.parameters()))

    m = _Tiny()
    opt = torch.optim.SGD(m.parameters(), lr=0.1)
    step = TrainStep(m, opt, dist_ctx=FakeDist(), autocast_dtype=None)
    step(torch.randn(4, 4), torch.randn(4, 2))
    step(torch.randn(4, 4), torch.randn(4, 2))
    assert len(calls) == 2 and all(c > 0 for c in calls)


def test_trainstep_custom_loss_fn():
    m = _Tiny()
    opt = torch.optim.SGD(m.parameters(), lr=0.1)
    seen = []
    step = TrainStep(m, opt, autocast_dtype=None,
                     loss_fn=lambda out, y: seen.append(1) or
                     (out - y).abs().mean())
    step(torch.randn(4, 4), torch.randn(4, 2))
    assert seen == [1]
