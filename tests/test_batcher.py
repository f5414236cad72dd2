"""MicroBatcher: concurrent requests coalesce into one predictor call
with per-request results identical to individual calls."""

import threading

import numpy as np

from deeprest_amd.serve.batcher import MicroBatcher


class _CountingPredictor:
    """Deterministic stub: output encodes the input so splits are checkable."""

    def __init__(self):
        self.calls = 0
        self.lock = threading.Lock()

    def predict(self, windows):
        with self.lock:
            self.calls += 1
        w = np.asarray(windows)
        return {"m0": w.sum(axis=(1, 2), keepdims=True) + np.zeros((len(w), 1, 3))}


def _mk(n, seed):
    return np.random.default_rng(seed).random((n, 4, 5))


def test_concurrent_requests_coalesce():
    pred = _CountingPredictor()
    b = MicroBatcher(pred, max_batch=64, max_wait_ms=25.0)
    results = {}
    errors = []

    def call(i):
        try:
            results[i] = b.predict(_mk(2 + i % 3, seed=i))
        except Exception as e:  # noqa: BLE001
            errors.append(e)

    threads = [threading.Thread(target=call, args=(i,)) for i in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=30)
    assert not errors
    assert len(results) == 8
    # coalescing happened: fewer predictor calls than requests
    assert pred.calls < 8
    assert b.requests_served == 8
    # each caller got ITS windows' results
    for i, out in results.items():
        w = _mk(2 + i % 3, seed=i)
        np.testing.assert_allclose(out["m0"][:, 0, 0], w.sum(axis=(1, 2)),
                                   rtol=1e-12)


def test_max_batch_triggers_immediate_flush():
    pred = _CountingPredictor()
    b = MicroBatcher(pred, max_batch=4, max_wait_ms=10_000.0)  # timer unused
    out = b.predict(_mk(4, seed=1))           # fills the batch alone
    assert pred.calls == 1
    assert out["m0"].shape[0] == 4


def test_predictor_error_propagates_to_waiters():
    class Boom:
        def predict(self, w):
            raise RuntimeError("boom")

    b = MicroBatcher(Boom(), max_batch=1, max_wait_ms=1.0)
    try:
        b.predict(_mk(1, seed=0))
        raise AssertionError("expected RuntimeError")
    except RuntimeError as e:
        assert "boom" in str(e)
