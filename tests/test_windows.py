import numpy as np

from deeprest_amd.data.windows import MinMaxScaler, minmax_apply, minmax_fit, sliding_window


def test_sliding_window_1d():
    ts = np.arange(10)
    w = sliding_window(ts, 3)
    assert w.shape == (7, 3)
    assert w[0].tolist() == [0, 1, 2]
    assert w[-1].tolist() == [6, 7, 8]  # last start = len-ws, exclusive


def test_sliding_window_2d():
    ts = np.arange(20).reshape(10, 2)
    w = sliding_window(ts, 4)
    assert w.shape == (6, 4, 2)
    np.testing.assert_array_equal(w[2], ts[2:6])


def test_sliding_window_too_short():
    assert sliding_window(np.arange(3), 5).shape == (0, 5)


def test_minmax_train_split_only():
    M = np.array([0.0, 10.0, 100.0, 1000.0])
    lo, hi = minmax_fit(M, split=2)
    assert (lo, hi) == (0.0, 10.0)
    out = minmax_apply(M, lo, hi)
    np.testing.assert_allclose(out, [0.0, 1.0, 10.0, 100.0])


def test_minmax_constant_series():
    M = np.full(5, 7.0)
    lo, hi = minmax_fit(M, 3)
    out = minmax_apply(M, lo, hi)
    np.testing.assert_allclose(out, M)  # unchanged when range is zero


def test_scaler_roundtrip():
    M = np.linspace(-5, 5, 50)
    sc = MinMaxScaler().fit(M, split=30)
    normalized = sc.transform(M)
    restored = sc.inverse_transform(normalized)
    np.testing.assert_allclose(restored, M, atol=1e-12)
    sc2 = MinMaxScaler.from_state_dict(sc.state_dict())
    assert sc2.min_val == sc.min_val and sc2.max_val == sc.max_val
