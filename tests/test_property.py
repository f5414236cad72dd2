"""Property-based tests (hypothesis) for the data layer.

Randomized span trees / series exercise invariants the example-based suites
can't sweep: arbitrary trace shapes and depths for the call-path feature
space, arbitrary series for windowing and normalization.
"""

import numpy as np
from hypothesis import given, settings, strategies as st

from deeprest_amd.data.featurize import FeatureSpace
from deeprest_amd.data.windows import MinMaxScaler, sliding_window
from deeprest_amd.ops.pinball import reference_pinball_loss

import torch


# ---- random span-tree strategy (bounded depth/fanout) ----
def _span(depth):
    base = st.fixed_dictionaries({
        "component": st.sampled_from(["svc-a", "svc-b", "svc-c", "db"]),
        "operation": st.sampled_from(["read", "write", "compose"]),
    })
    if depth == 0:
        return base.map(lambda d: {**d, "children": []})
    child = _span(depth - 1)
    return st.tuples(base, st.lists(child, max_size=3)).map(
        lambda t: {**t[0], "children": list(t[1])}
    )


def _all_prefix_paths(trace, prefix=()):
    """Brute-force enumeration of every root-to-node call path."""
    p = prefix + ((trace["component"], trace["operation"]),)
    out = [p]
    for ch in trace.get("children", []):
        out.extend(_all_prefix_paths(ch, p))
    return out


@settings(max_examples=40, deadline=None)
@given(st.lists(_span(3), min_size=1, max_size=6))
def test_feature_space_indexes_every_prefix_path(traces):
    fs = FeatureSpace()
    for tr in traces:
        fs.observe_trace(tr)
    want = set()
    for tr in traces:
        want.update(_all_prefix_paths(tr))
    assert len(fs) == len(want)

    # counting: per trace, vector total == number of spans; per path, count ==
    # number of occurrences of that exact prefix path
    for tr in traces:
        v = np.zeros(len(fs), dtype=np.int64)
        fs.count_trace(tr, v)
        paths = _all_prefix_paths(tr)
        assert v.sum() == len(paths)
        for p in set(paths):
            assert v[fs.index_of(p)] == paths.count(p)


@settings(max_examples=30, deadline=None)
@given(st.lists(_span(2), min_size=1, max_size=4), st.lists(_span(2), min_size=1, max_size=4))
def test_frozen_feature_space_never_grows(seen, unseen):
    fs = FeatureSpace()
    for tr in seen:
        fs.observe_trace(tr)
    n = len(fs)
    # counting unseen traces against the frozen space must not grow it and
    # must only count paths that already exist (reference featurize.py:27-33
    # drops unseen paths at extract time)
    for tr in unseen:
        v = np.zeros(n, dtype=np.int64)
        fs.count_trace(tr, v)
        seen_paths = {p for t in seen for p in _all_prefix_paths(t)}
        hits = [p for p in _all_prefix_paths(tr) if p in seen_paths]
        assert v.sum() == len(hits)
    assert len(fs) == n


@settings(max_examples=25, deadline=None)
@given(st.lists(_span(3), min_size=1, max_size=5),
       st.lists(_span(3), min_size=1, max_size=5))
def test_native_featurizer_property_parity(train_traces, new_traces):
    """The C++ trie featurizer must agree with the Python FeatureSpace on
    arbitrary span trees — both while growing the space and frozen."""
    from deeprest_amd.data.featurize import Featurizer, _native_featurize

    if _native_featurize() is None:
        return  # extension not built in this environment

    def windows(traces):
        return [{"metrics": [{"component": "svc-a", "resource": "cpu",
                              "value": 1.0}],
                 "traces": traces}]

    train = windows(train_traces)
    fresh = windows(new_traces)
    py = Featurizer(use_native=False).fit(train)
    nat = Featurizer(use_native=True).fit(train)
    assert py.feature_space.paths == nat.feature_space.paths
    dp = py.transform(train + fresh)   # frozen space: unseen paths dropped
    dn = nat.transform(train + fresh)
    np.testing.assert_array_equal(dp.traffic, dn.traffic)
    assert set(dp.invocations.keys()) == set(dn.invocations.keys())
    for k in dp.invocations:
        np.testing.assert_array_equal(dp.invocations[k], dn.invocations[k])


@settings(max_examples=30, deadline=None)
@given(
    st.integers(min_value=2, max_value=40).flatmap(
        lambda n: st.tuples(
            st.just(n),
            st.integers(min_value=1, max_value=n - 1),
            st.lists(st.floats(-1e3, 1e3, allow_nan=False), min_size=n, max_size=n),
        )
    )
)
def test_sliding_window_matches_naive(args):
    n, w, vals = args
    ts = np.asarray(vals, dtype=np.float64)
    got = sliding_window(ts, w)
    assert got.shape == (n - w, w)
    for i in range(n - w):
        np.testing.assert_array_equal(got[i], ts[i : i + w])


@settings(max_examples=30, deadline=None)
@given(st.lists(st.floats(0, 1e6, allow_nan=False), min_size=4, max_size=64))
def test_minmax_roundtrip(vals):
    M = np.asarray(vals, dtype=np.float64).reshape(-1, 1)
    sc = MinMaxScaler().fit(M, split=max(2, len(vals) // 2))
    z = sc.transform(M)
    back = sc.inverse_transform(z)
    np.testing.assert_allclose(back, M, rtol=1e-9, atol=1e-6)


@settings(max_examples=20, deadline=None)
@given(st.lists(_span(2), min_size=2, max_size=6), st.integers(0, 2**31 - 1),
       st.integers(1, 30))
def test_synthesizer_root_counts_and_conservation(traces, seed, count):
    """Synthesized vectors are sums of observed whole-trace vectors: the
    root-path count equals the requested calls, every count is achievable."""
    from deeprest_amd.data.synthesizer import TraceSynthesizer

    raw = [{"metrics": [{"component": "svc-a", "resource": "cpu", "value": 1.0}],
            "traces": traces}]
    syn = TraceSynthesizer().fit(raw)
    fs = syn.feature_space
    rng = np.random.default_rng(seed)
    api = syn.apis[0]
    vec = syn.synthesize({api: count}, rng=rng)
    comp, op = api.split("_", 1)
    root_idx = fs.index_of(((comp, op),))
    # every sampled trace contributes exactly one root-path occurrence
    assert vec[root_idx] == count
    # total span count is between count * min-spans and count * max-spans
    vecs, _ = syn.api2dist[api]
    per_trace = vecs.sum(axis=1)
    assert count * per_trace.min() <= vec.sum() <= count * per_trace.max()


@settings(max_examples=25, deadline=None)
@given(st.integers(0, 2**31 - 1))
def test_pinball_zero_at_exact_prediction_and_nonnegative(seed):
    rng = np.random.default_rng(seed)
    y = torch.from_numpy(rng.normal(size=(2, 5, 3)).astype(np.float32))
    exact = y.unsqueeze(-1).repeat(1, 1, 1, 3)
    assert float(reference_pinball_loss(exact, y, (0.05, 0.5, 0.95))) == 0.0
    off = exact + torch.from_numpy(rng.normal(size=exact.shape).astype(np.float32))
    assert float(reference_pinball_loss(off, y, (0.05, 0.5, 0.95))) >= 0.0


@settings(max_examples=25, deadline=None)
@given(st.lists(st.floats(-1e6, 1e6, allow_nan=False), min_size=4, max_size=40))
def test_minmax_state_roundtrip_and_zero_range(vals):
    M = np.asarray(vals, dtype=np.float64).reshape(-1, 1)
    sc = MinMaxScaler().fit(M, split=max(2, len(vals) // 2))
    sc2 = MinMaxScaler.from_state_dict(sc.state_dict())
    np.testing.assert_array_equal(sc.transform(M), sc2.transform(M))
    # degenerate constant series: transform/inverse must be total no-ops
    C = np.full((6, 1), 3.25)
    scc = MinMaxScaler().fit(C, split=3)
    np.testing.assert_array_equal(scc.transform(C), C)
    np.testing.assert_array_equal(scc.inverse_transform(C), C)
