import pickle

import pytest

from deeprest_amd.data.contract import (
    ContractError,
    Span,
    Window,
    load_raw_data,
    save_raw_data,
    validate_raw_data,
    windows_from_raw,
)


def test_validate_ok(tiny_raw_data):
    validate_raw_data(tiny_raw_data)


def test_validate_rejects_bad_shapes(tiny_raw_data):
    with pytest.raises(ContractError):
        validate_raw_data({"not": "a list"})
    bad = [dict(tiny_raw_data[0])]
    del bad[0]["traces"]
    with pytest.raises(ContractError):
        validate_raw_data(bad)
    bad = [{"metrics": [{"component": "a"}], "traces": []}]
    with pytest.raises(ContractError):
        validate_raw_data(bad)
    bad = [{"metrics": [], "traces": [{"component": "a"}]}]  # span missing operation
    with pytest.raises(ContractError):
        validate_raw_data(bad)


def test_roundtrip_pickle(tiny_raw_data, tmp_path):
    p = str(tmp_path / "raw.pkl")
    save_raw_data(tiny_raw_data, p)
    loaded = load_raw_data(p)
    assert loaded == tiny_raw_data
    # plain pickled list of dicts — the reference on-disk format
    with open(p, "rb") as f:
        assert isinstance(pickle.load(f), list)


def test_typed_view_roundtrip(tiny_raw_data):
    windows = windows_from_raw(tiny_raw_data)
    assert isinstance(windows[0], Window)
    assert windows[0].traces[0].component == "frontend"
    assert [w.to_dict() for w in windows] == tiny_raw_data


def test_span_walk_preorder():
    s = Span.from_dict(
        {"component": "a", "operation": "1", "children": [
            {"component": "b", "operation": "2", "children": [
                {"component": "c", "operation": "3", "children": []}]},
            {"component": "d", "operation": "4", "children": []},
        ]}
    )
    order = [n.component for n in s.walk()]
    assert order == ["a", "b", "c", "d"]


def test_validate_deep_span_chain_no_recursion_error():
    """REST /ingest accepts arbitrary payloads: a 10k-deep span chain must
    validate (or ContractError), never RecursionError."""
    deep = {"component": "c", "operation": "o", "children": []}
    node = deep
    for _ in range(10_000):
        child = {"component": "c", "operation": "o", "children": []}
        node["children"].append(child)
        node = child
    validate_raw_data([{"metrics": [], "traces": [deep]}])
    # typed view + featurizer also survive the same chain
    from deeprest_amd.data.contract import Span
    from deeprest_amd.data.featurize import Featurizer

    s = Span.from_dict(deep)
    assert sum(1 for _ in s.walk()) == 10_001
    data = Featurizer(use_native=False).fit_transform(
        [{"metrics": [{"component": "c", "resource": "cpu", "value": 1.0}],
          "traces": [deep]}])
    assert data.num_paths == 10_001


def test_shipped_fixtures_match_generator():
    """examples/fixtures must stay in sync with the synthetic generator
    (regenerate with examples/make_fixtures.py after generator changes)."""
    import os

    import numpy as np

    from deeprest_amd.data.contract import load_raw_data
    from deeprest_amd.data.featurize import FeaturizedData, Featurizer
    from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig

    root = os.path.join(os.path.dirname(__file__), "..", "examples", "fixtures")
    raw = load_raw_data(os.path.join(root, "raw_data.pkl"))
    data = FeaturizedData.load(os.path.join(root, "input.pkl"))

    app = SyntheticApp(SyntheticAppConfig(
        n_apis=2, n_components=5, windows_per_day=3, n_days=1,
        shapes_per_api=2, seed=1))
    fresh_raw = app.generate_raw()
    assert raw == fresh_raw, "fixtures stale: rerun examples/make_fixtures.py"
    fresh = Featurizer().fit_transform(fresh_raw)
    assert np.array_equal(fresh.traffic, data.traffic)
