import numpy as np
import pytest

from deeprest_amd.data.featurize import Featurizer
from deeprest_amd.data.synthesizer import TraceSynthesizer


def test_fit_discovers_apis(tiny_raw_data):
    syn = TraceSynthesizer().fit(tiny_raw_data)
    assert set(syn.apis) == {"frontend_/register", "frontend_/compose"}


def test_synthesize_counts_match_shapes(tiny_raw_data):
    fz = Featurizer(use_native=False).fit(tiny_raw_data)
    syn = TraceSynthesizer(feature_space=fz.feature_space).fit(tiny_raw_data)
    rng = np.random.default_rng(0)
    x = syn.synthesize({"frontend_/register": 5}, rng=rng)
    # register's shape is deterministic (only one observed): root+2 children x5
    assert x.tolist() == [5, 5, 5, 0, 0, 0]
    x = syn.synthesize({"frontend_/compose": 2, "frontend_/register": 1}, rng=rng)
    assert x.tolist() == [1, 1, 1, 2, 2, 2]


def test_synthesize_unknown_api_raises(tiny_raw_data):
    syn = TraceSynthesizer().fit(tiny_raw_data)
    with pytest.raises(KeyError):
        syn.synthesize({"nope_/x": 1})


def test_synthesize_mixture_distribution():
    # one api with two shapes at 3:1 observation ratio
    shape_a = {"component": "f", "operation": "/a", "children": []}
    shape_b = {"component": "f", "operation": "/a", "children": [
        {"component": "g", "operation": "op", "children": []}]}
    raw = [{"metrics": [], "traces": [shape_a, shape_a, shape_a, shape_b]}]
    syn = TraceSynthesizer().fit(raw)
    rng = np.random.default_rng(42)
    x = syn.synthesize({"f_/a": 4000}, rng=rng)
    # index 0 = root path (every call), index 1 = child path (~25% of calls)
    assert x[0] == 4000
    assert 800 <= x[1] <= 1200


def test_synthesize_series(tiny_raw_data):
    syn = TraceSynthesizer().fit(tiny_raw_data)
    plan = [{"frontend_/register": 1}, {"frontend_/compose": 2}, {}]
    series = syn.synthesize_series(plan, rng=np.random.default_rng(0))
    assert series.shape == (3, 6)
    assert series[2].sum() == 0
