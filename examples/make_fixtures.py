#!/usr/bin/env python3
"""Generate the toy demo fixtures (the reference ships raw_data.pkl and
input.pkl with 3 windows; ours come from the synthetic app so they are
reproducible): examples/fixtures/raw_data.pkl + input.pkl."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from deeprest_amd.data.contract import save_raw_data
from deeprest_amd.data.featurize import Featurizer
from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig

out = os.path.join(os.path.dirname(__file__), "fixtures")
os.makedirs(out, exist_ok=True)
app = SyntheticApp(SyntheticAppConfig(
    n_apis=2, n_components=5, windows_per_day=3, n_days=1,
    shapes_per_api=2, seed=1))
raw = app.generate_raw()
save_raw_data(raw, os.path.join(out, "raw_data.pkl"))
data = Featurizer().fit_transform(raw)
data.save(os.path.join(out, "input.pkl"))
print(f"wrote {len(raw)} windows, {data.num_paths} call paths, "
      f"{len(data.metric_names)} metrics -> {out}")
