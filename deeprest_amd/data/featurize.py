"""Call-path feature engineering.

DeepRest's feature space is the set of distinct *root-to-node call paths*
observed across all traces: a path is the sequence of ``component_operation``
identifiers from a trace root down to a node, and the per-window feature
vector counts how many times each path occurred in that window
(reference: resource-estimation/featurize.py:11-40).

This implementation keeps those semantics exactly (path identity, discovery
order of indices, per-window counting) but is a new design:

- paths are tuples of (component, operation) pairs, not stringified lists,
  so component/operation names may contain any characters (the reference
  breaks on '_' in names at featurize.py:92);
- traversal is iterative (stack-based), so 10^5-node traces from the
  4096-endpoint synthetic app do not hit the Python recursion limit;
- a C++ fast path (deeprest_amd._C.featurize_window) is used when the
  native extension is built, with identical output.

The on-disk ``input.pkl`` format ``[traffic, resources, invocations]``
(reference: resource-estimation/featurize.py:105-106) is kept compatible via
``FeaturizedData.to_input_list`` / ``from_input_list``.
"""

from __future__ import annotations

import pickle
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Sequence, Tuple

import numpy as np

Path = Tuple[Tuple[str, str], ...]  # ((component, operation), ...)


class FeatureSpace:
    """Discovery-ordered mapping from call paths to feature indices."""

    def __init__(self) -> None:
        self._index: Dict[Path, int] = {}
        self._paths: List[Path] = []

    def __len__(self) -> int:
        return len(self._index)

    def __contains__(self, path: Path) -> bool:
        return path in self._index

    @property
    def paths(self) -> List[Path]:
        return list(self._paths)

    def index_of(self, path: Path) -> int:
        return self._index[path]

    def observe_trace(self, trace: Dict[str, Any]) -> None:
        """Register every root-to-node path of one span tree (pre-order)."""
        # stack of (node, prefix) — prefix is the path to the node's parent
        stack: List[Tuple[Dict[str, Any], Path]] = [(trace, ())]
        while stack:
            node, prefix = stack.pop()
            path = prefix + ((node["component"], node["operation"]),)
            if path not in self._index:
                self._index[path] = len(self._index)
                self._paths.append(path)
            children = node.get("children", [])
            # reversed => children processed left-to-right (matches reference
            # discovery order, featurize.py:16-17)
            for child in reversed(children):
                stack.append((child, path))

    def count_trace(self, trace: Dict[str, Any], out: np.ndarray) -> None:
        """Accumulate path counts for one span tree into ``out`` (len == len(self))."""
        stack: List[Tuple[Dict[str, Any], Path]] = [(trace, ())]
        while stack:
            node, prefix = stack.pop()
            path = prefix + ((node["component"], node["operation"]),)
            idx = self._index.get(path)
            if idx is not None:
                out[idx] += 1
            for child in reversed(node.get("children", [])):
                stack.append((child, path))

    def components(self) -> List[str]:
        """All components appearing anywhere in the space, discovery order."""
        seen: Dict[str, None] = {}
        for path in self._paths:
            for component, _op in path:
                if component not in seen:
                    seen[component] = None
        return list(seen.keys())

    def api_endpoints(self) -> List[str]:
        """Root-level ``component_operation`` identifiers (the API surface)."""
        seen: Dict[str, None] = {}
        for path in self._paths:
            if len(path) == 1:
                component, op = path[0]
                ident = f"{component}_{op}"
                if ident not in seen:
                    seen[ident] = None
        return list(seen.keys())

    # ---- (de)serialization (part of the checkpoint format) ----
    def state_dict(self) -> Dict[str, Any]:
        return {"paths": self._paths}

    @staticmethod
    def from_state_dict(state: Dict[str, Any]) -> "FeatureSpace":
        fs = FeatureSpace()
        for path in state["paths"]:
            path = tuple(tuple(p) for p in path)
            fs._index[path] = len(fs._index)
            fs._paths.append(path)
        return fs


@dataclass
class FeaturizedData:
    """Output of featurization.

    traffic:     (T, P) int64 — per-window call-path counts
    resources:   {component_resource: (T,) float64} — target series,
                 first-seen order preserved
    invocations: {component: (T,) int64} plus 'general' (= #traces/window),
                 for the component-aware baseline
    resource_components: {identifier: component} so downstream code never
                 parses identifiers (the reference's '_'-split pitfall)
    """

    traffic: np.ndarray
    resources: Dict[str, np.ndarray]
    invocations: Dict[str, np.ndarray]
    feature_space: Optional[FeatureSpace] = None
    resource_components: Dict[str, str] = field(default_factory=dict)

    @property
    def num_windows(self) -> int:
        return int(self.traffic.shape[0])

    @property
    def num_paths(self) -> int:
        return int(self.traffic.shape[1])

    @property
    def metric_names(self) -> List[str]:
        return list(self.resources.keys())

    def to_input_list(self) -> list:
        """The reference's input.pkl payload [traffic, resources, invocations]."""
        return [self.traffic, self.resources, self.invocations]

    @staticmethod
    def from_input_list(payload: Sequence[Any]) -> "FeaturizedData":
        traffic, resources, invocations = payload
        return FeaturizedData(
            traffic=np.asarray(traffic),
            resources={k: np.asarray(v) for k, v in resources.items()},
            invocations={k: np.asarray(v) for k, v in invocations.items()},
        )

    def save(self, path: str) -> None:
        with open(path, "wb") as f:
            pickle.dump(self.to_input_list(), f)

    @staticmethod
    def load(path: str) -> "FeaturizedData":
        with open(path, "rb") as f:
            return FeaturizedData.from_input_list(pickle.load(f))


def _count_invocations_window(traces: List[Dict[str, Any]]) -> Dict[str, int]:
    """Per-component span counts for one window; 'general' = #root traces."""
    counts: Dict[str, int] = {"general": 0}
    for trace in traces:
        counts["general"] += 1
        stack = [trace]
        while stack:
            node = stack.pop()
            comp = node["component"]
            counts[comp] = counts.get(comp, 0) + 1
            stack.extend(node.get("children", []))
    return counts


def _native_featurize():
    """Return the C++ trie featurizer entry point, or None if not built."""
    try:
        from deeprest_amd.ops.native import load_native

        ext = load_native()
        if ext is not None and hasattr(ext, "featurize_walk"):
            return ext.featurize_walk
    except Exception:
        pass
    return None


class Featurizer:
    """fit/transform over the raw-data contract."""

    def __init__(self, feature_space: Optional[FeatureSpace] = None, use_native: bool = True) -> None:
        self.feature_space = feature_space or FeatureSpace()
        self.use_native = use_native

    def fit(self, raw_data: Sequence[Dict[str, Any]]) -> "Featurizer":
        for window in raw_data:
            for trace in window["traces"]:
                self.feature_space.observe_trace(trace)
        return self

    def _walk(self, raw_data: Sequence[Dict[str, Any]], grow: bool):
        """Native trie walk: (traffic, invocations); grows the space if asked."""
        native = _native_featurize()
        fs = self.feature_space
        new_paths, traffic, inv_names, inv_mat = native(
            list(raw_data), fs.paths, grow
        )
        for path in new_paths:
            path = tuple(tuple(p) for p in path)
            if path not in fs:
                fs._index[path] = len(fs._index)
                fs._paths.append(path)
        traffic = traffic.numpy()
        inv_np = inv_mat.numpy()
        invocations = {
            name: inv_np[i].copy() for i, name in enumerate(inv_names)
        }
        # components known to the space but absent from this data slice
        for comp in fs.components():
            invocations.setdefault(comp, np.zeros(len(raw_data), dtype=np.int64))
        return traffic, invocations

    def transform(self, raw_data: Sequence[Dict[str, Any]]) -> FeaturizedData:
        fs = self.feature_space
        T = len(raw_data)
        P = len(fs)

        if self.use_native and _native_featurize() is not None:
            traffic, invocations = self._walk(raw_data, grow=False)
        else:
            traffic = np.zeros((T, P), dtype=np.int64)
            for t, window in enumerate(raw_data):
                row = traffic[t]
                for trace in window["traces"]:
                    fs.count_trace(trace, row)

            # invocation counts for the component-aware baseline — include
            # components seen in the DATA even when absent from a frozen
            # feature space (matches the native walk's semantics)
            components = fs.components()
            invocations = {c: np.zeros(T, dtype=np.int64) for c in components}
            invocations["general"] = np.zeros(T, dtype=np.int64)
            for t, window in enumerate(raw_data):
                counts = _count_invocations_window(window["traces"])
                for comp, n in counts.items():
                    if comp not in invocations:
                        invocations[comp] = np.zeros(T, dtype=np.int64)
                    invocations[comp][t] = n
        return self._assemble(raw_data, traffic, invocations)

    def _assemble(self, raw_data, traffic, invocations) -> FeaturizedData:
        T = len(raw_data)
        # target series, first-seen order (reference featurize.py:68-75)
        resources: Dict[str, List[float]] = {}
        resource_components: Dict[str, str] = {}
        for window in raw_data:
            for metric in window["metrics"]:
                ident = f"{metric['component']}_{metric['resource']}"
                if ident not in resources:
                    resources[ident] = []
                    resource_components[ident] = metric["component"]
                resources[ident].append(float(metric["value"]))
        resources_np = {k: np.asarray(v, dtype=np.float64) for k, v in resources.items()}
        for name, series in resources_np.items():
            if series.shape[0] != T:
                raise ValueError(
                    f"resource series '{name}' has {series.shape[0]} samples for {T} windows; "
                    "every window must report every metric"
                )

        return FeaturizedData(
            traffic=traffic,
            resources=resources_np,
            invocations=invocations,
            feature_space=self.feature_space,
            resource_components=resource_components,
        )

    def fit_transform(self, raw_data: Sequence[Dict[str, Any]]) -> FeaturizedData:
        if self.use_native and _native_featurize() is not None:
            # one native pass: grow the trie AND count (discovery order kept)
            traffic, invocations = self._walk(raw_data, grow=True)
            return self._assemble(raw_data, traffic, invocations)
        return self.fit(raw_data).transform(raw_data)
