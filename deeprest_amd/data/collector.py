"""Collection-plane adapter: Jaeger-style traces + Prometheus-style samples
-> the raw-data contract.

The reference's collection plane is Jaeger (spans with trace/span ids and
parent references) and Prometheus (per-component resource samples on a fixed
scrape interval) — SURVEY.md L3/L4.  This module keeps that collector-side
API: feed it the JSON shapes those tools export and it produces contract
windows, discretized on the scrape interval
(reference: resource-estimation/README.md:29 — "window size can be defined
as the scrape interval").

Jaeger span shape accepted (the /api/traces export format, reduced to the
fields DeepRest uses):
    {"traceID": ..., "spanID": ..., "operationName": ...,
     "startTime": microseconds, "processID"/"serviceName": component,
     "references": [{"refType": "CHILD_OF", "spanID": parent}]}

Prometheus sample shape:
    {"component": ..., "resource": ..., "value": float, "timestamp": seconds}
"""

from __future__ import annotations

from collections import defaultdict
from typing import Any, Dict, Iterable, List, Optional, Tuple


def span_tree_from_jaeger(trace: Dict[str, Any]) -> Optional[Dict[str, Any]]:
    """One Jaeger trace (dict with 'spans' and optional 'processes') ->
    contract span tree rooted at the parentless span."""
    spans = trace.get("spans", [])
    if not spans:
        return None
    processes = trace.get("processes", {})

    def component_of(span: Dict[str, Any]) -> str:
        if "serviceName" in span:
            return span["serviceName"]
        pid = span.get("processID")
        if pid is not None and pid in processes:
            return processes[pid].get("serviceName", str(pid))
        return str(pid)

    nodes: Dict[str, Dict[str, Any]] = {}
    parent_of: Dict[str, Optional[str]] = {}
    order: List[str] = []
    for span in spans:
        sid = span["spanID"]
        nodes[sid] = {
            "component": component_of(span),
            "operation": span.get("operationName", ""),
            "children": [],
        }
        parent = None
        for ref in span.get("references", []):
            if ref.get("refType", "CHILD_OF") == "CHILD_OF":
                parent = ref.get("spanID")
        parent_of[sid] = parent
        order.append(sid)

    root = None
    for sid in order:
        p = parent_of[sid]
        if p is not None and p in nodes:
            nodes[p]["children"].append(nodes[sid])
        else:
            if root is None:
                root = nodes[sid]
    return root


def trace_start_time(trace: Dict[str, Any]) -> float:
    """Earliest span start (Jaeger startTime is microseconds) -> seconds."""
    starts = [s.get("startTime", 0) for s in trace.get("spans", [])]
    return (min(starts) / 1e6) if starts else 0.0


class Collector:
    """Accumulates Jaeger traces and Prometheus samples; emits contract
    windows discretized on the scrape interval."""

    def __init__(self, window_sec: float = 5.0, t0: Optional[float] = None) -> None:
        self.window_sec = window_sec
        self.t0 = t0
        self._traces: List[Tuple[float, Dict[str, Any]]] = []
        self._samples: List[Dict[str, Any]] = []

    def add_traces(self, traces: Iterable[Dict[str, Any]]) -> int:
        n = 0
        for trace in traces:
            tree = span_tree_from_jaeger(trace)
            if tree is None:
                continue
            self._traces.append((trace_start_time(trace), tree))
            n += 1
        return n

    def add_metric_samples(self, samples: Iterable[Dict[str, Any]]) -> int:
        n = 0
        for s in samples:
            self._samples.append(dict(s))
            n += 1
        return n

    def windows(self) -> List[Dict[str, Any]]:
        """Discretize everything collected into contract windows."""
        if not self._traces and not self._samples:
            return []
        times = [t for t, _ in self._traces] + [
            float(s.get("timestamp", 0.0)) for s in self._samples
        ]
        t0 = self.t0 if self.t0 is not None else min(times)
        t_end = max(times)
        n_win = int((t_end - t0) / self.window_sec) + 1

        win_traces: Dict[int, List[Dict[str, Any]]] = defaultdict(list)
        for t, tree in self._traces:
            w = int((t - t0) / self.window_sec)
            if 0 <= w < n_win:
                win_traces[w].append(tree)

        # per window, per (component, resource): last sample wins (gauge style)
        win_metrics: Dict[int, Dict[Tuple[str, str], float]] = defaultdict(dict)
        for s in self._samples:
            w = int((float(s.get("timestamp", 0.0)) - t0) / self.window_sec)
            if 0 <= w < n_win:
                win_metrics[w][(s["component"], s["resource"])] = float(s["value"])

        # metric identity must appear in EVERY window (featurizer requirement):
        # carry the last seen value forward, 0.0 before first observation
        all_keys: List[Tuple[str, str]] = []
        seen = set()
        for w in range(n_win):
            for key in win_metrics.get(w, {}):
                if key not in seen:
                    seen.add(key)
                    all_keys.append(key)

        out = []
        last: Dict[Tuple[str, str], float] = {k: 0.0 for k in all_keys}
        for w in range(n_win):
            for key, val in win_metrics.get(w, {}).items():
                last[key] = val
            metrics = [
                {"component": c, "resource": r, "value": last[(c, r)]}
                for (c, r) in all_keys
            ]
            out.append({"metrics": metrics, "traces": win_traces.get(w, [])})
        return out
