"""The raw-data ingestion contract.

This is the stable interface DeepRest exposes for bringing your own
application data (reference: resource-estimation/README.md:29-63): a pickled
ordered list of per-time-window dicts

    {'metrics': [{'component': str, 'resource': str, 'value': float}, ...],
     'traces':  [span-tree, ...]}

where a span tree is

    {'component': str, 'operation': str, 'children': [span-tree, ...]}

We keep that format byte-compatible (plain dicts/lists pickled) and add a
typed in-memory view on top of it.  Everything downstream (featurizer,
synthesizer, trainer, REST ingestion) speaks this contract.
"""

from __future__ import annotations

import pickle
from dataclasses import dataclass, field
from typing import Any, Dict, Iterable, List, Sequence


@dataclass
class Span:
    """One node of a trace tree (component, operation, children)."""

    component: str
    operation: str
    children: List["Span"] = field(default_factory=list)

    @property
    def op_id(self) -> str:
        """The component_operation identifier used by the call-path space."""
        return f"{self.component}_{self.operation}"

    def to_dict(self) -> Dict[str, Any]:
        """Iterative (deep chains must not exhaust the Python stack)."""
        out = {"component": self.component, "operation": self.operation,
               "children": []}
        stack = [(self, out)]
        while stack:
            src, dst = stack.pop()
            for c in src.children:
                cd = {"component": c.component, "operation": c.operation,
                      "children": []}
                dst["children"].append(cd)
                stack.append((c, cd))
        return out

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "Span":
        """Iterative build (deep chains must not exhaust the Python stack)."""
        root = Span(component=d["component"], operation=d["operation"])
        stack = [(d, root)]
        while stack:
            src, dst = stack.pop()
            for c in src.get("children", []):
                child = Span(component=c["component"], operation=c["operation"])
                dst.children.append(child)
                stack.append((c, child))
        return root

    def walk(self):
        """Depth-first pre-order traversal (iterative; deep trees safe)."""
        stack = [self]
        while stack:
            node = stack.pop()
            yield node
            # reversed so children are visited left-to-right
            stack.extend(reversed(node.children))


@dataclass
class Window:
    """One discretized time window: resource samples + collected traces."""

    metrics: List[Dict[str, Any]] = field(default_factory=list)
    traces: List[Span] = field(default_factory=list)

    def to_dict(self) -> Dict[str, Any]:
        return {
            "metrics": list(self.metrics),
            "traces": [t.to_dict() for t in self.traces],
        }

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "Window":
        return Window(
            metrics=list(d.get("metrics", [])),
            traces=[Span.from_dict(t) for t in d.get("traces", [])],
        )


class ContractError(ValueError):
    pass


def _validate_span(node: Any, where: str) -> None:
    # iterative: REST /ingest feeds arbitrary payloads here, and a deep
    # span chain must produce a ContractError, not a RecursionError
    stack = [(node, where)]
    while stack:
        cur, w = stack.pop()
        if not isinstance(cur, dict):
            raise ContractError(f"{w}: span must be a dict, got {type(cur).__name__}")
        for key in ("component", "operation"):
            if key not in cur or not isinstance(cur[key], str):
                raise ContractError(f"{w}: span missing string field '{key}'")
        children = cur.get("children", [])
        if not isinstance(children, list):
            raise ContractError(f"{w}: span 'children' must be a list")
        for i, child in enumerate(children):
            stack.append((child, f"{w}.children[{i}]"))


def validate_raw_data(raw_data: Any) -> None:
    """Validate the plain-dict raw-data structure; raise ContractError on issues.

    Mirrors the reference input format (resource-estimation/README.md:29-63).
    """
    if not isinstance(raw_data, (list, tuple)):
        raise ContractError(f"raw_data must be a list, got {type(raw_data).__name__}")
    for wi, window in enumerate(raw_data):
        where = f"raw_data[{wi}]"
        if not isinstance(window, dict):
            raise ContractError(f"{where}: window must be a dict")
        if "metrics" not in window or "traces" not in window:
            raise ContractError(f"{where}: window needs 'metrics' and 'traces' keys")
        if not isinstance(window["metrics"], list):
            raise ContractError(f"{where}.metrics must be a list")
        for mi, metric in enumerate(window["metrics"]):
            mwhere = f"{where}.metrics[{mi}]"
            if not isinstance(metric, dict):
                raise ContractError(f"{mwhere}: metric must be a dict")
            for key in ("component", "resource", "value"):
                if key not in metric:
                    raise ContractError(f"{mwhere}: metric missing '{key}'")
            if not isinstance(metric["value"], (int, float)):
                raise ContractError(f"{mwhere}: metric 'value' must be numeric")
        if not isinstance(window["traces"], list):
            raise ContractError(f"{where}.traces must be a list")
        for ti, trace in enumerate(window["traces"]):
            _validate_span(trace, f"{where}.traces[{ti}]")


def load_raw_data(path: str, validate: bool = True) -> List[Dict[str, Any]]:
    """Load a raw_data pickle in the reference on-disk format."""
    with open(path, "rb") as f:
        raw = pickle.load(f)
    if validate:
        validate_raw_data(raw)
    return raw


def save_raw_data(raw_data: Sequence[Dict[str, Any]], path: str) -> None:
    with open(path, "wb") as f:
        pickle.dump(list(raw_data), f)


def windows_from_raw(raw_data: Iterable[Dict[str, Any]]) -> List[Window]:
    """Typed view of plain raw_data."""
    return [Window.from_dict(w) for w in raw_data]


def raw_from_windows(windows: Iterable[Window]) -> List[Dict[str, Any]]:
    return [w.to_dict() for w in windows]
