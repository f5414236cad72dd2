"""Live collection loop: poll Jaeger + Prometheus into the Collector.

Closes the reference's L3 loop (Jaeger query API +
Prometheus scrape API, 5 s interval — minikube-openebs/
monitor-openebs-pg.yaml:39): a thin poller that, on an interval, pulls

- finished traces from Jaeger's  ``GET /api/traces?service=S&start=U&end=U``
  (the same export format ``span_tree_from_jaeger`` consumes), and
- resource samples from Prometheus's ``GET /api/v1/query?query=Q&time=T``
  instant-query API,

and feeds them to a :class:`~deeprest_amd.data.collector.Collector`, which
discretizes into raw-data-contract windows.  Pure stdlib HTTP (urllib) —
works against real Jaeger/Prometheus or the in-test fake servers.
"""

from __future__ import annotations

import json
import time
import urllib.parse
import urllib.request
from typing import Callable, Dict, Iterable, List, Optional

from .collector import Collector


def _get_json(url: str, timeout: float) -> dict:
    with urllib.request.urlopen(url, timeout=timeout) as resp:
        return json.loads(resp.read().decode("utf-8"))


class CollectionPoller:
    """Polls Jaeger + Prometheus on an interval into a Collector.

    ``queries`` maps resource-type name -> PromQL (e.g. ``{"cpu":
    'rate(container_cpu_usage_seconds_total[1m])*1000'}``); the component
    name is read from ``component_label`` on each result's metric labels.
    """

    def __init__(
        self,
        collector: Collector,
        jaeger_url: str,
        prometheus_url: str,
        services: Iterable[str],
        queries: Dict[str, str],
        component_label: str = "component",
        interval_sec: float = 5.0,
        lookback_sec: Optional[float] = None,
        http_timeout: float = 10.0,
        trace_limit: int = 2000,
    ) -> None:
        self.collector = collector
        self.jaeger_url = jaeger_url.rstrip("/")
        self.prometheus_url = prometheus_url.rstrip("/")
        self.services = list(services)
        self.queries = dict(queries)
        self.component_label = component_label
        self.interval_sec = interval_sec
        self.lookback_sec = lookback_sec or 2 * interval_sec
        self.http_timeout = http_timeout
        self.trace_limit = trace_limit
        self._seen_traces: set = set()
        self._last_poll: Optional[float] = None

    # ------------------------------------------------------------ one poll
    def poll_once(self, now: Optional[float] = None) -> Dict[str, int]:
        now = time.time() if now is None else now
        start = (self._last_poll if self._last_poll is not None
                 else now - self.lookback_sec)
        # overlap one interval: Jaeger indexes traces when they COMPLETE,
        # so a hard cursor at last-poll time drops late arrivals; the
        # seen-set dedups the overlap
        start = max(0.0, start - self.interval_sec)

        n_traces = 0
        for service in self.services:
            q = urllib.parse.urlencode({
                "service": service,
                "start": int(start * 1e6),
                "end": int(now * 1e6),
                "limit": self.trace_limit,
            })
            body = _get_json(f"{self.jaeger_url}/api/traces?{q}",
                             self.http_timeout)
            fresh = []
            for trace in body.get("data", []) or []:
                tid = trace.get("traceID")
                if tid is not None and tid in self._seen_traces:
                    continue
                if tid is not None:
                    self._seen_traces.add(tid)
                fresh.append(trace)
            n_traces += self.collector.add_traces(fresh)
        if len(self._seen_traces) > 50 * self.trace_limit:
            self._seen_traces.clear()   # bounded memory; overlap is short

        n_samples = 0
        samples: List[dict] = []
        for resource, promql in self.queries.items():
            q = urllib.parse.urlencode({"query": promql, "time": now})
            body = _get_json(f"{self.prometheus_url}/api/v1/query?{q}",
                             self.http_timeout)
            result = (body.get("data") or {}).get("result", []) or []
            for row in result:
                labels = row.get("metric", {})
                comp = labels.get(self.component_label)
                if comp is None:
                    comp = labels.get("pod") or labels.get("service")
                if comp is None:
                    continue
                ts, val = row.get("value", [now, "nan"])
                try:
                    v = float(val)
                except (TypeError, ValueError):
                    continue
                samples.append({"component": comp, "resource": resource,
                                "value": v, "timestamp": float(ts)})
        n_samples = self.collector.add_metric_samples(samples)

        self._last_poll = now
        return {"traces": n_traces, "samples": n_samples}

    # ---------------------------------------------------------------- loop
    def run(self, max_polls: Optional[int] = None,
            duration_sec: Optional[float] = None,
            sleep_fn: Callable[[float], None] = time.sleep,
            on_poll: Optional[Callable[[Dict[str, int]], None]] = None) -> int:
        """Poll every ``interval_sec`` until max_polls/duration reached.
        Returns the number of polls performed.  HTTP errors of one poll are
        logged-and-skipped (the collection plane outliving a flaky scrape
        is the whole point of a poller)."""
        t_end = (time.time() + duration_sec) if duration_sec else None
        polls = 0
        while True:
            if max_polls is not None and polls >= max_polls:
                break
            if t_end is not None and time.time() >= t_end:
                break
            try:
                stats = self.poll_once()
                if on_poll is not None:
                    on_poll(stats)
            except Exception as exc:  # noqa: BLE001 — keep the loop alive
                import sys

                print(f"poll failed: {exc}", file=sys.stderr)
            polls += 1
            if max_polls is not None and polls >= max_polls:
                break
            sleep_fn(self.interval_sec)
        return polls
