"""Live collection loop: CollectionPoller against fake Jaeger/Prometheus
HTTP servers (stdlib http.server in a thread) -> contract windows."""

import json
import threading
from http.server import BaseHTTPRequestHandler, HTTPServer
from urllib.parse import parse_qs, urlparse

from deeprest_amd.data.collector import Collector
from deeprest_amd.data.contract import validate_raw_data
from deeprest_amd.data.poller import CollectionPoller


def _trace(tid, t_us, comp="frontend", op="/api-0"):
    return {
        "traceID": tid,
        "spans": [
            {"traceID": tid, "spanID": f"{tid}-root", "operationName": op,
             "startTime": t_us, "processID": "p1", "references": []},
            {"traceID": tid, "spanID": f"{tid}-child", "operationName": "op1",
             "startTime": t_us + 10, "processID": "p2",
             "references": [{"refType": "CHILD_OF", "spanID": f"{tid}-root"}]},
        ],
        "processes": {"p1": {"serviceName": comp},
                      "p2": {"serviceName": "svc-a"}},
    }


class _Fake(BaseHTTPRequestHandler):
    polls = {"traces": 0, "prom": 0}

    def do_GET(self):  # noqa: N802
        u = urlparse(self.path)
        if u.path == "/api/traces":
            _Fake.polls["traces"] += 1
            qs = parse_qs(u.query)
            assert "service" in qs and "start" in qs and "end" in qs
            t_us = 1_700_000_000 * 1_000_000
            # same trace returned on every poll (overlap) + a fresh one
            body = {"data": [
                _trace("t-dup", t_us),
                _trace(f"t-{_Fake.polls['traces']}", t_us + 2_000_000),
            ]}
        elif u.path == "/api/v1/query":
            _Fake.polls["prom"] += 1
            qs = parse_qs(u.query)
            q = qs["query"][0]
            val = "123.5" if "cpu" in q else "7.0"
            body = {"status": "success", "data": {"result": [
                {"metric": {"component": "frontend"},
                 "value": [1_700_000_000 + _Fake.polls["prom"], val]},
                {"metric": {"pod": "svc-a"},     # component-label fallback
                 "value": [1_700_000_000, val]},
            ]}}
        else:
            self.send_response(404)
            self.end_headers()
            return
        data = json.dumps(body).encode()
        self.send_response(200)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(data)))
        self.end_headers()
        self.wfile.write(data)

    def log_message(self, *a):  # silence
        pass


def test_poller_end_to_end():
    _Fake.polls = {"traces": 0, "prom": 0}
    srv = HTTPServer(("127.0.0.1", 0), _Fake)
    thread = threading.Thread(target=srv.serve_forever, daemon=True)
    thread.start()
    try:
        base = f"http://127.0.0.1:{srv.server_port}"
        collector = Collector(window_sec=5.0)
        poller = CollectionPoller(
            collector, jaeger_url=base, prometheus_url=base,
            services=["frontend"],
            queries={"cpu": "rate(container_cpu_usage_seconds_total[1m])",
                     "memory": "container_memory_working_set_bytes"},
            interval_sec=5.0)
        n = poller.run(max_polls=3, sleep_fn=lambda s: None)
        assert n == 3
        assert _Fake.polls["traces"] == 3 and _Fake.polls["prom"] == 6
        windows = collector.windows()
        assert windows, "no contract windows produced"
        validate_raw_data(windows)
        # the duplicated trace was deduped: 1 dup + 3 fresh = 4 trees total
        total_traces = sum(len(w["traces"]) for w in windows)
        assert total_traces == 4
        # both resources present for the frontend component
        resources = {(m["component"], m["resource"])
                     for w in windows for m in w["metrics"]}
        assert ("frontend", "cpu") in resources
        assert ("frontend", "memory") in resources
        assert ("svc-a", "cpu") in resources   # pod-label fallback
    finally:
        srv.shutdown()


def test_poller_survives_http_errors():
    collector = Collector()
    poller = CollectionPoller(
        collector, jaeger_url="http://127.0.0.1:1",   # connection refused
        prometheus_url="http://127.0.0.1:1",
        services=["s"], queries={"cpu": "q"}, interval_sec=0.0,
        http_timeout=0.2)
    # loop keeps running through failures
    assert poller.run(max_polls=2, sleep_fn=lambda s: None) == 2


def test_cli_collect_command(tmp_path):
    """`deeprest-amd collect` drives the poller and writes contract data."""
    _Fake.polls = {"traces": 0, "prom": 0}
    srv = HTTPServer(("127.0.0.1", 0), _Fake)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    try:
        from deeprest_amd.cli import main
        from deeprest_amd.data.contract import load_raw_data

        base = f"http://127.0.0.1:{srv.server_port}"
        out = str(tmp_path / "raw.pkl")
        rc = main(["collect", "--jaeger", base, "--prometheus", base,
                   "--services", "frontend", "--query", "cpu=up",
                   "--max-polls", "2", "--interval", "5", "--out", out])
        assert rc == 0
        raw = load_raw_data(out)
        validate_raw_data(raw)
        assert any(w["traces"] for w in raw)
    finally:
        srv.shutdown()
