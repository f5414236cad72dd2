"""Command-line interface.

Replaces the reference's edit-the-constants workflow
(reference: resource-estimation/estimate.py:13-18)
with subcommands over the typed config:

  python -m deeprest_amd.cli featurize --raw raw_data.pkl --out input.pkl
  python -m deeprest_amd.cli train [--config cfg.yaml] [--set train.epochs=10]
  python -m deeprest_amd.cli experiment --name exp1 --out results.pkl
  python -m deeprest_amd.cli synthesize --raw raw_data.pkl --list-apis
  python -m deeprest_amd.cli serve --checkpoint ckpt.pt --port 2021
"""

from __future__ import annotations

import argparse
import sys

import numpy as np


def _load_config(args):
    from .engine.config import EngineConfig, apply_cli_overrides

    cfg = EngineConfig.load(args.config) if args.config else EngineConfig()
    if args.set:
        cfg = apply_cli_overrides(cfg, args.set)
    return cfg


def _load_data(cfg):
    from .data.featurize import FeaturizedData, Featurizer
    from .data.contract import load_raw_data
    from .data.synthetic import SyntheticApp, SyntheticAppConfig

    d = cfg.data
    if d.input_path:
        return FeaturizedData.load(d.input_path)
    if d.raw_path:
        return Featurizer().fit_transform(load_raw_data(d.raw_path))
    app = SyntheticApp(SyntheticAppConfig(
        n_apis=d.synth_apis, n_components=d.synth_components,
        windows_per_day=d.synth_windows_per_day, n_days=d.synth_days,
        seed=d.synth_seed,
    ))
    print(f"[deeprest-amd] no input given; synthetic app "
          f"({d.synth_apis} endpoints, {d.synth_components} components)")
    return app.generate_featurized()


def cmd_featurize(args):
    from .data.contract import load_raw_data
    from .data.featurize import Featurizer

    raw = load_raw_data(args.raw)
    data = Featurizer().fit_transform(raw)
    data.save(args.out)
    print(f"featurized {data.num_windows} windows -> {data.num_paths} call paths, "
          f"{len(data.metric_names)} metrics; saved {args.out}")
    return 0


def cmd_train(args):
    import torch

    from .engine.trainer import Trainer

    cfg = _load_config(args)
    data = _load_data(cfg)
    trainer = Trainer(data, cfg)
    result = trainer.train()
    print(result.summary())
    if result.coverage:
        import numpy as _np

        covs = [c["coverage"] for c in result.coverage.values()]
        print(f"quantile-band coverage: mean {float(_np.mean(covs)):.3f} "
              f"min {float(_np.min(covs)):.3f} (nominal "
              f"{1.0 - 2 * cfg.model.quantiles[0]:.2f})")
    print(f"samples/sec: {result.samples_per_sec:.1f}")
    if cfg.train.checkpoint_path:
        print(f"checkpoint: {cfg.train.checkpoint_path}")
    return 0


def cmd_experiment(args):
    from .engine.experiment import run_experiment, run_scenario_suite

    cfg = _load_config(args)
    if args.suite:
        # unseen-traffic scenario suite needs a traffic-plan-capable app,
        # i.e. the synthetic generator (external data has no counterfactuals)
        from .data.synthetic import ALL_RESOURCES, SyntheticApp, SyntheticAppConfig

        d = cfg.data
        app = SyntheticApp(SyntheticAppConfig(
            n_apis=d.synth_apis, n_components=d.synth_components,
            windows_per_day=d.synth_windows_per_day, n_days=d.synth_days,
            seed=d.synth_seed,
            # the reference's full resource set (utils.py:8-26)
            resources=ALL_RESOURCES,
        ))
        store = run_scenario_suite(app, base_name=args.name, config=cfg)
        from .engine.experiment import scenario_error_tables

        for exp, per_est in scenario_error_tables(store).items():
            print(f"===== {exp} =====")
            for est, t in per_est.items():
                print(f"   {est:>9} => Median: {t['median']:.4f} | "
                      f"95-th: {t['p95']:.4f} | 99-th: {t['p99']:.4f} | "
                      f"Max: {t['max']:.4f}")
    else:
        data = _load_data(cfg)
        store = run_experiment(data, args.name, config=cfg)
    store.save(args.out)
    print(f"experiment '{args.name}' written to {args.out}")
    return 0


def cmd_synthesize(args):
    from .data.contract import load_raw_data
    from .data.synthesizer import TraceSynthesizer

    raw = load_raw_data(args.raw)
    syn = TraceSynthesizer().fit(raw)
    print(f"{len(syn.apis)} API endpoints found:")
    for api in syn.apis:
        print(f"    > {api}")
    if args.api:
        calls = {args.api: args.count}
        vec = syn.synthesize(calls, rng=np.random.default_rng(args.seed))
        print(f"feature vector for {calls}: {vec.tolist()}")
    return 0


def cmd_collect(args):
    from .data.collector import Collector
    from .data.contract import save_raw_data
    from .data.poller import CollectionPoller

    queries = {}
    for spec in args.query or []:
        resource, _, promql = spec.partition("=")
        if not promql:
            raise SystemExit(f"--query needs resource=promql, got {spec!r}")
        queries[resource] = promql
    collector = Collector(window_sec=args.interval)
    poller = CollectionPoller(
        collector,
        jaeger_url=args.jaeger,
        prometheus_url=args.prometheus,
        services=args.services.split(","),
        queries=queries,
        component_label=args.component_label,
        interval_sec=args.interval,
    )
    n = poller.run(
        max_polls=args.max_polls,
        duration_sec=args.duration,
        on_poll=lambda s: print(f"poll: +{s['traces']} traces, "
                                f"+{s['samples']} samples"),
    )
    windows = collector.windows()
    save_raw_data(windows, args.out)
    print(f"{n} polls -> {len(windows)} contract windows -> {args.out}")
    return 0


def cmd_serve(args):
    import uvicorn

    from .serve.api import create_app

    app = create_app(checkpoint_path=args.checkpoint, results_path=args.results,
                     micro_batch=args.micro_batch)
    uvicorn.run(app, host=args.host, port=args.port)
    return 0


def main(argv=None):
    p = argparse.ArgumentParser(prog="deeprest-amd")
    sub = p.add_subparsers(dest="cmd", required=True)

    f = sub.add_parser("featurize")
    f.add_argument("--raw", required=True)
    f.add_argument("--out", default="input.pkl")

    for name in ("train", "experiment"):
        t = sub.add_parser(name)
        t.add_argument("--config", default=None)
        t.add_argument("--set", nargs="*", action="extend", default=[])
        if name == "experiment":
            t.add_argument("--name", required=True)
            t.add_argument("--out", default="results.pkl")
            t.add_argument("--suite", action="store_true",
                           help="unseen-traffic scenario suite (synthetic app)")

    s = sub.add_parser("synthesize")
    s.add_argument("--raw", required=True)
    s.add_argument("--api", default=None)
    s.add_argument("--count", type=int, default=10)
    s.add_argument("--seed", type=int, default=0)
    s.add_argument("--list-apis", action="store_true")

    c = sub.add_parser("collect", help="live Jaeger/Prometheus polling loop"
                       " -> raw_data.pkl (the reference's L3 collection plane)")
    c.add_argument("--jaeger", required=True, help="Jaeger query base URL")
    c.add_argument("--prometheus", required=True, help="Prometheus base URL")
    c.add_argument("--services", required=True,
                   help="comma-separated Jaeger service names")
    c.add_argument("--query", action="append", metavar="RESOURCE=PROMQL",
                   help="repeatable, e.g. cpu=rate(container_cpu_usage"
                        "_seconds_total[1m])*1000")
    c.add_argument("--component-label", default="component")
    c.add_argument("--interval", type=float, default=5.0,
                   help="poll + window seconds (reference scrape: 5 s)")
    c.add_argument("--duration", type=float, default=None)
    c.add_argument("--max-polls", type=int, default=None)
    c.add_argument("--out", default="raw_data.pkl")

    v = sub.add_parser("serve")
    v.add_argument("--checkpoint", default=None)
    v.add_argument("--results", default=None, help="results.pkl to browse at /results")
    # 127.0.0.1 by default: the API is unauthenticated, so external
    # binding must be an explicit opt-in (--host 0.0.0.0)
    v.add_argument("--host", default="127.0.0.1")
    v.add_argument("--port", type=int, default=2021)
    v.add_argument("--micro-batch", action="store_true",
                   help="coalesce concurrent /predict requests into one "
                        "hipGraph replay (serve/batcher.py)")

    args = p.parse_args(argv)
    return {
        "featurize": cmd_featurize,
        "train": cmd_train,
        "experiment": cmd_experiment,
        "synthesize": cmd_synthesize,
        "collect": cmd_collect,
        "serve": cmd_serve,
    }[args.cmd](args)


if __name__ == "__main__":
    sys.exit(main())
