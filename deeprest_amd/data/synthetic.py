"""Synthetic application / trace generator.

Replaces the reference's L1-L3 stack (DeathStarBench social network + Locust
load generation + Jaeger/Prometheus collection) with a
configurable generator that emits data in the raw-data contract:

- a random microservice call graph: each API endpoint owns a few trace
  *shapes* (span trees over components/operations) with empirical weights,
  mirroring how compose/read endpoints fan out in the reference app
  (reference: locust/locustfile-normal.py:88-144 drives those shapes);
- diurnal traffic with two Gaussian peaks per simulated day and noise
  (reference: locust/locustfile-normal.py:53-74), API mix from a Zipf
  popularity prior;
- ground-truth resource model: per component and resource type the
  utilization is an affine/EMA function of that component's invocation
  counts plus noise — so an estimator that learns traffic->resource
  causality can actually fit it;
- anomaly injection (utilization NOT justified by traffic) for the
  sanity-check capability (reference README.md:3: ransomware/cryptojacking
  detection).

Two output paths:
- ``generate_raw()``   — full span trees in the contract format (tests,
  small configs);
- ``generate_featurized()`` — the (T, P) traffic matrix and (T,) resource
  series directly, skipping tree construction (bench-scale configs:
  256/4096 endpoints).  Both paths agree: shapes' path-count vectors are
  precomputed from the same trees the raw path emits.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any, Dict, List, Optional, Sequence, Tuple

import numpy as np

from .contract import validate_raw_data
from .featurize import FeatureSpace, FeaturizedData

DEFAULT_RESOURCES = ("cpu", "memory", "write-iops")
# the reference estimates five resource types (reference:
# resource-estimation/utils.py:8-26); pass resources=ALL_RESOURCES for the
# full set — write-tp couples to write-iops via a per-component KB/op
# factor, and disk usage integrates write throughput (monotone growth)
ALL_RESOURCES = ("cpu", "memory", "write-iops", "write-tp", "usage")


@dataclass
class SyntheticAppConfig:
    n_apis: int = 8                      # API endpoints (root operations)
    n_components: int = 12               # microservice components
    resources: Tuple[str, ...] = DEFAULT_RESOURCES
    shapes_per_api: int = 3              # distinct trace shapes per endpoint
    max_depth: int = 4
    max_children: int = 3
    windows_per_day: int = 240           # discretization windows per simulated day
    n_days: int = 4
    base_calls: float = 20.0             # mean calls/window at trough
    peak_calls: float = 120.0            # mean calls/window at peaks
    noise: float = 0.05                  # traffic noise fraction
    resource_noise: float = 0.02         # resource observation noise fraction
    seed: int = 1234

    @property
    def n_windows(self) -> int:
        return self.windows_per_day * self.n_days


def _make_shape(
    rng: np.random.Generator,
    components: List[str],
    root_component: str,
    root_op: str,
    max_depth: int,
    max_children: int,
) -> Dict[str, Any]:
    """Random span tree rooted at (root_component, root_op)."""

    def grow(depth: int) -> List[Dict[str, Any]]:
        if depth >= max_depth:
            return []
        n_children = int(rng.integers(0, max_children + 1)) if depth > 0 else int(
            rng.integers(1, max_children + 1)
        )
        children = []
        for _ in range(n_children):
            comp = components[int(rng.integers(0, len(components)))]
            op = f"op{int(rng.integers(0, 4))}"
            children.append(
                {"component": comp, "operation": op, "children": grow(depth + 1)}
            )
        return children

    return {"component": root_component, "operation": root_op, "children": grow(0)}


class SyntheticApp:
    def __init__(self, config: Optional[SyntheticAppConfig] = None, **kwargs) -> None:
        if config is None:
            config = SyntheticAppConfig(**kwargs)
        self.config = config
        rng = np.random.default_rng(config.seed)
        self._rng = rng

        self.components = [f"svc-{i:04d}" for i in range(config.n_components)]
        # frontend component hosts all API root spans (like nginx-thrift)
        self.frontend = "frontend"
        self.apis: List[str] = []
        self.shapes: Dict[str, List[Dict[str, Any]]] = {}
        self.shape_weights: Dict[str, np.ndarray] = {}
        for i in range(config.n_apis):
            op = f"/api-{i:04d}"
            api_id = f"{self.frontend}_{op}"
            self.apis.append(api_id)
            shapes = [
                _make_shape(rng, self.components, self.frontend, op,
                            config.max_depth, config.max_children)
                for _ in range(config.shapes_per_api)
            ]
            self.shapes[api_id] = shapes
            # the shape MIX drifts over the timeline (real apps: cache-hit
            # ratios, request variants change) — a static component-count
            # linear map cannot track the resulting cost drift, call-path
            # features can
            w0 = rng.dirichlet(np.ones(config.shapes_per_api) * 2.0)
            w1 = rng.dirichlet(np.ones(config.shapes_per_api) * 2.0)
            self.shape_weights[api_id] = (w0, w1)

        # API popularity (Zipf-ish) and diurnal peaks
        pop = 1.0 / (1.0 + np.arange(config.n_apis)) ** 0.8
        self.popularity = pop / pop.sum()
        self.peak_positions = rng.uniform(0.15, 0.85, size=(config.n_days, 2))

        # shared feature space over all shapes (discovery order: api, then shape)
        self.feature_space = FeatureSpace()
        for api in self.apis:
            for shape in self.shapes[api]:
                self.feature_space.observe_trace(shape)
        P = len(self.feature_space)

        # precompute per-shape path-count and component-count vectors
        # per-api path-count matrices are stored COMPACT: an api's shapes only
        # touch its own ~dozen call paths, so (cols, (S, K) counts) instead of
        # a dense (S, P) row keeps construction and the traffic accumulation
        # linear in n_apis (dense was quadratic: P itself grows with n_apis —
        # 20 GB of zeros and ~13 min of matmuls at the 4096-endpoint config)
        self._shape_vec: Dict[str, tuple] = {}
        self._shape_comp: Dict[str, np.ndarray] = {}
        all_components = [self.frontend] + self.components
        self.component_index = {c: i for i, c in enumerate(all_components)}
        self.all_components = all_components
        for api in self.apis:
            per_shape: List[Dict[int, int]] = []
            comps = np.zeros((config.shapes_per_api, len(all_components)), dtype=np.int64)
            for s, shape in enumerate(self.shapes[api]):
                counts: Dict[int, int] = {}
                stack = [(shape, ())]
                while stack:
                    node, prefix = stack.pop()
                    path = prefix + ((node["component"], node["operation"]),)
                    idx = self.feature_space.index_of(path)
                    counts[idx] = counts.get(idx, 0) + 1
                    comps[s, self.component_index[node["component"]]] += 1
                    for child in node.get("children", []):
                        stack.append((child, path))
                per_shape.append(counts)
            cols = sorted(set().union(*per_shape))
            col_pos = {c: i for i, c in enumerate(cols)}
            mat = np.zeros((config.shapes_per_api, len(cols)), dtype=np.int64)
            for s, counts in enumerate(per_shape):
                for c, v in counts.items():
                    mat[s, col_pos[c]] = v
            self._shape_vec[api] = (np.asarray(cols, dtype=np.int64), mat)
            self._shape_comp[api] = comps

        # ground-truth resource model parameters per (component, resource).
        # The per-(api, shape) cost multipliers make utilization depend on the
        # PATH MIX, not just total component invocations — the premise of
        # call-path-aware estimation (a component-count linear model cannot
        # resolve two shapes hitting the same component at different cost).
        C = len(all_components)
        R = len(config.resources)
        self._res_base = rng.uniform(5.0, 50.0, size=(C, R))
        self._res_gain = rng.uniform(0.2, 2.0, size=(C, R))
        self._res_ema = rng.uniform(0.0, 0.9, size=(C, R))  # memory-like persistence
        # weighted per-shape component loads: shape s of api hits component c
        # with cost count * cost_mult (cost_mult in [0.2, 3])
        self._shape_comp_w: Dict[str, np.ndarray] = {}
        for api in self.apis:
            cost = rng.uniform(0.2, 3.0, size=self._shape_comp[api].shape)
            self._shape_comp_w[api] = self._shape_comp[api] * cost

        # five-resource-type dynamics, drawn from a SEPARATE derived stream
        # so 3-resource configs keep their round-1 RNG sequence
        # byte-identical (fixtures, bench configs):
        # - write-tp = write-iops x per-component KB/op (correlated columns,
        #   like a block device whose request size is a component property)
        # - usage integrates write throughput: monotone disk growth
        self._res_index = {r: i for i, r in enumerate(config.resources)}
        rng5 = np.random.default_rng(config.seed + 90001)
        self._kb_per_op = rng5.uniform(4.0, 64.0, size=C)
        self._usage_rate = rng5.uniform(0.002, 0.02, size=C)
        # (EMA state, usage level) at the end of the last from-scratch
        # generation — scenario query periods continue from here
        self._base_carry: Optional[tuple] = None

    # ------------------------------------------------------------------ traffic
    def traffic_plan(self, scale: float = 1.0, shape: str = "waves",
                     composition: Optional[Sequence[float]] = None) -> np.ndarray:
        """(T, n_apis) expected call counts.

        Scenario knobs mirror the reference's evaluation variants
        (reference: locust/locustfile-{scale,shape,composition}.py):
        - scale: user multiplier (unseen-scale scenario, e.g. 3.0);
        - shape: 'waves' (two diurnal Gaussian peaks/day) or 'flat'
          (steady at peak level — unseen-shape scenario);
        - composition: per-API mix weights overriding the Zipf popularity
          (unseen-composition scenario).
        """
        cfg = self.config
        t = np.arange(cfg.windows_per_day) / cfg.windows_per_day
        days = []
        for d in range(cfg.n_days):
            if shape == "flat":
                day = np.ones_like(t)
            else:
                p1, p2 = self.peak_positions[d]
                day = (
                    np.exp(-0.5 * ((t - p1) / 0.08) ** 2)
                    + np.exp(-0.5 * ((t - p2) / 0.08) ** 2)
                )
            days.append(day)
        shape_all = np.concatenate(days)  # (T,)
        level = (cfg.base_calls + (cfg.peak_calls - cfg.base_calls) * shape_all) * scale
        if composition is not None:
            mix = np.asarray(composition, dtype=np.float64)
            mix = mix / mix.sum()
        else:
            mix = self.popularity
        lam = level[:, None] * mix[None, :]
        noise = 1.0 + cfg.noise * self._rng.standard_normal(size=lam.shape)
        counts = self._rng.poisson(np.maximum(lam * noise, 0.0)).astype(np.int64)
        return counts

    def _sample_shape_counts(self, api_calls: np.ndarray) -> Dict[str, np.ndarray]:
        """Per api: (T, shapes_per_api) multinomial split of each window's
        calls, with the mix drifting linearly from w0 to w1 over the run."""
        out = {}
        T = api_calls.shape[0]
        alpha = np.linspace(0.0, 1.0, max(T, 2))[:T]
        for a, api in enumerate(self.apis):
            w0, w1 = self.shape_weights[api]
            calls = api_calls[:, a]
            picks = np.zeros((T, len(w0)), dtype=np.int64)
            for t in np.nonzero(calls > 0)[0]:
                wt = (1.0 - alpha[t]) * w0 + alpha[t] * w1
                picks[t] = self._rng.multinomial(int(calls[t]), wt)
            out[api] = picks
        return out

    def _resources_from_invocations(self, inv: np.ndarray,
                                    continue_state: bool = False) -> np.ndarray:
        """(T, C, R) ground-truth utilization from (T, C) invocation counts.

        ``continue_state=True`` starts from where the most recent
        from-scratch generation ended (EMA state + disk-usage level): a
        scenario's query period is the SAME deployment continuing, not a
        fresh one — restarting monotone metrics at zero would make query
        ground truth incomparable to levels learned in training.
        """
        cfg = self.config
        T, C = inv.shape
        R = len(cfg.resources)
        ri = self._res_index
        vals = np.zeros((T, C, R))
        if continue_state and self._base_carry is not None:
            state = self._base_carry[0].copy()
            u = self._base_carry[1]
            usage0 = u.copy() if u is not None else None
        else:
            state = np.zeros((C, R))
            usage0 = (self._res_base[:, ri["usage"]].copy()
                      if "usage" in ri else None)
        for t in range(T):
            drive = inv[t][:, None] * self._res_gain  # (C, R)
            state = self._res_ema * state + (1.0 - self._res_ema) * drive
            obs = self._res_base + state
            vals[t] = obs
        if "write-tp" in ri and "write-iops" in ri:
            # throughput in KB = IOps x per-component request size
            vals[:, :, ri["write-tp"]] = (
                vals[:, :, ri["write-iops"]] * self._kb_per_op[None, :])
        if "usage" in ri:
            # disk usage integrates write throughput (falls back to its own
            # affine response when no write-tp column exists)
            growth_src = (vals[:, :, ri["write-tp"]] if "write-tp" in ri
                          else vals[:, :, ri["usage"]])
            increments = self._usage_rate[None, :] * growth_src
            vals[:, :, ri["usage"]] = usage0[None, :] + np.cumsum(
                increments, axis=0)
        if not continue_state:
            self._base_carry = (
                state.copy(),
                vals[-1, :, ri["usage"]].copy() if "usage" in ri else None)
        vals *= 1.0 + cfg.resource_noise * self._rng.standard_normal(size=vals.shape)
        if "usage" in ri:
            # observation noise must not break monotonicity of disk usage
            vals[:, :, ri["usage"]] = np.maximum.accumulate(
                vals[:, :, ri["usage"]], axis=0)
        return np.maximum(vals, 0.0)

    # ------------------------------------------------------------------ outputs
    def generate_raw(self, plan: Optional[np.ndarray] = None,
                     continue_state: bool = False) -> List[Dict[str, Any]]:
        """Full contract-format raw_data with span trees.

        ``plan``: optional (T, n_apis) expected-call matrix from
        ``traffic_plan(...)`` — pass a scenario variant (unseen scale /
        shape / composition) to generate query-period data for what-if
        evaluation (reference: locustfile-{scale,shape,composition}.py).
        ``continue_state``: the plan continues the last from-scratch run's
        deployment (EMA + disk-usage state carry over).
        """
        cfg = self.config
        api_calls = self.traffic_plan() if plan is None else np.asarray(plan)
        shape_counts = self._sample_shape_counts(api_calls)
        T = cfg.n_windows
        C = len(self.all_components)

        inv = np.zeros((T, C), dtype=np.int64)
        drive = np.zeros((T, C))
        for api in self.apis:
            inv += shape_counts[api] @ self._shape_comp[api]
            drive += shape_counts[api] @ self._shape_comp_w[api]
        res = self._resources_from_invocations(drive, continue_state)

        raw = []
        for t in range(T):
            traces = []
            for api in self.apis:
                for s, shape in enumerate(self.shapes[api]):
                    traces.extend([shape] * int(shape_counts[api][t, s]))
            metrics = []
            for ci, comp in enumerate(self.all_components):
                for ri, resource in enumerate(cfg.resources):
                    metrics.append(
                        {"component": comp, "resource": resource,
                         "value": float(res[t, ci, ri])}
                    )
            raw.append({"metrics": metrics, "traces": traces})
        validate_raw_data(raw)
        return raw

    def generate_featurized(self, plan: Optional[np.ndarray] = None,
                            continue_state: bool = False) -> FeaturizedData:
        """Fast path: traffic matrix + resource series without building trees.

        ``plan`` as in :meth:`generate_raw` — a scenario traffic plan for
        query-period/what-if data sharing this app's feature space;
        ``continue_state`` continues the last from-scratch deployment.
        """
        cfg = self.config
        api_calls = self.traffic_plan() if plan is None else np.asarray(plan)
        shape_counts = self._sample_shape_counts(api_calls)
        T = cfg.n_windows
        P = len(self.feature_space)
        C = len(self.all_components)

        traffic = np.zeros((T, P), dtype=np.int64)
        inv = np.zeros((T, C), dtype=np.int64)
        drive = np.zeros((T, C))
        for api in self.apis:
            cols, mat = self._shape_vec[api]
            traffic[:, cols] += shape_counts[api] @ mat
            inv += shape_counts[api] @ self._shape_comp[api]
            drive += shape_counts[api] @ self._shape_comp_w[api]
        res = self._resources_from_invocations(drive, continue_state)

        resources = {}
        resource_components = {}
        for ci, comp in enumerate(self.all_components):
            for ri, resource in enumerate(cfg.resources):
                ident = f"{comp}_{resource}"
                resources[ident] = res[:, ci, ri].copy()
                resource_components[ident] = comp

        invocations = {c: inv[:, i].copy() for c, i in self.component_index.items()}
        invocations["general"] = api_calls.sum(axis=1).astype(np.int64)

        return FeaturizedData(
            traffic=traffic,
            resources=resources,
            invocations=invocations,
            feature_space=self.feature_space,
            resource_components=resource_components,
        )

    def inject_anomaly(
        self,
        data: FeaturizedData,
        component: str,
        resource: str,
        start: int,
        length: int,
        magnitude: float = 3.0,
    ) -> FeaturizedData:
        """Add traffic-unjustified utilization (cryptojacking-style CPU thief,
        reference locust/pow.py:29-38) to one component's series in-place."""
        ident = f"{component}_{resource}"
        series = data.resources[ident]
        baseline = float(np.median(series))
        series[start : start + length] += magnitude * max(baseline, 1.0)
        return data
