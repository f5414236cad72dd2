"""DeepRestNet — the MI355X-native estimation engine.

Architecture (the north star's re-design of the reference QuantileRNN,
reference: resource-estimation/qrnn.py:6-56):

  traffic (B, T, P call-path counts)
    -> input projection (P -> D)                      [rocBLAS GEMM]
    -> attention traffic encoder, n_layers x
         pre-LN MHA over the T axis + FFN            [HIP MHA + LayerNorm
                                                       kernels, rocBLAS GEMMs]
    -> call-graph propagation over the component
       graph (mean-aggregate neighbor embeddings)    [small GEMMs]
    -> per-component FiLM-conditioned GRU decoders   [fused HIP GRU sequence
       over time (optionally bidirectional)           kernel — the hot op]
    -> per-resource-type quantile heads (H -> Q)     [small GEMMs]
  output: (B, T, M, Q) quantile predictions for every component_resource
          metric, Q = (.05, .50, .95)

Where the reference runs one bidirectional GRU per metric and mixes experts
by averaging all other experts' outputs (qrnn.py:46-53, O(M^2) in metrics),
this design shares one encoder, conditions a shared recurrent decoder per
*component* via FiLM, and propagates information along the actual
application call graph — O(M), and every stage maps onto MFMA-shaped
compute.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Sequence, Tuple

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F

from ..data.featurize import FeaturizedData
from ..ops import fused_gru_sequence, layer_norm, mha_forward, pinball_loss
from ..ops.linear_bigk import bigk_linear


# --------------------------------------------------------------------- spec
@dataclass
class ModelSpec:
    """Static application structure the model is built around."""

    num_paths: int
    components: List[str]                 # decoder index -> component name
    resources: List[str]                  # head index -> resource type
    metric_names: List[str]               # output order (dataset order)
    comp_of: List[int]                    # metric -> component index
    res_of: List[int]                     # metric -> resource-type index
    adjacency: np.ndarray                 # (C, C) row-normalized call graph

    @property
    def num_components(self) -> int:
        return len(self.components)

    @property
    def num_metrics(self) -> int:
        return len(self.metric_names)


def _split_metric_ident(ident: str, known_components: Sequence[str]) -> Tuple[str, str]:
    """component_resource -> (component, resource), robust to '_' in names."""
    best = None
    for comp in known_components:
        if ident.startswith(comp + "_") and (best is None or len(comp) > len(best)):
            best = comp
    if best is not None:
        return best, ident[len(best) + 1 :]
    comp, _, res = ident.rpartition("_")
    return comp, res


def build_model_spec(data: FeaturizedData) -> ModelSpec:
    """Derive the static structure (components, resources, call graph)."""
    known = [c for c in data.invocations.keys() if c != "general"]
    comps: List[str] = []
    ress: List[str] = []
    comp_of: List[int] = []
    res_of: List[int] = []
    for ident in data.metric_names:
        if ident in data.resource_components:
            comp = data.resource_components[ident]
            res = ident[len(comp) + 1 :]
        else:
            comp, res = _split_metric_ident(ident, known)
        if comp not in comps:
            comps.append(comp)
        if res not in ress:
            ress.append(res)
        comp_of.append(comps.index(comp))
        res_of.append(ress.index(res))

    C = len(comps)
    A = np.eye(C, dtype=np.float64)
    if data.feature_space is not None:
        cidx = {c: i for i, c in enumerate(comps)}
        for path in data.feature_space.paths:
            if len(path) < 2:
                continue
            parent_comp = path[-2][0]
            child_comp = path[-1][0]
            pi, ci = cidx.get(parent_comp), cidx.get(child_comp)
            if pi is not None and ci is not None and pi != ci:
                A[pi, ci] = 1.0
                A[ci, pi] = 1.0
    A = A / A.sum(axis=1, keepdims=True)
    return ModelSpec(
        num_paths=data.num_paths,
        components=comps,
        resources=ress,
        metric_names=list(data.metric_names),
        comp_of=comp_of,
        res_of=res_of,
        adjacency=A,
    )


# ------------------------------------------------------------------- config
@dataclass
class DeepRestNetConfig:
    d_model: int = 256
    n_heads: int = 8
    n_layers: int = 2
    d_ff: int = 512
    hidden: int = 128                     # GRU hidden size
    comp_dim: int = 64                    # component embedding size
    quantiles: Tuple[float, ...] = (0.05, 0.50, 0.95)
    dropout: float = 0.1
    bidirectional: bool = True
    prop_rounds: int = 2                  # call-graph propagation rounds
    fp8_inference: bool = False           # fp8 MFMA GRU decode at eval time
    linear_bias: bool = False             # biases on encoder/x_proj Linears:
                                          # biasless default (LayerNorms and
                                          # metric_bias absorb shifts) saves
                                          # ~1.1 ms/step of bias-grad
                                          # reductions at accuracy parity
                                          # (profiles/r02_perf_notes.md)

    def to_dict(self) -> dict:
        return {
            "d_model": self.d_model, "n_heads": self.n_heads,
            "n_layers": self.n_layers, "d_ff": self.d_ff,
            "hidden": self.hidden, "comp_dim": self.comp_dim,
            "quantiles": tuple(self.quantiles), "dropout": self.dropout,
            "bidirectional": self.bidirectional, "prop_rounds": self.prop_rounds,
            "fp8_inference": self.fp8_inference, "linear_bias": self.linear_bias,
        }


# ------------------------------------------------------------------ modules
class _LayerNormOp(nn.Module):
    """nn.LayerNorm replacement routed through the HIP kernel on GPU."""

    def __init__(self, dim: int, eps: float = 1e-5) -> None:
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.bias = nn.Parameter(torch.zeros(dim))
        self.eps = eps

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return layer_norm(x, self.weight, self.bias, self.eps)


class _EncoderLayer(nn.Module):
    def __init__(self, cfg: DeepRestNetConfig) -> None:
        super().__init__()
        d = cfg.d_model
        self.ln1 = _LayerNormOp(d)
        self.ln2 = _LayerNormOp(d)
        lb = cfg.linear_bias
        self.qkv = nn.Linear(d, 3 * d, bias=lb)
        self.proj = nn.Linear(d, d, bias=lb)
        self.ff1 = nn.Linear(d, cfg.d_ff, bias=lb)
        self.ff2 = nn.Linear(cfg.d_ff, d, bias=lb)
        self.n_heads = cfg.n_heads
        self.dropout = nn.Dropout(cfg.dropout)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, T, D = x.shape
        h = self.ln1(x)
        qkv = self.qkv(h).view(B, T, 3, self.n_heads, D // self.n_heads)
        q, k, v = (qkv[:, :, i].transpose(1, 2) for i in range(3))  # (B,h,T,dh)
        attn = mha_forward(q, k, v)
        attn = attn.transpose(1, 2).reshape(B, T, D)
        x = x + self.dropout(self.proj(attn))
        h = self.ln2(x)
        x = x + self.dropout(self.ff2(F.gelu(self.ff1(h))))
        return x


class _GraphPropagation(nn.Module):
    """Mean-aggregate message passing over the application call graph."""

    def __init__(self, cfg: DeepRestNetConfig, num_components: int,
                 adjacency: np.ndarray) -> None:
        super().__init__()
        self.emb = nn.Parameter(torch.randn(num_components, cfg.comp_dim) * 0.02)
        self.w_self = nn.ModuleList(
            [nn.Linear(cfg.comp_dim, cfg.comp_dim) for _ in range(cfg.prop_rounds)]
        )
        self.w_nbr = nn.ModuleList(
            [nn.Linear(cfg.comp_dim, cfg.comp_dim) for _ in range(cfg.prop_rounds)]
        )
        self.register_buffer(
            "adj", torch.from_numpy(np.asarray(adjacency, dtype=np.float32))
        )

    def forward(self) -> torch.Tensor:
        e = self.emb
        for ws, wn in zip(self.w_self, self.w_nbr):
            e = F.relu(ws(e) + wn(self.adj @ e))
        return e                                               # (C, comp_dim)


class _GRUDecoderBank(nn.Module):
    """Shared-weight, per-component FiLM-conditioned GRU over time."""

    def __init__(self, cfg: DeepRestNetConfig, num_components: int) -> None:
        super().__init__()
        H = cfg.hidden
        D = cfg.d_model
        self.hidden = H
        self.bidirectional = cfg.bidirectional
        self.x_proj = nn.Linear(D, 3 * H, bias=cfg.linear_bias)
        self.w_hh = nn.Parameter(torch.empty(3 * H, H))
        self.b_hh = nn.Parameter(torch.zeros(3 * H))
        self.cond_gamma = nn.Linear(cfg.comp_dim, 3 * H)
        self.cond_beta = nn.Linear(cfg.comp_dim, 3 * H)
        self.h0_proj = nn.Linear(cfg.comp_dim, H)
        if cfg.bidirectional:
            self.x_proj_r = nn.Linear(D, 3 * H, bias=cfg.linear_bias)
            self.w_hh_r = nn.Parameter(torch.empty(3 * H, H))
            self.b_hh_r = nn.Parameter(torch.zeros(3 * H))
            self.cond_gamma_r = nn.Linear(cfg.comp_dim, 3 * H)
            self.cond_beta_r = nn.Linear(cfg.comp_dim, 3 * H)
            self.h0_proj_r = nn.Linear(cfg.comp_dim, H)
        self.reset_parameters()

    def reset_parameters(self) -> None:
        stdv = 1.0 / (self.hidden ** 0.5)
        with torch.no_grad():
            self.w_hh.uniform_(-stdv, stdv)
            if self.bidirectional:
                self.w_hh_r.uniform_(-stdv, stdv)
            # FiLM starts as identity: gamma ~ 1, beta ~ 0
            for g in (self.cond_gamma, getattr(self, "cond_gamma_r", None)):
                if g is not None:
                    g.weight.mul_(0.1)
                    g.bias.fill_(1.0)
            for b in (self.cond_beta, getattr(self, "cond_beta_r", None)):
                if b is not None:
                    b.weight.mul_(0.1)
                    b.bias.zero_()

    def forward(self, enc: torch.Tensor, comp: torch.Tensor, fp8: bool = False):
        """enc: (B, T, D); comp: (C, comp_dim) -> tuple of direction outputs
        (each (B, T, C, H)); the heads consume the halves separately so no
        concat copy is ever materialized."""
        B = enc.shape[0]
        C = comp.shape[0]
        xg = self.x_proj(enc)                                  # (B, T, 3H)
        gamma = self.cond_gamma(comp)                          # (C, 3H)
        beta = self.cond_beta(comp)
        h0 = torch.tanh(self.h0_proj(comp)).unsqueeze(0).expand(B, C, self.hidden)
        out = fused_gru_sequence(xg, self.w_hh, self.b_hh, h0.contiguous(),
                                 gamma, beta, reverse=False, fp8=fp8)
        if not self.bidirectional:
            return (out,)
        xg_r = self.x_proj_r(enc)
        gamma_r = self.cond_gamma_r(comp)
        beta_r = self.cond_beta_r(comp)
        h0_r = torch.tanh(self.h0_proj_r(comp)).unsqueeze(0).expand(B, C, self.hidden)
        out_r = fused_gru_sequence(xg_r, self.w_hh_r, self.b_hh_r, h0_r.contiguous(),
                                   gamma_r, beta_r, reverse=True, fp8=fp8)
        return (out, out_r)


# -------------------------------------------------------------------- model
class DeepRestNet(nn.Module):
    def __init__(self, spec: ModelSpec, cfg: Optional[DeepRestNetConfig] = None) -> None:
        super().__init__()
        self.spec = spec
        self.cfg = cfg or DeepRestNetConfig()
        cfg = self.cfg

        # in_norm immediately re-centers, so the in_proj bias is redundant
        # when biasless mode is on
        self.in_proj = nn.Linear(spec.num_paths, cfg.d_model,
                                 bias=cfg.linear_bias)
        self.in_norm = _LayerNormOp(cfg.d_model)
        self.layers = nn.ModuleList([_EncoderLayer(cfg) for _ in range(cfg.n_layers)])
        self.graph = _GraphPropagation(cfg, spec.num_components, spec.adjacency)
        self.decoder = _GRUDecoderBank(cfg, spec.num_components)
        self.dropout = nn.Dropout(cfg.dropout)

        dirs = 2 if cfg.bidirectional else 1
        Q = len(cfg.quantiles)
        # bias=False: metric_bias below subsumes it, and the per-head bias
        # gradient was a pathological 28M->9 reduction on the GPU
        self.heads = nn.ModuleList(
            [nn.Linear(cfg.hidden * dirs, Q, bias=False)
             for _ in range(len(spec.resources))]
        )
        self.metric_bias = nn.Parameter(torch.zeros(spec.num_metrics, Q))
        self.register_buffer("comp_of", torch.tensor(spec.comp_of, dtype=torch.long))
        self.register_buffer("res_of", torch.tensor(spec.res_of, dtype=torch.long))
        # fast path: when the metric order IS (component-major x resource) the
        # per-metric gather is a plain reshape (true for the synthetic apps;
        # avoids an indexing_backward scatter over ~10^6 rows)
        C, R, M = spec.num_components, len(spec.resources), spec.num_metrics
        grid_c = [c for c in range(C) for _ in range(R)]
        grid_r = list(range(R)) * C
        self._gather_is_reshape = (M == C * R and spec.comp_of == grid_c
                                   and spec.res_of == grid_r)

        # positional encoding over the window (sinusoidal, not learned: windows
        # slide, so absolute position has no meaning beyond phase)
        self._pos_cache: Optional[torch.Tensor] = None

    def _pos_encoding(self, T: int, device, dtype) -> torch.Tensor:
        c = self._pos_cache
        if c is not None and c.shape[0] >= T and c.device == device:
            return c[:T].to(dtype)
        D = self.cfg.d_model
        pos = torch.arange(T, device=device, dtype=torch.float32).unsqueeze(1)
        i = torch.arange(0, D, 2, device=device, dtype=torch.float32)
        div = torch.exp(-np.log(10000.0) * i / D)
        pe = torch.zeros(T, D, device=device)
        pe[:, 0::2] = torch.sin(pos * div)
        pe[:, 1::2] = torch.cos(pos * div)
        self._pos_cache = pe
        return pe.to(dtype)

    def forward(self, traffic: torch.Tensor) -> torch.Tensor:
        """traffic: (B, T, P) normalized call-path counts -> (B, T, M, Q)."""
        B, T, P = traffic.shape
        x = self.in_proj(traffic)
        x = x + self._pos_encoding(T, x.device, x.dtype)
        x = self.in_norm(x)
        for layer in self.layers:
            x = layer(x)
        comp = self.graph()                                   # (C, comp_dim)
        fp8 = (self.cfg.fp8_inference and not self.training
               and traffic.is_cuda and not torch.is_grad_enabled())
        h_dirs = self.decoder(x, comp, fp8=fp8)               # tuple of (B,T,C,H)
        return self._apply_heads(h_dirs)

    def _apply_heads(self, h_dirs) -> torch.Tensor:
        """Per-resource heads as a sum of per-direction GEMMs (no concat),
        each with a batched-K backward (ops/linear_bigk.py)."""
        H = self.cfg.hidden
        w_all = torch.cat([h.weight for h in self.heads], dim=0)   # (R*Q, H*dirs)
        out = None
        for d, h_d in enumerate(h_dirs):
            h_d = self.dropout(h_d)
            part = bigk_linear(h_d, w_all[:, d * H : (d + 1) * H].contiguous())
            out = part if out is None else out + part
        B, T, C = h_dirs[0].shape[0], h_dirs[0].shape[1], h_dirs[0].shape[2]
        R = len(self.heads)
        Q = len(self.cfg.quantiles)
        outs = out.view(B, T, C, R, Q)
        if self._gather_is_reshape:
            preds = outs.reshape(B, T, C * R, Q)
        else:
            preds = outs[:, :, self.comp_of, self.res_of, :]  # (B, T, M, Q)
        return preds + self.metric_bias

    def loss(self, outputs: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
        return pinball_loss(outputs, labels, self.cfg.quantiles)

    # -------------------------------------------------- long-horizon path
    @torch.no_grad()
    def forward_long(self, traffic: torch.Tensor, chunk_size: int = 4096) -> torch.Tensor:
        """Streaming inference over arbitrarily long horizons (SURVEY.md 5.7,
        BASELINE config 5: 7-day@1s ~ 604,800 steps).

        The encoder attends within chunks (attention cost stays
        O(chunk^2) per chunk); the GRU decoders carry hidden state across
        chunk boundaries — a forward sweep left->right and, when
        bidirectional, a reverse sweep right->left — so recurrence is exact
        over the whole horizon while peak memory is O(chunk).

        traffic: (B, T_total, P) -> (B, T_total, M, Q)
        """
        B, T_total, P = traffic.shape
        cfg = self.cfg
        dec = self.decoder
        comp = self.graph()
        C = comp.shape[0]
        H = dec.hidden
        dev, dt = traffic.device, traffic.dtype

        bounds = list(range(0, T_total, chunk_size)) + [T_total]
        chunks = list(zip(bounds[:-1], bounds[1:]))

        def encode(x):
            x = self.in_proj(x)
            x = x + self._pos_encoding(x.shape[1], x.device, x.dtype)
            x = self.in_norm(x)
            for layer in self.layers:
                x = layer(x)
            return x

        # fp8 MFMA decode (BASELINE config 5) applies to the long-horizon
        # path exactly as to forward()
        fp8 = bool(cfg.fp8_inference and traffic.is_cuda)

        # forward sweep
        h_f = torch.tanh(dec.h0_proj(comp)).unsqueeze(0).expand(B, C, H).contiguous().to(dt)
        gamma = dec.cond_gamma(comp)
        beta = dec.cond_beta(comp)
        fwd_outs = []
        enc_cache = []
        for (s, e) in chunks:
            enc = encode(traffic[:, s:e])
            enc_cache.append(enc)
            xg = dec.x_proj(enc)
            out = fused_gru_sequence(xg, dec.w_hh, dec.b_hh, h_f, gamma, beta,
                                     reverse=False, fp8=fp8)
            h_f = out[:, -1].contiguous()
            fwd_outs.append(out)

        if cfg.bidirectional:
            gamma_r = dec.cond_gamma_r(comp)
            beta_r = dec.cond_beta_r(comp)
            h_r = torch.tanh(dec.h0_proj_r(comp)).unsqueeze(0).expand(B, C, H).contiguous().to(dt)
            rev_outs = [None] * len(chunks)
            for ci in range(len(chunks) - 1, -1, -1):
                xg_r = dec.x_proj_r(enc_cache[ci])
                out_r = fused_gru_sequence(xg_r, dec.w_hh_r, dec.b_hh_r, h_r,
                                           gamma_r, beta_r, reverse=True,
                                           fp8=fp8)
                h_r = out_r[:, 0].contiguous()
                rev_outs[ci] = out_r
            h_chunks = list(zip(fwd_outs, rev_outs))
        else:
            h_chunks = [(f,) for f in fwd_outs]

        preds = [self._apply_heads(h_dirs) for h_dirs in h_chunks]
        return torch.cat(preds, dim=1)

    # ---- checkpoint helpers (spec travels with the weights) ----
    def full_state(self) -> dict:
        return {
            "state_dict": self.state_dict(),
            "config": self.cfg.to_dict(),
            "spec": {
                "num_paths": self.spec.num_paths,
                "components": self.spec.components,
                "resources": self.spec.resources,
                "metric_names": self.spec.metric_names,
                "comp_of": self.spec.comp_of,
                "res_of": self.spec.res_of,
                "adjacency": self.spec.adjacency,
            },
        }

    @staticmethod
    def from_full_state(state: dict) -> "DeepRestNet":
        spec = ModelSpec(**state["spec"])
        config = dict(state["config"])
        # checkpoints from before the biasless default carried biases
        config.setdefault("linear_bias", True)
        cfg = DeepRestNetConfig(**config)
        model = DeepRestNet(spec, cfg)
        model.load_state_dict(state["state_dict"])
        return model
