"""deeprest_amd — MI355X-native per-endpoint resource estimation framework.

A from-scratch rebuild of the capabilities of IBM/DeepRest (EuroSys'22)
designed for AMD Instinct MI355X (gfx950, CDNA4):

- ``data``:     ingestion contract (Jaeger-style span trees + Prometheus-style
                metrics), call-path featurizer, trace synthesizer, synthetic
                application generator.
- ``models``:   the estimation engine (attention traffic encoder, call-graph
                propagation, per-resource GRU decoders with quantile heads)
                plus the three comparison baselines (history-only ANN,
                component-aware linear scaling, trace-level ridge).
- ``ops``:      hand-written CDNA4 HIP kernels (fused GRU cell, MHA over the
                endpoint x time window, LayerNorm, pinball loss, fused Adam)
                with PyTorch autograd bindings; CPU fallbacks for testing.
- ``parallel``: single-node data parallelism over RCCL/xGMI with a fused
                gradient bucket.
- ``engine``:   training loop with the three-estimator comparison harness,
                checkpoint/resume, typed config.
- ``serve``:    hipGraph-captured batched predictor, results writer
                (web-demo compatible schema), REST API, anomaly scorer.

Reference parity is tracked against /root/reference (IBM/DeepRest); citations
in docstrings use ``reference:<path>:<line>`` form.
"""

__version__ = "0.1.0"
