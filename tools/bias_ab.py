import sys
import torch
sys.path.insert(0, ".")
import deeprest_amd.models.net as netmod
from bench import accuracy_probe

dev = torch.device("cuda")
orig = netmod.DeepRestNetConfig
for lb in (True, False):
    class Patched(orig):  # probe builds DeepRestNetConfig(dropout=0.1)
        def __init__(self, **kw):
            kw.setdefault("linear_bias", lb)
            super().__init__(**kw)
    netmod.DeepRestNetConfig = Patched
    import bench
    bench.DeepRestNetConfig = Patched
    acc = accuracy_probe(50, dev)
    print(f"linear_bias={lb}: {acc['mean_median_abs_err']} "
          f"beats_comp={acc['deepr_beats_comp']} ext={acc['extended']}",
          flush=True)
    netmod.DeepRestNetConfig = orig
