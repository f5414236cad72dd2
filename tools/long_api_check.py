import sys, time
import torch
sys.path.insert(0, ".")
from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig
from deeprest_amd.models.net import DeepRestNet, DeepRestNetConfig, build_model_spec

dev = torch.device("cuda")
app = SyntheticApp(SyntheticAppConfig(n_apis=256, n_components=63,
                                      windows_per_day=240, n_days=1, seed=7))
data = app.generate_featurized()
spec = build_model_spec(data)
torch.manual_seed(0)
for fp8 in (False, True):
    model = DeepRestNet(spec, DeepRestNetConfig(dropout=0.0,
                                                fp8_inference=fp8)).to(dev).eval()
    x = torch.rand(1, 122880, spec.num_paths, device=dev)
    with torch.no_grad():
        model.forward_long(x[:, :8192], chunk_size=4096)   # warmup
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        out = model.forward_long(x, chunk_size=4096)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
    print(f"fp8={fp8}: {122880/dt:.0f} steps/s through model.forward_long",
          flush=True)
