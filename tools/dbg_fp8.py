import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import numpy as np
import torch
from deeprest_amd.ops import fused_gru_sequence

def ref_fp8_sim(xg, w_hh, b_hh, h0):
    """reference GRU with W and h quantized to e4m3 before each GEMM."""
    q = lambda t: t.to(torch.float8_e4m3fn).float()
    B, T, G = xg.shape
    _, C, H = h0.shape
    wq = q(w_hh).t()
    h = h0
    outs = []
    for t in range(T):
        g = xg[:, t, None, :].expand(B, C, G)
        hh = q(h) @ wq + b_hh
        r = torch.sigmoid(g[..., :H] + hh[..., :H])
        z = torch.sigmoid(g[..., H:2*H] + hh[..., H:2*H])
        n = torch.tanh(g[..., 2*H:] + r * hh[..., 2*H:])
        h = (1 - z) * n + z * h
        outs.append(h)
    return torch.stack(outs, 1)

dev = torch.device("cuda:0")
torch.manual_seed(10)
B, T, C, H = 3, 12, 6, 128
xg = torch.randn(B, T, 3*H, device=dev) * 0.4
w_hh = torch.randn(3*H, H, device=dev) / np.sqrt(H)
b_hh = torch.randn(3*H, device=dev) * 0.1
h0 = torch.randn(B, C, H, device=dev) * 0.3
with torch.no_grad():
    out8 = fused_gru_sequence(xg, w_hh, b_hh, h0, fp8=True)
    refq = ref_fp8_sim(xg, w_hh, b_hh, h0)
e = (out8.float() - refq).abs()
print("kernel-vs-fp8sim max", e.max().item(), "mean", e.mean().item())
# single step diagnosis
with torch.no_grad():
    o1 = fused_gru_sequence(xg[:, :1]*0, w_hh, torch.zeros_like(b_hh), h0, fp8=True)
    r1 = ref_fp8_sim(xg[:, :1]*0, w_hh, torch.zeros_like(b_hh), h0)
e1 = (o1.float() - r1).abs()
print("1-step max", e1.max().item(), "mean", e1.mean().item())
