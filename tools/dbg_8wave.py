import numpy as np
import torch
from deeprest_amd.ops import fused_gru_sequence, reference_gru_sequence

dev = torch.device("cuda:0")
torch.manual_seed(8)
B, T, C, H = 400, 3, 64, 128
xg = torch.randn(B, T, 3 * H, device=dev) * 0.4
w_hh = torch.randn(3 * H, H, device=dev) / np.sqrt(H)
b_hh = torch.randn(3 * H, device=dev) * 0.1
h0 = torch.randn(B, C, H, device=dev) * 0.3
gamma = 1.0 + 0.1 * torch.randn(C, 3 * H, device=dev)
beta = 0.1 * torch.randn(C, 3 * H, device=dev)
args_t = [t.detach().clone().requires_grad_(True) for t in (xg, w_hh, b_hh, h0, gamma, beta)]
args_r = [t.detach().clone().requires_grad_(True) for t in (xg, w_hh, b_hh, h0, gamma, beta)]
out_t = fused_gru_sequence(*args_t)
out_r = reference_gru_sequence(*args_r)
d = (out_t.float() - out_r).abs()
print("fwd max err", d.max().item(), "mean", d.mean().item())
g = torch.randn_like(out_r)
out_t.backward(g)
out_r.backward(g)
for name, at_, ar_ in zip(["xg","w","b","h0","gam","bet"], args_t, args_r):
    e = (at_.grad.float() - ar_.grad.float()).abs()
    rel = e.max() / (ar_.grad.float().abs().max() + 1e-9)
    print(f"{name}: max abs {e.max().item():.4f} relmax {rel.item():.4f}")
