"""Chunked-GEMM probe for the in_proj shapes (BT=122880, K=12672, N=256).

The bench's in_proj GEMMs (F.linear TN layout) run at ~1.5 TB/s: the
6.3 MB weight panel is re-read by every M-tile and thrashes per-XCD L2.
K-chunked accumulation keeps each weight chunk L2-resident. Also checks
whether the plain NN layout (pre-transposed weight) is simply faster.
"""
import time

import torch
import torch.nn.functional as F

torch.manual_seed(0)
dev = torch.device("cuda")
BT, K, N = 122880, 12672, 256
A = torch.randn(BT, K, device=dev, dtype=torch.bfloat16)
W = torch.randn(N, K, device=dev, dtype=torch.bfloat16)     # nn.Linear layout
Wt = W.t().contiguous()                                     # (K, N) NN layout
dY = torch.randn(BT, N, device=dev, dtype=torch.bfloat16)


def timeit(fn, iters=20):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return round((time.perf_counter() - t0) / iters * 1000, 3)


print("fwd linear(A, W)  [TN, the model's path]:", timeit(lambda: F.linear(A, W)), "ms")
print("fwd mm(A, Wt)     [NN, pre-transposed]  :", timeit(lambda: torch.mm(A, Wt)), "ms")
for nch in (4, 8):
    ks = K // nch
    def chunked():
        acc = torch.mm(A[:, :ks], Wt[:ks])
        for i in range(1, nch):
            acc += torch.mm(A[:, i * ks:(i + 1) * ks], Wt[i * ks:(i + 1) * ks])
        return acc
    print(f"fwd NN chunked x{nch}                     :", timeit(chunked), "ms")

print("dW mm(dY.t, A)    [the model's dW]      :", timeit(lambda: torch.mm(dY.t(), A)), "ms")
print("dW mm(A.t, dY)    [transposed variant]  :", timeit(lambda: torch.mm(A.t(), dY)), "ms")
for nch in (4, 8):
    bs = BT // nch
    def chunked_dw():
        acc = torch.mm(dY[:bs].t(), A[:bs])
        for i in range(1, nch):
            acc += torch.mm(dY[i * bs:(i + 1) * bs].t(), A[i * bs:(i + 1) * bs])
        return acc
    print(f"dW chunked x{nch}                         :", timeit(chunked_dw), "ms")

print("dX mm(dY, W)      [the model's dX]      :", timeit(lambda: torch.mm(dY, W)), "ms")
