import sys
import time
import torch

torch.manual_seed(0)
dev = torch.device("cuda")
BT, K, N = 122880, 12672, 256
A = torch.randn(BT, K, device=dev, dtype=torch.bfloat16)
B = torch.randn(K, N, device=dev, dtype=torch.bfloat16)
dY = torch.randn(BT, N, device=dev, dtype=torch.bfloat16)

def timeit(fn, iters=20):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000

# forward: C = A @ B
C = torch.empty(BT, N, device=dev, dtype=torch.bfloat16)
print("fwd full     :", round(timeit(lambda: torch.mm(A, B, out=C)), 3), "ms")
Cf = torch.empty(BT, N, device=dev, dtype=torch.float32)
for nch in (4, 8, 16):
    ks = K // nch
    def chunked():
        Cf.zero_()
        for i in range(nch):
            Cf.addmm_(A[:, i*ks:(i+1)*ks], B[i*ks:(i+1)*ks], beta=1.0)
    print(f"fwd chunk{nch:3d} :", round(timeit(chunked), 3), "ms")

# dW = A^T @ dY  (the big-K gradient)
W = torch.empty(K, N, device=dev, dtype=torch.bfloat16)
print("dW full      :", round(timeit(lambda: torch.mm(A.t(), dY, out=W)), 3), "ms")
Wf = torch.empty(K, N, device=dev, dtype=torch.float32)
for nch in (4, 8, 16):
    bs = BT // nch
    def chunked_dw():
        Wf.zero_()
        for i in range(nch):
            Wf.addmm_(A[i*bs:(i+1)*bs].t(), dY[i*bs:(i+1)*bs], beta=1.0)
    print(f"dW chunk{nch:3d}  :", round(timeit(chunked_dw), 3), "ms")

# dX = dY @ B^T (N=K-wide output, K=256 small) — reference
X2 = torch.empty(BT, K, device=dev, dtype=torch.bfloat16)
print("dX full      :", round(timeit(lambda: torch.mm(dY, B.t(), out=X2)), 3), "ms")
