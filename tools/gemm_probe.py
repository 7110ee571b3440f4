"""Decode-shape GEMM roofline probe on MI355X.

Times F.linear for llama3-8b decode shapes at M=64 and reports effective
weight-streaming TB/s. Run with/without:
  TORCH_BLAS_PREFER_HIPBLASLT=1
  PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1
"""

import sys
import time

import torch
import torch.nn.functional as F

M = int(sys.argv[1]) if len(sys.argv) > 1 else 64
shapes = [  # (in, out) llama3-8b decode projections + lm_head
    (4096, 6144), (4096, 4096), (4096, 28672), (14336, 4096), (4096, 128256),
]
dev = "cuda"
for K, N in shapes:
    w = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
    x = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
    for _ in range(10):
        F.linear(x, w)
    torch.cuda.synchronize()
    t0 = time.time()
    iters = 200
    for _ in range(iters):
        F.linear(x, w)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / iters
    gb = N * K * 2 / 1e9
    print(f"M={M} K={K:6d} N={N:6d}: {dt*1e6:8.1f} us  "
          f"{gb/dt/1000:6.2f} TB/s weights  "
          f"{2*M*N*K/dt/1e12:7.1f} TF")
