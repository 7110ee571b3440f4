"""Skinny packed GEMM sweep: shapes x KC x split, TB/s + correctness.

python tools/sg_sweep.py [M]
"""
import sys
import time

import torch

sys.path.insert(0, ".")
import agentainer_amd.ops as O
from agentainer_amd.ops import pack_weight

mod = O._load_hip()
M = int(sys.argv[1]) if len(sys.argv) > 1 else 64
SHAPES = [(6144, 4096), (4096, 4096), (28672, 4096), (4096, 14336)]

for N, K in SHAPES:
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.05
    wp = pack_weight(w)
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    ref = torch.nn.functional.linear(x.float(), w.float())
    # library baseline
    for _ in range(20):
        torch.nn.functional.linear(x, w)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(200):
        torch.nn.functional.linear(x, w)
    torch.cuda.synchronize()
    dlib = (time.time() - t0) / 200
    gb = N * K * 2 / 1e9
    print(f"N={N:6d} K={K:6d} lib: {dlib*1e6:7.1f} us {gb/dlib/1000:5.2f} TB/s")
    for kc in (128, 256):
        for split in (1, 2, 4, 8, 16):
            if K % (kc * split) or (N // 64) * split > 4096:
                continue
            ws = O._skinny_ws(x.device, N, split)
            out = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
            try:
                mod.skinny_gemm_packed(out, x, wp, N, K, ws, split, False, kc)
            except RuntimeError as e:
                print(f"  kc={kc} split={split:2d}: launch error {e}")
                continue
            torch.cuda.synchronize()
            err = (out.float() - ref).abs().max().item()
            rel = err / ref.abs().max().item()
            for _ in range(20):
                mod.skinny_gemm_packed(out, x, wp, N, K, ws, split, False, kc)
            torch.cuda.synchronize()
            t0 = time.time()
            for _ in range(200):
                mod.skinny_gemm_packed(out, x, wp, N, K, ws, split, False, kc)
            torch.cuda.synchronize()
            d = (time.time() - t0) / 200
            flag = "BAD" if rel > 2e-2 else "ok"
            print(f"  kc={kc} split={split:2d}: {d*1e6:7.1f} us "
                  f"{gb/d/1000:5.2f} TB/s  rel={rel:.1e} {flag}")
