"""Decisive probes for the skinny-GEMM bandwidth ceiling.

1. torch streaming rate (sum) over the same weight buffer sizes.
2. skinny_gemm_packed time vs K at fixed N (fixed-cost extraction).
3. skinny at N=4096 with different split choices.
"""
import sys, time
import torch
sys.path.insert(0, ".")
import agentainer_amd.ops as O
from agentainer_amd.ops import pack_weight

def t(fn, n=100):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / n

print("== torch .sum streaming (same-buffer loop)")
for mb in [34, 235, 1026]:
    n = mb * 1024 * 1024 // 2
    w = torch.randn(n, dtype=torch.bfloat16, device="cuda")
    d = t(lambda: w.sum())
    print(f"  {mb:5d} MB: {d*1e6:8.1f} us  {n*2/d/1e12:5.2f} TB/s")

mod = O._load_hip()
print("== skinny fixed-cost sweep (N=4096, split=1, 64 blocks)")
for K in [512, 1024, 2048, 4096, 8192]:
    w = torch.randn(4096, K, dtype=torch.bfloat16, device="cuda")
    wp = pack_weight(w)
    x = torch.randn(64, K, dtype=torch.bfloat16, device="cuda")
    out = torch.empty(64, 4096, dtype=torch.bfloat16, device="cuda")
    ws = O._skinny_ws(x.device, 4096, 1)
    d = t(lambda: mod.skinny_gemm_packed(out, x, wp, 4096, K, ws, 1, False, 256))
    print(f"  K={K:6d}: {d*1e6:8.1f} us  {4096*K*2/d/1e12:5.2f} TB/s")

print("== skinny split sweep (N=4096, K=4096)")
w = torch.randn(4096, 4096, dtype=torch.bfloat16, device="cuda")
wp = pack_weight(w)
x = torch.randn(64, 4096, dtype=torch.bfloat16, device="cuda")
out = torch.empty(64, 4096, dtype=torch.bfloat16, device="cuda")
for split in [1, 2, 4, 8, 16]:
    if 4096 % (256 * split):
        continue
    ws = O._skinny_ws(x.device, 4096, split)
    d = t(lambda: mod.skinny_gemm_packed(out, x, wp, 4096, 4096, ws, split, False, 256))
    print(f"  split={split:2d} ({4096//64*split:4d} blocks): {d*1e6:8.1f} us  "
          f"{4096*4096*2/d/1e12:5.2f} TB/s")

print("== skinny multi-buffer (defeat L3), N=4096 K=4096 split=8")
ws = O._skinny_ws(x.device, 4096, 8)
wps = [pack_weight(torch.randn(4096, 4096, dtype=torch.bfloat16, device="cuda"))
       for _ in range(16)]
i = [0]
def rot():
    mod.skinny_gemm_packed(out, x, wps[i[0] % 16], 4096, 4096, ws, 8, False, 256)
    i[0] += 1
d = t(rot)
print(f"  rotating 16 buffers: {d*1e6:8.1f} us  {4096*4096*2/d/1e12:5.2f} TB/s")
