"""Per-shape A/B: packed skinny GEMM (nt off) vs hipBLASLt at M=64."""
import sys, time
import torch
import torch.nn.functional as F
sys.path.insert(0, ".")
import agentainer_amd.ops as O
from agentainer_amd.ops import pack_weight

def t(fn, n=200):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / n

mod = O._load_hip()
for K, N in [(4096, 6144), (4096, 4096), (4096, 28672), (14336, 4096),
             (4096, 128256)]:
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
    wp = pack_weight(w)
    x = torch.randn(64, K, dtype=torch.bfloat16, device="cuda")
    split = O._skinny_split(N // 64, K)
    ws = O._skinny_ws(x.device, N, split)
    out = torch.empty(64, N, dtype=torch.bfloat16, device="cuda")
    d_sk = t(lambda: mod.skinny_gemm_packed(out, x, wp, N, K, ws, split, False, 256))
    d_bl = t(lambda: F.linear(x, w))
    win = "skinny" if d_sk < d_bl else "blaslt"
    print(f"K={K:6d} N={N:6d} split={split}: skinny {d_sk*1e6:7.1f}us "
          f"({N*K*2/d_sk/1e12:.2f}TB/s)  blaslt {d_bl*1e6:7.1f}us "
          f"({N*K*2/d_bl/1e12:.2f}TB/s)  -> {win}")
