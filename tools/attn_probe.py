"""Isolated decode-attention probe: time + effective KV bandwidth.

python tools/attn_probe.py [ctx] [batch]
"""
import math
import sys
import time

import torch

sys.path.insert(0, ".")
from agentainer_amd import ops

ctx = int(sys.argv[1]) if len(sys.argv) > 1 else 320
B = int(sys.argv[2]) if len(sys.argv) > 2 else 64
n_q, n_kv, D, PS = 32, 8, 128, 16
max_pages = -(-ctx // PS)
P = B * max_pages + 1
kc = torch.zeros(P, n_kv, D // 8, PS, 8, dtype=torch.bfloat16, device="cuda")
vc = torch.zeros(P, n_kv, PS, D, dtype=torch.bfloat16, device="cuda")
kc.normal_()
vc.normal_()
pt = torch.arange(1, P, dtype=torch.int32, device="cuda").view(B, max_pages)
sl = torch.full((B,), ctx, dtype=torch.int32, device="cuda")
q = torch.randn(B, n_q, D, dtype=torch.bfloat16, device="cuda")
out = torch.empty_like(q)
scale = 1.0 / math.sqrt(D)
for _ in range(20):
    ops.paged_decode_attention(out, q, kc, vc, pt, sl, scale)
torch.cuda.synchronize()
t0 = time.time()
iters = 300
for _ in range(iters):
    ops.paged_decode_attention(out, q, kc, vc, pt, sl, scale)
torch.cuda.synchronize()
dt = (time.time() - t0) / iters
kv_bytes = B * n_kv * ctx * 2 * D * 2
print(f"ctx={ctx} B={B}: {dt*1e6:8.1f} us  KV {kv_bytes/1e6:.1f} MB  "
      f"{kv_bytes/dt/1e12:5.2f} TB/s")
