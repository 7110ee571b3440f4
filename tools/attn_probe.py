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
dt = torch.float8_e4m3fn if (len(sys.argv) > 3 and sys.argv[3] == "fp8") \
    else torch.bfloat16
n_q, n_kv, D, PS = 32, 8, 128, 16
max_pages = -(-ctx // PS)
P = B * max_pages + 1
kc = torch.randn(P, n_kv, D // 8, PS, 8, device="cuda").to(dt)
vc = torch.randn(P, n_kv, PS, D, device="cuda").to(dt)
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
el = (time.time() - t0) / iters
kv_bytes = B * n_kv * ctx * 2 * D * kc.element_size()
print(f"ctx={ctx} B={B} {str(dt).split('.')[-1]}: {el*1e6:8.1f} us  "
      f"KV {kv_bytes/1e6:.1f} MB  {kv_bytes/el/1e12:5.2f} TB/s")
