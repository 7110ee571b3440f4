import sys, torch
sys.path.insert(0, ".")
import agentainer_amd.ops as O
from agentainer_amd.ops import pack_weight
mod = O._load_hip()
K, N, split = 4096, 4096, 4
w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
wp = pack_weight(w)
x = torch.randn(64, K, dtype=torch.bfloat16, device="cuda")
out = torch.empty(64, N, dtype=torch.bfloat16, device="cuda")
ws = O._skinny_ws(x.device, N, split)
for _ in range(30):
    mod.skinny_gemm_packed(out, x, wp, N, K, ws, split, False, 256)
torch.cuda.synchronize()
