"""Reduced MXFP4 kernel diagnostic (full-fp4): unit vs real scales."""
import sys
import torch

sys.path.insert(0, ".")
import agentainer_amd.ops as O

mod = O._load_hip()
torch.manual_seed(1)
M, N, K = 16, 64, 256
GRID = torch.tensor([0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0,
                     -0.0, -0.5, -1.0, -1.5, -2.0, -3.0, -4.0, -6.0])
w = GRID[torch.randint(0, 16, (N, K))].to(torch.bfloat16).cuda()
w.view(N, K // 64, 64)[:, :, 0] = 6.0  # every block amax = 6 -> e8m0 127
wp, wsc = O.quantize_weight_mxfp4(w)
print("w scales all 127:", bool((wsc == 127).all().item()))
wd = O.dequantize_mxfp4(wp, wsc, N, K).cuda()
x = (torch.randn(M, K) * 0.5).to(torch.bfloat16).cuda()
x4 = torch.empty(M, K // 2, dtype=torch.uint8, device="cuda")
sx = torch.empty(M, dtype=torch.float32, device="cuda")
mod.quant_fp4_rows(x4, sx, x.contiguous())
b = x4.cpu()
codes = torch.stack([(b & 0xF).long(), (b >> 4).long()], -1).view(M, K)
xq = (GRID[codes % 16]) * sx.cpu()[:, None]
ref = xq.cuda() @ wd.float().t()
ws = O._skinny_ws(x.device, N, 1)
out = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
for combo in (0, 1):
    mod.skinny_gemm_mxfp4(out, x4, sx, wp, wsc, N, K, ws, 1, combo)
    torch.cuda.synchronize()
    rel = ((out.float() - ref).abs().max() / ref.abs().max()).item()
    print(f"unit-scale combo {combo}: rel {rel:.4f} "
          f"{'PASS' if rel < 2e-2 else ''}")
print("got ", out[0, :5].float().tolist())
print("want", ref[0, :5].tolist())
# real block scales: scale block 1 of every row by 4
w2 = w.clone()
w2.view(N, K // 64, 64)[:, 1, :] *= 4.0
wp2, wsc2 = O.quantize_weight_mxfp4(w2)
print("scale bytes:", torch.unique(wsc2).tolist())
wd2 = O.dequantize_mxfp4(wp2, wsc2, N, K).cuda()
ref2 = xq.cuda() @ wd2.float().t()
for combo in (0, 1):
    mod.skinny_gemm_mxfp4(out, x4, sx, wp2, wsc2, N, K, ws, 1, combo)
    torch.cuda.synchronize()
    rel = ((out.float() - ref2).abs().max() / ref2.abs().max()).item()
    print(f"block-scale combo {combo}: rel {rel:.4f} "
          f"{'PASS' if rel < 2e-2 else ''}")
