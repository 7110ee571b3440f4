"""ops — dispatch layer over the gfx950 HIP kernels.

Policy (the driver's "native code not loaded" check depends on this):
  * On a CUDA/ROCm device: the in-tree HIP extension (_hip_ops.so) is
    REQUIRED. If it is missing or fails to import, GPU ops raise
    RuntimeError — there is no silent eager fallback on a GPU box.
  * On CPU: the fp32 torch references (ops.reference) run, so the engine
    and control plane are testable without a GPU.

Build the extension with `python -m agentainer_amd.ops.build` (or via
__graft_entry__.build()).
"""

from __future__ import annotations

import os
from typing import Optional

import torch

from . import reference

_hip = None
_hip_err: Optional[str] = None


def _load_hip():
    global _hip, _hip_err
    if _hip is not None or _hip_err is not None:
        return _hip
    so = os.path.join(os.path.dirname(__file__), "_hip_ops.so")
    if not os.path.exists(so):
        _hip_err = f"HIP extension not built ({so} missing); run python -m agentainer_amd.ops.build"
        return None
    try:
        import importlib.util

        spec = importlib.util.spec_from_file_location("agentainer_amd.ops._hip_ops", so)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        _hip = mod
    except Exception as exc:  # noqa: BLE001
        _hip_err = f"HIP extension failed to load: {exc}"
    return _hip


def hip_available() -> bool:
    return _load_hip() is not None


def _dispatch(name: str, *tensors: torch.Tensor):
    """Return the HIP module if the leading tensor is on GPU, else None.
    On GPU with no extension: fail loudly (never a silent fallback)."""
    if tensors[0].is_cuda:
        mod = _load_hip()
        if mod is None:
            raise RuntimeError(
                f"op {name!r} on GPU requires the gfx950 HIP extension: {_hip_err}")
        return mod
    return None


def rmsnorm(out, x, w, eps: float = 1e-5):
    mod = _dispatch("rmsnorm", x)
    if mod:
        mod.rmsnorm(out, x, w, float(eps))
    else:
        reference.rmsnorm(out, x, w, eps)
    return out


def fused_add_rmsnorm(out, x, residual, w, eps: float = 1e-5):
    mod = _dispatch("fused_add_rmsnorm", x)
    if mod:
        mod.fused_add_rmsnorm(out, x, residual, w, float(eps))
    else:
        reference.fused_add_rmsnorm(out, x, residual, w, eps)
    return out


def silu_mul(out, gate, up):
    mod = _dispatch("silu_mul", gate)
    if mod:
        mod.silu_mul(out, gate, up)
    else:
        reference.silu_mul(out, gate, up)
    return out


make_cos_sin_table = reference.make_cos_sin_table


def rope_inplace(q, k, cos_sin, positions):
    mod = _dispatch("rope_inplace", q)
    if mod:
        mod.rope_inplace(q, k, cos_sin, positions)
    else:
        reference.rope_inplace(q, k, cos_sin, positions)


def rope_append(q, k, v, k_cache, v_cache, cos_sin, positions, slot_mapping):
    """Fused rope_inplace(q,k) + kv_append(k,v). CPU falls back to the two
    reference ops."""
    mod = _dispatch("rope_append", q)
    if mod:
        mod.rope_append(q, k, v, k_cache, v_cache, cos_sin, positions,
                        slot_mapping)
    else:
        reference.rope_inplace(q, k, cos_sin, positions)
        reference.kv_append(k_cache, v_cache, k, v, slot_mapping)


def kv_append(k_cache, v_cache, k, v, slot_mapping):
    mod = _dispatch("kv_append", k)
    if mod:
        mod.kv_append(k_cache, v_cache, k, v, slot_mapping)
    else:
        reference.kv_append(k_cache, v_cache, k, v, slot_mapping)


def paged_decode_attention(out, q, k_cache, v_cache, page_table, seq_lens,
                           scale: float):
    mod = _dispatch("paged_decode_attention", q)
    if mod:
        mod.paged_decode_attention(out, q, k_cache, v_cache, page_table,
                                   seq_lens, float(scale))
    else:
        reference.paged_decode_attention(out, q, k_cache, v_cache, page_table,
                                         seq_lens, scale)
    return out


def paged_prefill_attention(out, q, k_cache, v_cache, page_table, seq_lens,
                            query_starts, query_lens, scale: float,
                            max_qlen: int = 0):
    mod = _dispatch("paged_prefill_attention", q)
    if mod:
        mod.paged_prefill_attention(out, q, k_cache, v_cache, page_table,
                                    seq_lens, query_starts, query_lens,
                                    float(scale), int(max_qlen))
    else:
        reference.paged_prefill_attention(out, q, k_cache, v_cache, page_table,
                                          seq_lens, query_starts, query_lens,
                                          scale)
    return out


def greedy_sample(out, logits):
    mod = _dispatch("greedy_sample", logits)
    if mod:
        mod.greedy_sample(out, logits)
    else:
        reference.greedy_sample(out, logits)
    return out


def topp_sample(out, logits, temps, top_ps, seeds):
    mod = _dispatch("topp_sample", logits)
    if mod:
        mod.topp_sample(out, logits, temps, top_ps, seeds)
    else:
        reference.topp_sample(out, logits, temps, top_ps, seeds)
    return out


_ws_cache = {}
_EMPTY_WS = {}
SKINNY_DISABLED = os.environ.get("AGENTAINER_DISABLE_SKINNY", "0") == "1"
SKINNY_FORCED = os.environ.get("AGENTAINER_FORCE_SKINNY", "0") == "1"
# shapes where the hand-written skinny GEMM beat hipBLASLt on MI355X,
# with the (split_k, k_chunk) that won the tools/sg_sweep.py grid
# (M=64; see profiles/README.md). Library keeps qkv/gate_up: at
# N>=6144 its MT256 tiles already stream near the roofline.
_SKINNY_SHAPES = {
    (4096, 4096): (4, 128),    # o-proj: 14.2 us vs lib 18.6
    (4096, 14336): (8, 128),   # down-proj: 30.3 us vs lib 34.4
}
SKINNY_NT = os.environ.get("AGENTAINER_SKINNY_NT", "0") == "1"
SKINNY_KC = int(os.environ.get("AGENTAINER_SKINNY_KC", "0"))  # 0 = tuned
# A/B hook: extra tuned shapes as "N:K:split:kc,..." (perf experiments)
for _spec in os.environ.get("AGENTAINER_SKINNY_SHAPES", "").split(","):
    if _spec.strip():
        _n, _k, _s, _c = (int(v) for v in _spec.split(":"))
        _SKINNY_SHAPES[(_n, _k)] = (_s, _c)


def _skinny_split(ntiles: int, K: int) -> int:
    # target ~256 blocks (1 per CU — measured optimum: 2 blocks/CU halve
    # the per-block LDS and contend on the chunk barriers)
    split = 1
    while split < 16 and ntiles * split < 256 and K % (256 * split * 2) == 0:
        split *= 2
    return split


def pack_weight(w: torch.Tensor) -> torch.Tensor:
    """Shuffle a static [N, K] weight into MFMA-fragment-linear layout
    packed[n_tile][k_chunk][lane][8] so the skinny GEMM's weight stream is
    contiguous per wave. Done once at model load; flat [N*K] result."""
    N, K = w.shape
    assert N % 16 == 0 and K % 32 == 0
    p = w.view(N // 16, 16, K // 32, 4, 8).permute(0, 2, 3, 1, 4).contiguous()
    return p.view(-1)


def _skinny_ws(device, N: int, split: int) -> torch.Tensor:
    if split > 1:
        key = (device, N, split)
        ws = _ws_cache.get(key)
        if ws is None or ws.numel() < split * 64 * N:
            ws = torch.empty(split * 64 * N, dtype=torch.float32, device=device)
            _ws_cache[key] = ws
        return ws
    ws = _EMPTY_WS.get(device)
    if ws is None:
        ws = torch.empty(1, dtype=torch.float32, device=device)
        _EMPTY_WS[device] = ws
    return ws


def linear(x: torch.Tensor, w: torch.Tensor,
           w_packed: Optional[torch.Tensor] = None) -> torch.Tensor:
    """x[M,K] @ w[N,K]^T. Decode-shaped (M<=64) GPU GEMMs go to the
    hand-written skinny MFMA kernel — packed-weight variant when the
    caller pre-packed W (streams at the HBM roofline); everything else to
    hipBLASLt via F.linear."""
    M, K = x.shape
    N = w.size(0)
    tuned = _SKINNY_SHAPES.get((N, K))
    if (x.is_cuda and M <= 64 and N % 64 == 0 and K % 256 == 0
            and (tuned is not None or SKINNY_FORCED) and not SKINNY_DISABLED):
        mod = _dispatch("skinny_gemm", x)
        split, kc = tuned if tuned else (_skinny_split(N // 64, K), 128)
        if SKINNY_KC:
            kc = SKINNY_KC
        if K % (kc * split):
            split = _skinny_split(N // 64, K)
        out = torch.empty(M, N, dtype=x.dtype, device=x.device)
        ws = _skinny_ws(x.device, N, split)
        if w_packed is not None:
            mod.skinny_gemm_packed(out, x.contiguous(), w_packed, N, K, ws,
                                   split, SKINNY_NT, kc)
        else:
            mod.skinny_gemm(out, x.contiguous(), w, ws, split)
        return out
    return torch.nn.functional.linear(x, w)


def quantize_weight_fp8(w: torch.Tensor):
    """Offline per-output-channel OCP e4m3 quantization of a [N, K] weight:
    returns (fragment-linear packed bytes, f32 scales[N]). Serves the fp8
    MFMA expert decode GEMM (BASELINE config 5)."""
    N, K = w.shape
    sw = w.float().abs().amax(dim=1).clamp(min=1e-8) / 448.0
    scaled = (w.float() / sw[:, None]).clamp(-448.0, 448.0)
    try:
        w8 = scaled.to(torch.float8_e4m3fn).view(torch.uint8)
    except (RuntimeError, TypeError):
        w8 = scaled.cpu().to(torch.float8_e4m3fn).view(torch.uint8).to(w.device)
    packed = (w8.view(N // 16, 16, K // 32, 4, 8)
              .permute(0, 2, 3, 1, 4).contiguous().view(-1))
    return packed, sw.float().contiguous()


def linear_fp8(x: torch.Tensor, w_packed: torch.Tensor, sw: torch.Tensor,
               N: int) -> torch.Tensor:
    """x[M,K] bf16 @ fp8-quantized W^T: dynamic per-token activation quant
    then the fp8 MFMA skinny GEMM (half the weight bytes of bf16)."""
    M, K = x.shape
    mod = _dispatch("skinny_gemm_fp8", x)
    assert mod is not None, "linear_fp8 is GPU-only"
    xc = x.contiguous()
    x8 = torch.empty(M, K, dtype=torch.uint8, device=x.device)
    sx = torch.empty(M, dtype=torch.float32, device=x.device)
    mod.quant_fp8_rows(x8, sx, xc)
    split = _skinny_split(N // 64, K)
    ws = _skinny_ws(x.device, N, split)
    out = torch.empty(M, N, dtype=x.dtype, device=x.device)
    mod.skinny_gemm_fp8(out, x8, sx, w_packed, sw, N, K, ws, split)
    return out


# e2m1 (fp4) code values, index = code (bit 3 = sign)
_FP4_GRID = torch.tensor([0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0])
# operand-role combo for the scaled MFMA (validated on hardware via
# tests/test_gpu_ops.py::test_mxfp4_gemm; override for probing only)
MXFP4_COMBO = int(os.environ.get("AGENTAINER_MXFP4_COMBO", "0"))


def quantize_weight_mxfp4(w: torch.Tensor):
    """Offline MXFP4 quantization of a [N, K] weight for the block-scaled
    MFMA expert GEMM (gfx950 v_mfma_scale_f32_16x16x128_f8f6f4): e2m1
    codes, 2 per byte, with one e8m0 scale per (row, 64-element k-block)
    — the scale granularity the instruction's scale lanes expose
    (tools/mx_probe.py). Returns (packed codes [N*K/2] u8 fragment-linear,
    scales [N/16 * K/128 * 32] u8 e8m0, kernel-ordered)."""
    N, K = w.shape
    assert N % 16 == 0 and K % 128 == 0, "MXFP4: N%16==0, K%128==0"
    wf = w.float().view(N, K // 32, 32)          # true MX block-32
    amax = wf.abs().amax(dim=-1).clamp(min=1e-8)
    # e8m0 scale 2^e with amax/2^e <= 6 (e2m1 max)
    e = torch.ceil(torch.log2(amax / 6.0)).clamp(-127, 127)
    scale = torch.pow(2.0, e)
    q = wf / scale.unsqueeze(-1)
    # nearest e2m1 via bucketize on the midpoints (device-native: this
    # runs at model load for up to 512 Mixtral expert matrices)
    mids = torch.tensor([0.25, 0.75, 1.25, 1.75, 2.5, 3.5, 5.0],
                        device=wf.device)
    idx = torch.bucketize(q.abs().contiguous(), mids)
    codes = (idx + torch.where(q < 0, 8, 0)).to(torch.uint8).view(N, K)
    # pack 2 codes/byte along k (lo nibble = even k)
    by = (codes[:, 0::2] | (codes[:, 1::2] << 4))  # [N, K/2]
    # fragment-linear: B lane map (probed):
    # B[(l/16)*32 + j][l%16] -> W[tile*16 + l%16][kc*128 + (l/16)*32 + j],
    # 32 codes = 16 bytes per lane
    byv = by.view(N // 16, 16, K // 128, 4, 16)     # [T][col][kc][q][16B]
    packed = byv.permute(0, 2, 3, 1, 4).contiguous().view(-1)
    # scales: one e8m0 per (col, 32-block); scale lane l covers
    # (kb = l/16, col = l%16) -> scv[T][kc][kb][col], 64 B per fragment
    e8 = (e + 127).clamp(0, 254).to(torch.uint8).view(N, K // 32)
    scv = e8.view(N // 16, 16, K // 128, 4).permute(0, 2, 3, 1).contiguous()
    return packed.to(w.device), scv.view(-1).to(w.device)


def dequantize_mxfp4(packed: torch.Tensor, scales: torch.Tensor,
                     N: int, K: int) -> torch.Tensor:
    """CPU reference inverse of quantize_weight_mxfp4 (tests/oracles)."""
    p = packed.cpu().view(N // 16, K // 128, 4, 16, 16)
    by = p.permute(0, 3, 1, 2, 4).contiguous().view(N, K // 2)
    lo = (by & 0xF).long()
    hi = (by >> 4).long()
    codes = torch.stack([lo, hi], dim=-1).view(N, K)
    vals = _FP4_GRID[codes & 7] * torch.where(codes >= 8, -1.0, 1.0)
    sc = scales.cpu().view(N // 16, K // 128, 4, 16).permute(0, 3, 1, 2)
    sc = sc.contiguous().view(N, K // 32).float()
    sc = torch.pow(2.0, sc - 127)
    return (vals.view(N, K // 32, 32) * sc.unsqueeze(-1)).view(N, K)


def linear_mxfp4(x: torch.Tensor, w_packed: torch.Tensor,
                 w_scales: torch.Tensor, N: int) -> torch.Tensor:
    """x[M,K] bf16 @ MXFP4 W^T: per-row e2m1 activation quant + the
    block-scaled 16x16x128 fp4 MFMA — quarter the weight bytes of bf16.
    (Mixed fp8-activation x fp4-weight through the scaled MFMA NaNs on
    this silicon/compiler — tools/mx_probe.py — so both operands run
    e2m1; activations carry a per-row f32 scale applied in the
    epilogue.)"""
    M, K = x.shape
    mod = _dispatch("skinny_gemm_mxfp4", x)
    assert mod is not None, "linear_mxfp4 is GPU-only"
    xc = x.contiguous()
    x8 = torch.empty(M, K // 2, dtype=torch.uint8, device=x.device)
    sx = torch.empty(M, dtype=torch.float32, device=x.device)
    mod.quant_fp4_rows(x8, sx, xc)
    split = _skinny_split(N // 64, K)
    ws = _skinny_ws(x.device, N, split)
    out = torch.empty(M, N, dtype=x.dtype, device=x.device)
    mod.skinny_gemm_mxfp4(out, x8, sx, w_packed, w_scales, N, K, ws, split,
                          MXFP4_COMBO)
    return out


def linear_quant(x: torch.Tensor, qt) -> torch.Tensor:
    """Dispatch a quantized decode GEMM: qt = (kind, packed, scales, N)
    with kind "mxfp4" or "fp8" (engine.dense_quant / expert paths)."""
    kind, packed, scales, N = qt
    if kind == "mxfp4":
        return linear_mxfp4(x, packed, scales, N)
    return linear_fp8(x, packed, scales, N)


def gather_kv_pages(dst, k_cache, v_cache, page_ids):
    mod = _dispatch("gather_kv_pages", k_cache)
    if mod:
        mod.gather_kv_pages(dst, k_cache, v_cache, page_ids)
    else:
        reference.gather_kv_pages(dst, k_cache, v_cache, page_ids)


def scatter_kv_pages(k_cache, v_cache, src, page_ids):
    mod = _dispatch("scatter_kv_pages", k_cache)
    if mod:
        mod.scatter_kv_pages(k_cache, v_cache, src, page_ids)
    else:
        reference.scatter_kv_pages(k_cache, v_cache, src, page_ids)
