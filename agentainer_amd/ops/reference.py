"""Pure-PyTorch fp32 reference implementations of every HIP op.

Dual purpose (SURVEY.md §4 test strategy):
  * the numerics oracle for the GPU kernel tests (kernel output vs the
    fp32 reference of the same op, tolerance-gated);
  * the CPU execution path of the engine, so the full control plane +
    scheduler run (and are tested) on GPU-less CI.

Tensors use the same shapes/layouts as the HIP kernels, including the
paged-cache layouts k_cache [P, n_kv, D/8, PS, 8] / v_cache [P, n_kv, PS, D].
"""

from __future__ import annotations


import torch


def rmsnorm(out: torch.Tensor, x: torch.Tensor, w: torch.Tensor, eps: float) -> None:
    xf = x.float()
    r = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    out.copy_((r * w.float()).to(out.dtype))


def fused_add_rmsnorm(out: torch.Tensor, x: torch.Tensor, residual: torch.Tensor,
                      w: torch.Tensor, eps: float) -> None:
    s = (x.float() + residual.float())
    residual.copy_(s.to(residual.dtype))
    sf = residual.float()  # match kernel: normalizes the bf16-rounded sum
    r = sf * torch.rsqrt(sf.pow(2).mean(-1, keepdim=True) + eps)
    out.copy_((r * w.float()).to(out.dtype))


def silu_mul(out: torch.Tensor, gate: torch.Tensor, up: torch.Tensor) -> None:
    g = gate.float()
    out.copy_((g * torch.sigmoid(g) * up.float()).to(out.dtype))


def make_cos_sin_table(max_pos: int, dim: int, theta: float = 500000.0,
                       device="cpu") -> torch.Tensor:
    """[max_pos, dim] f32 rows laid out [cos(0..d/2) | sin(0..d/2)]."""
    inv = 1.0 / (theta ** (torch.arange(0, dim, 2, dtype=torch.float64) / dim))
    pos = torch.arange(max_pos, dtype=torch.float64)
    ang = torch.outer(pos, inv)
    tab = torch.cat([torch.cos(ang), torch.sin(ang)], dim=-1).float()
    return tab.contiguous().to(device)


def rope_inplace(q: torch.Tensor, k: torch.Tensor, cos_sin: torch.Tensor,
                 positions: torch.Tensor) -> None:
    D = q.size(-1)
    half = D // 2
    c = cos_sin[positions.long(), :half].unsqueeze(1).float()  # [T,1,half]
    s = cos_sin[positions.long(), half:].unsqueeze(1).float()
    for t in (q, k):
        x1 = t[..., :half].float()
        x2 = t[..., half:].float()
        t[..., :half] = (x1 * c - x2 * s).to(t.dtype)
        t[..., half:] = (x1 * s + x2 * c).to(t.dtype)


def kv_append(k_cache: torch.Tensor, v_cache: torch.Tensor, k: torch.Tensor,
              v: torch.Tensor, slot_mapping: torch.Tensor) -> None:
    n_kv, D = k.size(1), k.size(2)
    PS = k_cache.size(3)
    for t in range(k.size(0)):
        slot = int(slot_mapping[t])
        if slot < 0:
            continue
        page, off = slot // PS, slot % PS
        # k_cache [P, n_kv, D/8, PS, 8]; .to() casts bf16 -> fp8 when the
        # pool is e4m3 (index-assignment alone won't cast to float8)
        k_cache[page, :, :, off, :] = (
            k[t].reshape(n_kv, D // 8, 8).to(k_cache.dtype))
        v_cache[page, :, off, :] = v[t].to(v_cache.dtype)


def _gather_kv(k_cache, v_cache, page_table_row, length, g):
    """K/V [length, D] f32 for kv head g of one sequence."""
    PS = k_cache.size(3)
    D = v_cache.size(3)
    ks, vs = [], []
    for tok in range(length):
        page = int(page_table_row[tok // PS])
        off = tok % PS
        ks.append(k_cache[page, g, :, off, :].reshape(D).float())
        vs.append(v_cache[page, g, off, :].float())
    return torch.stack(ks), torch.stack(vs)


def paged_decode_attention(out: torch.Tensor, q: torch.Tensor,
                           k_cache: torch.Tensor, v_cache: torch.Tensor,
                           page_table: torch.Tensor, seq_lens: torch.Tensor,
                           scale: float) -> None:
    B, n_q, D = q.shape
    n_kv = k_cache.size(1)
    ratio = n_q // n_kv
    for b in range(B):
        length = int(seq_lens[b])
        if length <= 0:
            out[b] = 0  # graph pad row (dev_seq_lens == -1); HIP kernel skips
            continue
        for g in range(n_kv):
            K, V = _gather_kv(k_cache, v_cache, page_table[b], length, g)
            for qh in range(g * ratio, (g + 1) * ratio):
                s = (K @ q[b, qh].float()) * scale
                p = torch.softmax(s, dim=-1)
                out[b, qh] = (p @ V).to(out.dtype)


def paged_prefill_attention(out: torch.Tensor, q: torch.Tensor,
                            k_cache: torch.Tensor, v_cache: torch.Tensor,
                            page_table: torch.Tensor, seq_lens: torch.Tensor,
                            query_starts: torch.Tensor, query_lens: torch.Tensor,
                            scale: float) -> None:
    n_q = q.size(1)
    n_kv = k_cache.size(1)
    ratio = n_q // n_kv
    B = seq_lens.numel()
    for b in range(B):
        qlen = int(query_lens[b])
        if qlen == 0:
            continue
        total = int(seq_lens[b])
        start = int(query_starts[b])
        ctx_start = total - qlen
        for g in range(n_kv):
            K, V = _gather_kv(k_cache, v_cache, page_table[b], total, g)
            for qh in range(g * ratio, (g + 1) * ratio):
                Q = q[start:start + qlen, qh].float()       # [qlen, D]
                s = Q @ K.t() * scale                       # [qlen, total]
                qpos = ctx_start + torch.arange(qlen).unsqueeze(1)
                kpos = torch.arange(total).unsqueeze(0)
                s = s.masked_fill(kpos > qpos, float("-inf"))
                p = torch.softmax(s, dim=-1)
                out[start:start + qlen, qh] = (p @ V).to(out.dtype)


def greedy_sample(out: torch.Tensor, logits: torch.Tensor) -> None:
    out.copy_(logits.float().argmax(dim=-1))


def topp_sample(out: torch.Tensor, logits: torch.Tensor, temps: torch.Tensor,
                top_ps: torch.Tensor, seeds: torch.Tensor) -> None:
    """Reference nucleus sampling. Uses torch RNG seeded per row so results
    are deterministic given seeds (NOT bitwise-matched to the HIP kernel —
    distribution-level tests only)."""
    B, V = logits.shape
    for b in range(B):
        gen = torch.Generator(device="cpu").manual_seed(int(seeds[b]) & 0x7FFFFFFF)
        p = torch.softmax(logits[b].float() / max(float(temps[b]), 1e-6), dim=-1)
        sp, idx = p.sort(descending=True)
        cum = sp.cumsum(0)
        cut = int(torch.searchsorted(cum, float(top_ps[b])).item()) + 1
        cut = min(cut, V)
        sp = sp[:cut] / sp[:cut].sum()
        pick = idx[torch.multinomial(sp.cpu(), 1, generator=gen)]
        out[b] = pick.to(out.device)


def gather_kv_pages(dst: torch.Tensor, k_cache: torch.Tensor,
                    v_cache: torch.Tensor, page_ids: torch.Tensor) -> None:
    n_kv = k_cache.size(1)
    D = v_cache.size(3)
    PS = k_cache.size(3)
    plane = n_kv * D * PS
    flat = dst.view(-1)
    for i, p in enumerate(page_ids.tolist()):
        flat[i * 2 * plane:(i * 2 + 1) * plane] = k_cache[p].reshape(-1)
        flat[(i * 2 + 1) * plane:(i + 1) * 2 * plane] = v_cache[p].reshape(-1)


def scatter_kv_pages(k_cache: torch.Tensor, v_cache: torch.Tensor,
                     src: torch.Tensor, page_ids: torch.Tensor) -> None:
    n_kv = k_cache.size(1)
    D = v_cache.size(3)
    PS = k_cache.size(3)
    plane = n_kv * D * PS
    flat = src.view(-1)
    for i, p in enumerate(page_ids.tolist()):
        k_cache[p] = flat[i * 2 * plane:(i * 2 + 1) * plane].view_as(k_cache[p])
        v_cache[p] = flat[(i * 2 + 1) * plane:(i + 1) * 2 * plane].view_as(v_cache[p])
