"""CPU reference-op sanity: shapes, invariants, paged-cache round trips.

(The GPU kernels are validated against these references in test_gpu_ops.py.)
"""

import math

import pytest
import torch

from agentainer_amd import ops
from agentainer_amd.ops import reference as ref


def test_rmsnorm_matches_formula():
    x = torch.randn(4, 64, dtype=torch.bfloat16)
    w = torch.randn(64, dtype=torch.bfloat16)
    out = torch.empty_like(x)
    ops.rmsnorm(out, x, w, 1e-5)
    xf = x.float()
    want = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-5) * w.float()
    assert torch.allclose(out.float(), want, atol=0.02, rtol=0.02)


def test_fused_add_rmsnorm_updates_residual():
    x = torch.randn(4, 64, dtype=torch.bfloat16)
    res = torch.randn(4, 64, dtype=torch.bfloat16)
    w = torch.ones(64, dtype=torch.bfloat16)
    want_res = (x.float() + res.float()).to(torch.bfloat16)
    out = torch.empty_like(x)
    ops.fused_add_rmsnorm(out, x, res, w, 1e-5)
    assert torch.equal(res, want_res)


def test_silu_mul():
    g = torch.randn(8, 16, dtype=torch.bfloat16)
    u = torch.randn(8, 16, dtype=torch.bfloat16)
    out = torch.empty_like(g)
    ops.silu_mul(out, g, u)
    want = torch.nn.functional.silu(g.float()) * u.float()
    assert torch.allclose(out.float(), want, atol=0.02, rtol=0.05)


def test_rope_preserves_norm_and_position0():
    T, nq, nkv, D = 3, 2, 1, 32
    q = torch.randn(T, nq, D, dtype=torch.bfloat16)
    k = torch.randn(T, nkv, D, dtype=torch.bfloat16)
    q0, k0 = q.clone(), k.clone()
    tab = ops.make_cos_sin_table(16, D)
    pos = torch.tensor([0, 5, 7], dtype=torch.int32)
    ops.rope_inplace(q, k, tab, pos)
    # position 0: rotation is identity
    assert torch.allclose(q[0].float(), q0[0].float(), atol=0.02)
    # rotation preserves pair norms
    for t in range(T):
        n_before = q0[t, 0].float().view(2, D // 2).pow(2).sum(0)
        n_after = q[t, 0].float().view(2, D // 2).pow(2).sum(0)
        assert torch.allclose(n_before, n_after, atol=0.05, rtol=0.05)


def _mk_caches(P=8, n_kv=2, D=16, PS=4):
    k_cache = torch.zeros(P, n_kv, D // 8, PS, 8, dtype=torch.bfloat16)
    v_cache = torch.zeros(P, n_kv, PS, D, dtype=torch.bfloat16)
    return k_cache, v_cache


def test_kv_append_and_layouts():
    k_cache, v_cache = _mk_caches()
    T, n_kv, D, PS = 3, 2, 16, 4
    k = torch.randn(T, n_kv, D, dtype=torch.bfloat16)
    v = torch.randn(T, n_kv, D, dtype=torch.bfloat16)
    slots = torch.tensor([0, 1, 5], dtype=torch.long)  # page0 off0/1, page1 off1
    ops.kv_append(k_cache, v_cache, k, v, slots)
    assert torch.equal(k_cache[0, :, :, 0, :].reshape(n_kv, D), k[0])
    assert torch.equal(v_cache[0, :, 1, :], v[1])
    assert torch.equal(k_cache[1, :, :, 1, :].reshape(n_kv, D), k[2])


def test_decode_attention_vs_dense():
    torch.manual_seed(0)
    B, n_q, n_kv, D, PS = 2, 4, 2, 16, 4
    P = 8
    k_cache, v_cache = _mk_caches(P, n_kv, D, PS)
    lens = [6, 3]
    page_table = torch.tensor([[0, 1], [2, 0]], dtype=torch.int32)
    Ks = {}, {}
    K_all = torch.randn(B, max(lens), n_kv, D, dtype=torch.bfloat16)
    V_all = torch.randn(B, max(lens), n_kv, D, dtype=torch.bfloat16)
    for b in range(B):
        for t in range(lens[b]):
            page = int(page_table[b][t // PS])
            ops.kv_append(k_cache, v_cache, K_all[b, t:t + 1],
                          V_all[b, t:t + 1],
                          torch.tensor([page * PS + t % PS], dtype=torch.long))
    q = torch.randn(B, n_q, D, dtype=torch.bfloat16)
    out = torch.empty_like(q)
    scale = 1.0 / math.sqrt(D)
    ops.paged_decode_attention(out, q, k_cache, v_cache, page_table,
                               torch.tensor(lens, dtype=torch.int32), scale)
    # dense oracle
    ratio = n_q // n_kv
    for b in range(B):
        for h in range(n_q):
            g = h // ratio
            K = K_all[b, :lens[b], g].float()
            V = V_all[b, :lens[b], g].float()
            s = torch.softmax(K @ q[b, h].float() * scale, dim=-1)
            want = s @ V
            assert torch.allclose(out[b, h].float(), want, atol=0.03, rtol=0.05)


def test_prefill_attention_vs_dense_causal():
    torch.manual_seed(1)
    n_q, n_kv, D, PS = 4, 2, 16, 4
    k_cache, v_cache = _mk_caches(16, n_kv, D, PS)
    # seq with 3 cached tokens + 4 new tokens
    ctx, new = 3, 4
    total = ctx + new
    page_table = torch.tensor([[0, 1]], dtype=torch.int32)
    K_all = torch.randn(total, n_kv, D, dtype=torch.bfloat16)
    V_all = torch.randn(total, n_kv, D, dtype=torch.bfloat16)
    for t in range(total):
        page = int(page_table[0][t // PS])
        ops.kv_append(k_cache, v_cache, K_all[t:t + 1], V_all[t:t + 1],
                      torch.tensor([page * PS + t % PS], dtype=torch.long))
    q = torch.randn(new, n_q, D, dtype=torch.bfloat16)
    out = torch.empty_like(q)
    scale = 1.0 / math.sqrt(D)
    ops.paged_prefill_attention(
        out, q, k_cache, v_cache, page_table,
        torch.tensor([total], dtype=torch.int32),
        torch.tensor([0], dtype=torch.int32),
        torch.tensor([new], dtype=torch.int32), scale)
    ratio = n_q // n_kv
    for h in range(n_q):
        g = h // ratio
        K = K_all[:, g].float()
        V = V_all[:, g].float()
        s = q[:, h].float() @ K.t() * scale  # [new, total]
        for i in range(new):
            s[i, ctx + i + 1:] = float("-inf")
        p = torch.softmax(s, dim=-1)
        want = p @ V
        assert torch.allclose(out[:, h].float(), want, atol=0.03, rtol=0.05)


def test_greedy_sample():
    logits = torch.randn(4, 100, dtype=torch.bfloat16)
    out = torch.empty(4, dtype=torch.long)
    ops.greedy_sample(out, logits)
    assert torch.equal(out, logits.float().argmax(-1))


def test_topp_sample_reference_properties():
    torch.manual_seed(2)
    logits = torch.randn(8, 64, dtype=torch.bfloat16)
    logits[:, 7] += 8.0  # dominant token
    out = torch.empty(8, dtype=torch.long)
    temps = torch.full((8,), 0.7)
    tps = torch.full((8,), 0.1)  # tight nucleus -> dominant token always
    seeds = torch.arange(8, dtype=torch.int64)
    ops.topp_sample(out, logits, temps, tps, seeds)
    assert (out == 7).all()


def test_gather_scatter_pages_roundtrip():
    k_cache, v_cache = _mk_caches()
    k_cache.normal_()
    v_cache.normal_()
    n_kv, D, PS = 2, 16, 4
    plane = n_kv * D * PS
    ids = torch.tensor([1, 3], dtype=torch.int32)
    buf = torch.zeros(2 * 2 * plane, dtype=torch.bfloat16)
    ops.gather_kv_pages(buf, k_cache, v_cache, ids)
    k2, v2 = _mk_caches()
    ops.scatter_kv_pages(k2, v2, buf, ids)
    assert torch.equal(k2[1], k_cache[1]) and torch.equal(v2[3], v_cache[3])
    assert k2[0].abs().sum() == 0


def test_mxfp4_pack_roundtrip():
    """quantize_weight_mxfp4 <-> dequantize_mxfp4: every reconstructed
    value is within the e2m1 grid resolution of its 32-block's scale."""
    torch.manual_seed(3)
    N, K = 64, 512
    w = (torch.randn(N, K) * 0.07).to(torch.bfloat16)
    packed, scales = ops.quantize_weight_mxfp4(w)
    assert packed.numel() == N * K // 2
    assert scales.numel() == (N // 16) * (K // 128) * 64
    wd = ops.dequantize_mxfp4(packed, scales, N, K)
    blocks = w.float().view(N, K // 32, 32)
    err = (wd.view(N, K // 32, 32) - blocks).abs().amax(-1)
    bound = blocks.abs().amax(-1).clamp(min=1e-8) * 0.26 + 1e-6
    assert bool((err <= bound).all()), float((err / bound).max())
    # exactly representable values roundtrip exactly
    grid = torch.tensor([0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0, -6.0, -0.5])
    w2 = grid[torch.randint(0, 9, (16, 128))].to(torch.bfloat16)
    w2.view(16, 4, 32)[:, :, 0] = 6.0  # pin block scale to 2^0
    p2, s2 = ops.quantize_weight_mxfp4(w2)
    assert torch.equal(ops.dequantize_mxfp4(p2, s2, 16, 128), w2.float())


def test_linear_quant_dispatch_kinds():
    """linear_quant routes by kind tag; CPU tensors raise loudly (these
    paths are GPU-only by policy — no silent eager fallback)."""
    x = torch.randn(4, 256, dtype=torch.bfloat16)
    with pytest.raises((AssertionError, RuntimeError)):
        ops.linear_quant(x, ("mxfp4", torch.zeros(1, dtype=torch.uint8),
                             torch.zeros(1, dtype=torch.uint8), 64))
