"""GPU kernel numerics: every gfx950 HIP kernel vs the plain-PyTorch fp32
reference of the same op (SURVEY.md §4 implication (3)). All @pytest.mark.gpu."""

import math

import pytest
import torch

from agentainer_amd import ops
from agentainer_amd.ops import reference as ref

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)


@pytest.fixture(autouse=True, scope="module")
def _require_hip():
    assert ops.hip_available(), "HIP extension must be present on the GPU box"


DEV = "cuda"


def test_rmsnorm_matches_reference():
    torch.manual_seed(0)
    for T, H in [(1, 4096), (33, 4096), (256, 8192), (7, 512)]:
        x = torch.randn(T, H, dtype=torch.bfloat16, device=DEV)
        w = torch.randn(H, dtype=torch.bfloat16, device=DEV)
        out = torch.empty_like(x)
        ops.rmsnorm(out, x, w, 1e-5)
        want = torch.empty_like(x, device="cpu")
        ref.rmsnorm(want, x.cpu(), w.cpu(), 1e-5)
        assert torch.allclose(out.float().cpu(), want.float(), atol=0.02, rtol=0.02), \
            f"rmsnorm mismatch T={T} H={H}"


def test_fused_add_rmsnorm_matches_reference():
    torch.manual_seed(1)
    T, H = 64, 4096
    x = torch.randn(T, H, dtype=torch.bfloat16, device=DEV)
    res = torch.randn(T, H, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(H, dtype=torch.bfloat16, device=DEV)
    x_c, res_c = x.cpu().clone(), res.cpu().clone()
    out = torch.empty_like(x)
    ops.fused_add_rmsnorm(out, x, res, w, 1e-5)
    want = torch.empty_like(x_c)
    ref.fused_add_rmsnorm(want, x_c, res_c, w.cpu(), 1e-5)
    assert torch.allclose(res.float().cpu(), res_c.float(), atol=0.02, rtol=0.02)
    assert torch.allclose(out.float().cpu(), want.float(), atol=0.02, rtol=0.02)


def test_silu_mul_matches_reference():
    g = torch.randn(1024, 128, dtype=torch.bfloat16, device=DEV)
    u = torch.randn(1024, 128, dtype=torch.bfloat16, device=DEV)
    out = torch.empty_like(g)
    ops.silu_mul(out, g, u)
    want = torch.empty_like(g.cpu())
    ref.silu_mul(want, g.cpu(), u.cpu())
    assert torch.allclose(out.float().cpu(), want.float(), atol=0.02, rtol=0.05)


def test_rope_matches_reference():
    torch.manual_seed(2)
    T, nq, nkv, D = 17, 8, 2, 128
    q = torch.randn(T, nq, D, dtype=torch.bfloat16, device=DEV)
    k = torch.randn(T, nkv, D, dtype=torch.bfloat16, device=DEV)
    qc, kc = q.cpu().clone(), k.cpu().clone()
    tab = ops.make_cos_sin_table(256, D, device=DEV)
    pos = torch.randint(0, 256, (T,), dtype=torch.int32, device=DEV)
    ops.rope_inplace(q, k, tab, pos)
    ref.rope_inplace(qc, kc, tab.cpu(), pos.cpu())
    assert torch.allclose(q.float().cpu(), qc.float(), atol=0.03, rtol=0.03)
    assert torch.allclose(k.float().cpu(), kc.float(), atol=0.03, rtol=0.03)


def _gpu_caches(P, n_kv, D, PS):
    k = torch.zeros(P, n_kv, D // 8, PS, 8, dtype=torch.bfloat16, device=DEV)
    v = torch.zeros(P, n_kv, PS, D, dtype=torch.bfloat16, device=DEV)
    return k, v


def test_kv_append_matches_reference():
    torch.manual_seed(3)
    P, n_kv, D, PS, T = 8, 8, 128, 16, 20
    kc, vc = _gpu_caches(P, n_kv, D, PS)
    k = torch.randn(T, n_kv, D, dtype=torch.bfloat16, device=DEV)
    v = torch.randn(T, n_kv, D, dtype=torch.bfloat16, device=DEV)
    slots = torch.arange(T, dtype=torch.long, device=DEV) * 3 % (P * PS)
    ops.kv_append(kc, vc, k, v, slots)
    kr = torch.zeros_like(kc, device="cpu")
    vr = torch.zeros_like(vc, device="cpu")
    ref.kv_append(kr, vr, k.cpu(), v.cpu(), slots.cpu())
    assert torch.equal(kc.cpu(), kr)
    assert torch.equal(vc.cpu(), vr)


def _fill_seq(kc, vc, page_table_row, length, n_kv, D, PS, seed):
    gen = torch.Generator(device="cpu").manual_seed(seed)
    K = torch.randn(length, n_kv, D, generator=gen, dtype=torch.float32)
    V = torch.randn(length, n_kv, D, generator=gen, dtype=torch.float32)
    Kb, Vb = K.to(torch.bfloat16).to(DEV), V.to(torch.bfloat16).to(DEV)
    slots = []
    for t in range(length):
        page = int(page_table_row[t // PS])
        slots.append(page * PS + t % PS)
    ops.kv_append(kc, vc, Kb, Vb, torch.tensor(slots, dtype=torch.long, device=DEV))
    return Kb.cpu(), Vb.cpu()


@pytest.mark.parametrize("lens", [[5], [64], [1, 77, 200, 33], [129] * 8])
def test_decode_attention_matches_reference(lens):
    torch.manual_seed(4)
    n_q, n_kv, D, PS = 32, 8, 128, 16
    B = len(lens)
    max_pages = max(-(-l // PS) for l in lens)
    P = B * max_pages + 1
    kc, vc = _gpu_caches(P, n_kv, D, PS)
    page_table = torch.zeros(B, max_pages, dtype=torch.int32)
    next_page = 1
    for b in range(B):
        for i in range(-(-lens[b] // PS)):
            page_table[b, i] = next_page
            next_page += 1
    KVs = [_fill_seq(kc, vc, page_table[b], lens[b], n_kv, D, PS, seed=100 + b)
           for b in range(B)]
    q = torch.randn(B, n_q, D, dtype=torch.bfloat16, device=DEV)
    out = torch.empty_like(q)
    scale = 1.0 / math.sqrt(D)
    pt_dev = page_table.to(DEV)
    sl_dev = torch.tensor(lens, dtype=torch.int32, device=DEV)
    ops.paged_decode_attention(out, q, kc, vc, pt_dev, sl_dev, scale)
    want = torch.empty(B, n_q, D, dtype=torch.bfloat16)
    ref.paged_decode_attention(want, q.cpu(), kc.cpu(), vc.cpu(), page_table,
                               torch.tensor(lens, dtype=torch.int32), scale)
    diff = (out.float().cpu() - want.float()).abs().max().item()
    assert diff < 0.05, f"decode attn max abs diff {diff} (lens={lens})"


@pytest.mark.parametrize("ctx,new", [(0, 32), (0, 77), (48, 64), (130, 31), (0, 1)])
def test_prefill_attention_matches_reference(ctx, new):
    torch.manual_seed(5)
    n_q, n_kv, D, PS = 32, 8, 128, 16
    total = ctx + new
    max_pages = -(-total // PS)
    kc, vc = _gpu_caches(max_pages + 2, n_kv, D, PS)
    page_table = torch.arange(1, max_pages + 1, dtype=torch.int32).unsqueeze(0)
    _fill_seq(kc, vc, page_table[0], total, n_kv, D, PS, seed=7)
    q = torch.randn(new, n_q, D, dtype=torch.bfloat16, device=DEV)
    out = torch.zeros_like(q)
    scale = 1.0 / math.sqrt(D)
    args = (page_table.to(DEV), torch.tensor([total], dtype=torch.int32, device=DEV),
            torch.tensor([0], dtype=torch.int32, device=DEV),
            torch.tensor([new], dtype=torch.int32, device=DEV))
    ops.paged_prefill_attention(out, q, kc, vc, *args, scale)
    want = torch.zeros(new, n_q, D, dtype=torch.bfloat16)
    ref.paged_prefill_attention(want, q.cpu(), kc.cpu(), vc.cpu(), page_table,
                                torch.tensor([total], dtype=torch.int32),
                                torch.tensor([0], dtype=torch.int32),
                                torch.tensor([new], dtype=torch.int32), scale)
    diff = (out.float().cpu() - want.float()).abs().max().item()
    assert diff < 0.05, f"prefill attn max abs diff {diff} (ctx={ctx} new={new})"


def test_prefill_attention_varlen_batch():
    torch.manual_seed(6)
    n_q, n_kv, D, PS = 32, 8, 128, 16
    specs = [(0, 40), (16, 16), (64, 100)]  # (ctx, new)
    max_pages = max(-(-(c + n) // PS) for c, n in specs)
    kc, vc = _gpu_caches(len(specs) * max_pages + 1, n_kv, D, PS)
    page_table = torch.zeros(len(specs), max_pages, dtype=torch.int32)
    nxt = 1
    for b, (c, n) in enumerate(specs):
        for i in range(-(-(c + n) // PS)):
            page_table[b, i] = nxt
            nxt += 1
        _fill_seq(kc, vc, page_table[b], c + n, n_kv, D, PS, seed=50 + b)
    total_new = sum(n for _, n in specs)
    starts = []
    acc = 0
    for _, n in specs:
        starts.append(acc)
        acc += n
    q = torch.randn(total_new, n_q, D, dtype=torch.bfloat16, device=DEV)
    out = torch.zeros_like(q)
    scale = 1.0 / math.sqrt(D)
    seq_lens = torch.tensor([c + n for c, n in specs], dtype=torch.int32)
    q_starts = torch.tensor(starts, dtype=torch.int32)
    q_lens = torch.tensor([n for _, n in specs], dtype=torch.int32)
    ops.paged_prefill_attention(out, q, kc, vc, page_table.to(DEV),
                                seq_lens.to(DEV), q_starts.to(DEV),
                                q_lens.to(DEV), scale)
    want = torch.zeros_like(q.cpu())
    ref.paged_prefill_attention(want, q.cpu(), kc.cpu(), vc.cpu(), page_table,
                                seq_lens, q_starts, q_lens, scale)
    diff = (out.float().cpu() - want.float()).abs().max().item()
    assert diff < 0.05, f"varlen prefill max abs diff {diff}"


def test_greedy_sample_matches_argmax():
    torch.manual_seed(7)
    logits = torch.randn(64, 128256, dtype=torch.bfloat16, device=DEV)
    out = torch.empty(64, dtype=torch.long, device=DEV)
    ops.greedy_sample(out, logits)
    want = logits.float().argmax(-1)
    assert torch.equal(out, want)


def test_topp_sample_nucleus_property():
    torch.manual_seed(8)
    B, V = 32, 128256
    logits = torch.randn(B, V, dtype=torch.bfloat16, device=DEV)
    peak = torch.randint(0, V, (B,))
    for b in range(B):
        logits[b, peak[b]] += 12.0  # overwhelming mass on one token
    out = torch.empty(B, dtype=torch.long, device=DEV)
    temps = torch.full((B,), 0.8, device=DEV)
    tps = torch.full((B,), 0.3, device=DEV)
    seeds = torch.arange(B, dtype=torch.int64, device=DEV)
    ops.topp_sample(out, logits, temps, tps, seeds)
    assert torch.equal(out.cpu(), peak.long())


def test_topp_sample_distribution():
    """With top_p=1, temp=1, draws should roughly follow softmax over a
    3-token-dominated distribution."""
    B, V = 512, 1024
    logits = torch.full((B, V), -10.0, dtype=torch.bfloat16, device=DEV)
    logits[:, 0] = 1.0
    logits[:, 1] = 1.0
    logits[:, 2] = 1.0
    out = torch.empty(B, dtype=torch.long, device=DEV)
    temps = torch.ones(B, device=DEV)
    tps = torch.ones(B, device=DEV)
    seeds = torch.arange(B, dtype=torch.int64, device=DEV) * 7919
    ops.topp_sample(out, logits, temps, tps, seeds)
    picks = out.cpu()
    assert (picks < 3).float().mean() > 0.98
    counts = torch.bincount(picks.clamp(max=3), minlength=4)[:3].float()
    assert counts.min() > B / 3 * 0.5  # each of the 3 tokens drawn often


def test_gather_scatter_roundtrip_gpu():
    P, n_kv, D, PS = 8, 8, 128, 16
    kc, vc = _gpu_caches(P, n_kv, D, PS)
    kc.normal_()
    vc.normal_()
    plane = n_kv * D * PS
    ids = torch.tensor([2, 5, 7], dtype=torch.int32, device=DEV)
    buf = torch.zeros(3 * 2 * plane, dtype=torch.bfloat16, device=DEV)
    ops.gather_kv_pages(buf, kc, vc, ids)
    kc2, vc2 = _gpu_caches(P, n_kv, D, PS)
    ops.scatter_kv_pages(kc2, vc2, buf, ids)
    for p in [2, 5, 7]:
        assert torch.equal(kc2[p], kc[p])
        assert torch.equal(vc2[p], vc[p])


@pytest.mark.parametrize("M,N,K", [(64, 4096, 4096), (64, 6144, 4096),
                                   (64, 4096, 14336), (33, 28672, 4096),
                                   (1, 4096, 4096), (64, 128256, 4096)])
def test_skinny_gemm_matches_torch(M, N, K):
    torch.manual_seed(10)
    x = (torch.randn(M, K, dtype=torch.bfloat16, device=DEV) * 0.1)
    w = (torch.randn(N, K, dtype=torch.bfloat16, device=DEV) * 0.1)
    from agentainer_amd.ops import linear
    out = linear(x, w)
    want = torch.nn.functional.linear(x.float(), w.float())
    diff = (out.float() - want).abs()
    rel = diff.max().item() / max(want.abs().max().item(), 1e-6)
    assert rel < 0.02, f"skinny_gemm rel diff {rel} (M={M} N={N} K={K})"


def test_skinny_gemm_packed_matches_torch():
    torch.manual_seed(11)
    from agentainer_amd.ops import linear, pack_weight
    for M, N, K in [(64, 4096, 4096), (64, 6144, 4096), (33, 4096, 14336),
                    (64, 28672, 4096)]:
        x = (torch.randn(M, K, dtype=torch.bfloat16, device=DEV) * 0.1)
        w = (torch.randn(N, K, dtype=torch.bfloat16, device=DEV) * 0.1)
        out = linear(x, w, pack_weight(w))
        want = torch.nn.functional.linear(x.float(), w.float())
        rel = (out.float() - want).abs().max().item() / want.abs().max().item()
        assert rel < 0.02, f"packed skinny rel {rel} M={M} N={N} K={K}"


def test_rope_append_matches_two_ops():
    torch.manual_seed(12)
    T, n_q, n_kv, D, PS, P = 9, 8, 2, 128, 16, 4
    kc1, vc1 = _gpu_caches(P, n_kv, D, PS)
    kc2, vc2 = _gpu_caches(P, n_kv, D, PS)
    q1 = torch.randn(T, n_q, D, dtype=torch.bfloat16, device=DEV)
    k1 = torch.randn(T, n_kv, D, dtype=torch.bfloat16, device=DEV)
    v1 = torch.randn(T, n_kv, D, dtype=torch.bfloat16, device=DEV)
    q2, k2, v2 = q1.clone(), k1.clone(), v1.clone()
    tab = ops.make_cos_sin_table(64, D, device=DEV)
    pos = torch.randint(0, 64, (T,), dtype=torch.int32, device=DEV)
    slots = torch.arange(T, dtype=torch.long, device=DEV) * 5 % (P * PS)
    # two-op path
    ops.rope_inplace(q1, k1, tab, pos)
    ops.kv_append(kc1, vc1, k1, v1, slots)
    # fused path
    mod = ops._load_hip()
    mod.rope_append(q2, k2, v2, kc2, vc2, tab, pos, slots)
    assert torch.equal(q1, q2)
    assert torch.equal(kc1, kc2)
    assert torch.equal(vc1, vc2)


def test_skinny_gemm_fp8_matches_bf16():
    torch.manual_seed(13)
    from agentainer_amd.ops import linear_fp8, quantize_weight_fp8
    for M, N, K in [(64, 4096, 4096), (33, 14336, 4096), (64, 4096, 14336)]:
        x = torch.randn(M, K, dtype=torch.bfloat16, device=DEV) * 0.1
        w = torch.randn(N, K, dtype=torch.bfloat16, device=DEV) * 0.1
        wp, sw = quantize_weight_fp8(w)
        out = linear_fp8(x, wp, sw, N)
        want = torch.nn.functional.linear(x.float(), w.float())
        num = (out.float() - want).pow(2).sum().sqrt()
        den = want.pow(2).sum().sqrt()
        rel = (num / den).item()
        assert rel < 0.05, f"fp8 rel L2 err {rel} (M={M} N={N} K={K})"


def test_mixtral_fp8_generation(tmp_path):
    """tiny-mixtral with fp8 MFMA expert decode: generates, and stays close
    to the bf16 expert path (same weights)."""
    import tempfile
    from agentainer_amd.config import load_config
    from agentainer_amd.engine.llm import GenRequest, LLMEngine
    from agentainer_amd.registry import Manager
    from agentainer_amd.store import Store

    def run(expert_fp8):
        cfg = load_config(path="/nonexistent.yaml", env={})
        cfg.data["store"]["path"] = str(tmp_path / f"r{expert_fp8}")
        cfg.data["engine"]["sync_mode"] = True
        cfg.data["engine"]["kv_pool_gb"] = 1.0
        cfg.data["engine"]["expert_fp8"] = expert_fp8
        store = Store(str(tmp_path / f"s{expert_fp8}"), sync="never")
        eng = LLMEngine(store, cfg, device="cuda",
                        state_root=str(tmp_path / f"t{expert_fp8}"))
        man = Manager(store, eng, cfg)
        a = man.deploy(name="f8", model="tiny-mixtral")
        man.start(a.id)
        inst = eng._instances["tiny-mixtral"]
        req = GenRequest(agent_id=a.id, prompt_tokens=list(range(3, 35)),
                         max_new=8, temperature=0.0, top_p=1.0, seed=0)
        b = inst.binding(a.id)
        with inst._lock:
            b.queue.put(req)
            inst._pump_agent(b)
        for _ in range(16):
            inst.step()
            if req.done.is_set():
                break
        torch.cuda.synchronize()
        assert req.done.is_set() and not req.error, req.error
        eng.shutdown()
        return req.generated

    toks_fp8 = run(True)
    toks_bf16 = run(False)
    # full generations complete on both paths; numerical closeness of the
    # fp8 GEMM itself is asserted in test_skinny_gemm_fp8_matches_bf16
    # (random-init logits are near-uniform, so greedy picks legitimately
    # diverge under quantization and then compound)
    assert len(toks_fp8) == 8 and len(toks_bf16) == 8


# ---------------------------------------------------------------- fp8 KV

def _gpu_caches_fp8(P, n_kv, D, PS):
    k = torch.zeros(P, n_kv, D // 8, PS, 8, dtype=torch.float8_e4m3fn,
                    device=DEV)
    v = torch.zeros(P, n_kv, PS, D, dtype=torch.float8_e4m3fn, device=DEV)
    return k, v


def test_kv_append_fp8_matches_reference():
    torch.manual_seed(13)
    P, n_kv, D, PS, T = 8, 8, 128, 16, 20
    kc, vc = _gpu_caches_fp8(P, n_kv, D, PS)
    k = torch.randn(T, n_kv, D, dtype=torch.bfloat16, device=DEV)
    v = torch.randn(T, n_kv, D, dtype=torch.bfloat16, device=DEV)
    slots = torch.arange(T, dtype=torch.long, device=DEV) * 3 % (P * PS)
    ops.kv_append(kc, vc, k, v, slots)
    kr = torch.zeros_like(kc, device="cpu")
    vr = torch.zeros_like(vc, device="cpu")
    ref.kv_append(kr, vr, k.cpu(), v.cpu(), slots.cpu())
    # HIP __hip_fp8 and torch's float8 cast both round-to-nearest-even;
    # compare dequantized with a 1-ulp allowance for saturation edges
    dk = (kc.float().cpu() - kr.float()).abs().max().item()
    dv = (vc.float().cpu() - vr.float()).abs().max().item()
    assert dk < 0.07 and dv < 0.07, (dk, dv)


def test_decode_attention_fp8_matches_reference():
    torch.manual_seed(14)
    lens = [1, 16, 130, 300]
    n_q, n_kv, D, PS = 32, 8, 128, 16
    B = len(lens)
    max_pages = max(-(-l // PS) for l in lens)
    P = B * max_pages + 1
    kc, vc = _gpu_caches_fp8(P, n_kv, D, PS)
    page_table = torch.zeros(B, max_pages, dtype=torch.int32)
    next_page = 1
    for b in range(B):
        for i in range(-(-lens[b] // PS)):
            page_table[b, i] = next_page
            next_page += 1
    for b in range(B):
        _fill_seq(kc, vc, page_table[b], lens[b], n_kv, D, PS, seed=200 + b)
    q = torch.randn(B, n_q, D, dtype=torch.bfloat16, device=DEV)
    out = torch.empty_like(q)
    scale = 1.0 / math.sqrt(D)
    ops.paged_decode_attention(out, q, kc, vc, page_table.to(DEV),
                               torch.tensor(lens, dtype=torch.int32, device=DEV),
                               scale)
    want = torch.empty(B, n_q, D, dtype=torch.bfloat16)
    ref.paged_decode_attention(want, q.cpu(), kc.cpu(), vc.cpu(), page_table,
                               torch.tensor(lens, dtype=torch.int32), scale)
    diff = (out.float().cpu() - want.float()).abs().max().item()
    assert diff < 0.05, f"fp8 decode attn max abs diff {diff}"


def test_prefill_attention_fp8_matches_reference():
    torch.manual_seed(15)
    n_q, n_kv, D, PS = 32, 8, 128, 16
    ctx, new = 48, 64
    total = ctx + new
    max_pages = -(-total // PS)
    kc, vc = _gpu_caches_fp8(max_pages + 2, n_kv, D, PS)
    page_table = torch.arange(1, max_pages + 1, dtype=torch.int32).unsqueeze(0)
    _fill_seq(kc, vc, page_table[0], total, n_kv, D, PS, seed=77)
    q = torch.randn(new, n_q, D, dtype=torch.bfloat16, device=DEV)
    out = torch.zeros_like(q)
    scale = 1.0 / math.sqrt(D)
    ops.paged_prefill_attention(
        out, q, kc, vc, page_table.to(DEV),
        torch.tensor([total], dtype=torch.int32, device=DEV),
        torch.tensor([0], dtype=torch.int32, device=DEV),
        torch.tensor([new], dtype=torch.int32, device=DEV), scale)
    want = torch.zeros(new, n_q, D, dtype=torch.bfloat16)
    ref.paged_prefill_attention(
        want, q.cpu(), kc.cpu(), vc.cpu(), page_table,
        torch.tensor([total], dtype=torch.int32),
        torch.tensor([0], dtype=torch.int32),
        torch.tensor([new], dtype=torch.int32), scale)
    diff = (out.float().cpu() - want.float()).abs().max().item()
    assert diff < 0.05, f"fp8 prefill attn max abs diff {diff}"


def test_rope_append_fp8_matches_reference():
    torch.manual_seed(16)
    P, n_q, n_kv, D, PS, T = 6, 32, 8, 128, 16, 12
    kc, vc = _gpu_caches_fp8(P, n_kv, D, PS)
    cos_sin = ref.make_cos_sin_table(256, D, device=DEV)
    q = torch.randn(T, n_q, D, dtype=torch.bfloat16, device=DEV)
    k = torch.randn(T, n_kv, D, dtype=torch.bfloat16, device=DEV)
    v = torch.randn(T, n_kv, D, dtype=torch.bfloat16, device=DEV)
    pos = torch.arange(T, dtype=torch.int32, device=DEV)
    slots = (torch.arange(T, dtype=torch.long, device=DEV) * 5) % (P * PS)
    q_ref, k_ref = q.cpu().clone(), k.cpu().clone()
    ops.rope_append(q, k, v, kc, vc, cos_sin, pos, slots)
    ref.rope_inplace(q_ref, k_ref, cos_sin.cpu(), pos.cpu())
    kr = torch.zeros_like(kc, device="cpu")
    vr = torch.zeros_like(vc, device="cpu")
    ref.kv_append(kr, vr, k_ref, v.cpu(), slots.cpu())
    assert (q.float().cpu() - q_ref.float()).abs().max().item() < 2e-2
    dk = (kc.float().cpu() - kr.float()).abs().max().item()
    dv = (vc.float().cpu() - vr.float()).abs().max().item()
    assert dk < 0.3 and dv < 0.07, (dk, dv)  # rope'd k: e4m3 quantum at |x|~4


def test_engine_generation_fp8_kv(tmp_path):
    """End-to-end fp8-KV generation on device + stop/resume exactness."""
    from agentainer_amd.config import load_config
    from agentainer_amd.engine.llm import LLMEngine
    from agentainer_amd.registry import Manager
    from agentainer_amd.store import Store
    from test_gpu_engine import _gen

    cfg = load_config(path="/nonexistent.yaml", env={})
    root = str(tmp_path)
    cfg.data["store"]["path"] = root
    cfg.data["engine"]["sync_mode"] = True
    cfg.data["engine"]["kv_pool_gb"] = 1.0
    cfg.data["engine"]["kv_dtype"] = "fp8"
    store = Store(root + "/state", sync="never")
    engine = LLMEngine(store, cfg, device="cuda", state_root=root)
    manager = Manager(store, engine, cfg)
    try:
        a = manager.deploy(name="f8a", model="tiny-llama")
        manager.start(a.id)
        b = manager.deploy(name="f8b", model="tiny-llama")
        manager.start(b.id)
        inst = engine._instances["tiny-llama"]
        assert inst.kvm.dtype == torch.float8_e4m3fn
        prompt = list(range(3, 40))
        out_a = _gen(engine, manager, a, prompt)
        out_b = _gen(engine, manager, b, prompt)
        assert out_a == out_b and len(out_a) == 8
        manager.stop(a.id)   # offload fp8 pages to pinned host
        manager.resume(a.id)
        p2 = list(range(50, 70))
        assert _gen(engine, manager, a, p2) == _gen(engine, manager, b, p2)
    finally:
        engine.shutdown()


def test_mxfp4_gemm_matches_dequant_reference():
    """Block-scaled MXFP4 expert GEMM (v_mfma_scale_f32_16x16x128_f8f6f4)
    vs an oracle through the SAME e2m1 quantizations: both operands fp4,
    W with e8m0 block-32 scales, activations with a per-row f32 scale."""
    mod = ops._load_hip()
    torch.manual_seed(21)
    GRID = torch.tensor([0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0])
    for M, N, K in ((16, 256, 512), (64, 14336, 4096), (3, 4096, 14336)):
        w = (torch.randn(N, K) * 0.04).to(torch.bfloat16).cuda()
        x = (torch.randn(M, K) * 0.6).to(torch.bfloat16).cuda()
        wp, wsc = ops.quantize_weight_mxfp4(w)
        wd = ops.dequantize_mxfp4(wp, wsc, N, K).cuda()
        x4 = torch.empty(M, K // 2, dtype=torch.uint8, device="cuda")
        sx = torch.empty(M, dtype=torch.float32, device="cuda")
        mod.quant_fp4_rows(x4, sx, x.contiguous())
        b = x4.cpu()
        codes = torch.stack([(b & 0xF).long(), (b >> 4).long()], -1).view(M, K)
        xq = (GRID[codes & 7] * torch.where(codes >= 8, -1.0, 1.0)) \
            * sx.cpu()[:, None]
        ref = xq.cuda() @ wd.float().t()
        out = ops.linear_mxfp4(x, wp, wsc, N)
        rel = ((out.float() - ref).abs().max() / ref.abs().max()).item()
        assert rel < 2e-2, (M, N, K, rel)


def test_mixtral_fp4_experts_generate():
    """Mixtral decode with MXFP4 experts completes generations."""
    import tempfile

    from agentainer_amd.config import load_config
    from agentainer_amd.engine.llm import LLMEngine
    from agentainer_amd.registry import Manager
    from agentainer_amd.store import Store
    from test_gpu_engine import _gen

    tmp = tempfile.mkdtemp(prefix="fp4mx-")
    cfg = load_config(path="/nonexistent.yaml", env={})
    cfg.data["store"]["path"] = tmp
    cfg.data["engine"]["sync_mode"] = True
    cfg.data["engine"]["kv_pool_gb"] = 1.0
    cfg.data["engine"]["expert_fp4"] = True
    store = Store(tmp + "/state", sync="never")
    engine = LLMEngine(store, cfg, device="cuda", state_root=tmp)
    manager = Manager(store, engine, cfg)
    try:
        a = manager.deploy(name="mx4", model="tiny-mixtral")
        manager.start(a.id)
        inst = engine._instances["tiny-mixtral"]
        assert inst.model.layers[0].moe.gate_up_fp4[0] is not None
        out = _gen(engine, manager, a, list(range(3, 40)), max_new=6)
        assert len(out) == 6
    finally:
        engine.shutdown()
