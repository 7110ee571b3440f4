"""Llama-3 family — MI355X-first serving forward.

The model is written for the paged multi-tenant engine, not as a generic
HF module: every hot op dispatches to the gfx950 HIP kernels
(agentainer_amd.ops), plain GEMMs go to hipBLASLt via F.linear, and the
forward takes paged-KV attention metadata (prefill: varlen packed rows;
decode: one row per sequence). Weights are bf16, random-init by default
(BASELINE.json: synthetic prompts / random-init weights; no network for
checkpoints) with a safetensors loader for real weights.

Tensor-parallel sharding (TP over RCCL/xGMI) slices n_heads/intermediate
by rank: column-parallel qkv/gate/up, row-parallel o/down with all-reduce
(agentainer_amd.parallel).
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Dict, List, Optional

import torch
import torch.nn.functional as F

from .. import ops


@dataclass
class LlamaConfig:
    name: str = "tiny-llama"
    vocab_size: int = 512
    hidden_size: int = 512
    n_layers: int = 2
    n_heads: int = 4
    n_kv_heads: int = 1
    head_dim: int = 128
    intermediate_size: int = 1024
    rope_theta: float = 500000.0
    max_position: int = 8192
    norm_eps: float = 1e-5
    tie_embeddings: bool = True
    qkv_bias: bool = False  # Qwen2-family checkpoints carry q/k/v biases

    @property
    def q_size(self):
        return self.n_heads * self.head_dim

    @property
    def kv_size(self):
        return self.n_kv_heads * self.head_dim

    def kv_bytes_per_token(self, dtype_bytes: int = 2) -> int:
        return 2 * self.n_layers * self.n_kv_heads * self.head_dim * dtype_bytes


@dataclass
class AttnMetadata:
    """Paged-KV attention metadata for one engine step."""

    page_table: torch.Tensor      # [B, max_pages] int32
    seq_lens: torch.Tensor        # [B] int32 — total context length per seq
    slot_mapping: torch.Tensor    # [T] int64 — where each new token's K/V goes
    positions: torch.Tensor       # [T] int32
    is_prefill: bool = False
    query_starts: Optional[torch.Tensor] = None  # [B] int32 (prefill)
    query_lens: Optional[torch.Tensor] = None    # [B] int32 (prefill)
    # host-known max(query_lens): avoids a BLOCKING D2H read in the
    # attention launcher (measured ~2 ms/call behind a deep GPU queue)
    max_qlen: int = 0


class LlamaAttention(torch.nn.Module):
    def __init__(self, cfg: LlamaConfig, tp_rank: int = 0, tp_size: int = 1):
        super().__init__()
        assert cfg.n_heads % tp_size == 0, "n_heads must divide tp_size"
        assert cfg.n_kv_heads % tp_size == 0 or tp_size == 1, \
            "n_kv_heads must divide tp_size"
        self.cfg = cfg
        self.tp_size = tp_size
        self.n_heads = cfg.n_heads // tp_size
        self.n_kv = max(cfg.n_kv_heads // tp_size, 1)
        self.head_dim = cfg.head_dim
        self.scale = 1.0 / math.sqrt(cfg.head_dim)
        q_out = self.n_heads * cfg.head_dim
        kv_out = self.n_kv * cfg.head_dim
        self.qkv_proj = torch.nn.Parameter(
            torch.empty(q_out + 2 * kv_out, cfg.hidden_size, dtype=torch.bfloat16))
        self.o_proj = torch.nn.Parameter(
            torch.empty(cfg.hidden_size, q_out, dtype=torch.bfloat16))
        self.qkv_bias = (torch.nn.Parameter(
            torch.zeros(q_out + 2 * kv_out, dtype=torch.bfloat16))
            if cfg.qkv_bias else None)
        self.qkv_packed = None
        self.o_packed = None
        # opt-in quantized decode weights ("turbo": engine.dense_quant)
        self.qkv_q = None
        self.o_q = None

    def forward(self, h, k_cache, v_cache, md: AttnMetadata, cos_sin, tp_group=None):
        T = h.size(0)
        if self.qkv_q is not None and T <= 64 and h.is_cuda:
            qkv = ops.linear_quant(h, self.qkv_q)
        else:
            qkv = ops.linear(h, self.qkv_proj, self.qkv_packed)
        if self.qkv_bias is not None:
            qkv = qkv + self.qkv_bias
        q_sz = self.n_heads * self.head_dim
        kv_sz = self.n_kv * self.head_dim
        # strided views into the fused GEMM output — the HIP kernels take
        # token strides, so no contiguous copies are materialized
        q = qkv[:, :q_sz].view(T, self.n_heads, self.head_dim)
        k = qkv[:, q_sz:q_sz + kv_sz].view(T, self.n_kv, self.head_dim)
        v = qkv[:, q_sz + kv_sz:].view(T, self.n_kv, self.head_dim)
        ops.rope_append(q, k, v, k_cache, v_cache, cos_sin, md.positions,
                        md.slot_mapping)
        out = torch.empty(T, self.n_heads, self.head_dim, dtype=q.dtype,
                          device=q.device)
        if md.is_prefill:
            ops.paged_prefill_attention(out, q, k_cache, v_cache, md.page_table,
                                        md.seq_lens, md.query_starts,
                                        md.query_lens, self.scale,
                                        max_qlen=md.max_qlen)
        else:
            ops.paged_decode_attention(out, q, k_cache, v_cache, md.page_table,
                                       md.seq_lens, self.scale)
        if self.o_q is not None and T <= 64 and h.is_cuda:
            o = ops.linear_quant(out.view(T, q_sz), self.o_q)
        else:
            o = ops.linear(out.view(T, q_sz), self.o_proj, self.o_packed)
        if tp_group is not None:
            torch.distributed.all_reduce(o, group=tp_group)
        return o


class LlamaMLP(torch.nn.Module):
    def __init__(self, cfg: LlamaConfig, tp_size: int = 1):
        super().__init__()
        inter = cfg.intermediate_size // tp_size
        self.gate_up = torch.nn.Parameter(
            torch.empty(2 * inter, cfg.hidden_size, dtype=torch.bfloat16))
        self.down = torch.nn.Parameter(
            torch.empty(cfg.hidden_size, inter, dtype=torch.bfloat16))
        self.inter = inter
        self.gate_up_packed = None
        self.down_packed = None
        self.gate_up_q = None
        self.down_q = None

    def forward(self, h, tp_group=None):
        turbo = self.gate_up_q is not None and h.size(0) <= 64 and h.is_cuda
        if turbo:
            gu = ops.linear_quant(h, self.gate_up_q)
        else:
            gu = ops.linear(h, self.gate_up, self.gate_up_packed)
        gate, up = gu[:, :self.inter], gu[:, self.inter:]
        act = torch.empty(gu.size(0), self.inter, dtype=gu.dtype,
                          device=gu.device)
        ops.silu_mul(act, gate, up)
        if turbo:
            out = ops.linear_quant(act, self.down_q)
        else:
            out = ops.linear(act, self.down, self.down_packed)
        if tp_group is not None:
            torch.distributed.all_reduce(out, group=tp_group)
        return out


class LlamaLayer(torch.nn.Module):
    def __init__(self, cfg: LlamaConfig, tp_rank: int = 0, tp_size: int = 1):
        super().__init__()
        self.attn = LlamaAttention(cfg, tp_rank, tp_size)
        self.mlp = LlamaMLP(cfg, tp_size)
        self.input_ln = torch.nn.Parameter(torch.empty(cfg.hidden_size, dtype=torch.bfloat16))
        self.post_ln = torch.nn.Parameter(torch.empty(cfg.hidden_size, dtype=torch.bfloat16))
        self.eps = cfg.norm_eps

    def forward(self, hidden, residual, k_cache, v_cache, md, cos_sin, tp_group=None):
        normed = torch.empty_like(hidden)
        if residual is None:
            residual = hidden.clone()
            ops.rmsnorm(normed, hidden, self.input_ln, self.eps)
        else:
            ops.fused_add_rmsnorm(normed, hidden, residual, self.input_ln, self.eps)
        attn_out = self.attn(normed, k_cache, v_cache, md, cos_sin, tp_group)
        normed2 = torch.empty_like(hidden)
        ops.fused_add_rmsnorm(normed2, attn_out, residual, self.post_ln, self.eps)
        mlp_out = self.mlp(normed2, tp_group)
        return mlp_out, residual


class LlamaForCausalLM(torch.nn.Module):
    def __init__(self, cfg: LlamaConfig, device="cpu", tp_rank: int = 0,
                 tp_size: int = 1, seed: int = 0):
        super().__init__()
        self.cfg = cfg
        self.tp_rank = tp_rank
        self.tp_size = tp_size
        self.tp_group = None
        # construct parameters directly on the target device (an 8B/70B
        # shard must never round-trip through host RAM)
        with torch.device(device):
            self.embed = torch.nn.Parameter(
                torch.empty(cfg.vocab_size, cfg.hidden_size, dtype=torch.bfloat16))
            self.layers = torch.nn.ModuleList(
                [LlamaLayer(cfg, tp_rank, tp_size) for _ in range(cfg.n_layers)])
            self.final_ln = torch.nn.Parameter(
                torch.empty(cfg.hidden_size, dtype=torch.bfloat16))
            if cfg.tie_embeddings:
                self.lm_head = self.embed
            else:
                self.lm_head = torch.nn.Parameter(
                    torch.empty(cfg.vocab_size, cfg.hidden_size, dtype=torch.bfloat16))
        self.register_buffer(
            "cos_sin",
            ops.make_cos_sin_table(cfg.max_position, cfg.head_dim, cfg.rope_theta,
                                   device=device),
            persistent=False)
        self.random_init(seed)

    @torch.no_grad()
    def random_init(self, seed: int = 0):
        dev = self.embed.device
        gen = torch.Generator(device=dev).manual_seed(seed)
        for name, p in self.named_parameters():
            if p.dim() >= 2:
                p.data.normal_(0.0, 0.02, generator=gen)
            else:
                p.fill_(1.0)  # norm weights

    @torch.no_grad()
    def forward(self, input_ids: torch.Tensor, md: AttnMetadata,
                kv_caches: List, last_rows: Optional[torch.Tensor] = None
                ) -> torch.Tensor:
        """Returns logits for `last_rows` (defaults: all rows)."""
        hidden = F.embedding(input_ids, self.embed)
        residual = None
        for i, layer in enumerate(self.layers):
            k_cache, v_cache = kv_caches[i]
            hidden, residual = layer(hidden, residual, k_cache, v_cache, md,
                                     self.cos_sin, self.tp_group)
        final = torch.empty_like(hidden)
        ops.fused_add_rmsnorm(final, hidden, residual, self.final_ln, self.cfg.norm_eps)
        if last_rows is not None:
            final = final[last_rows]
        return ops.linear(final, self.lm_head,
                          getattr(self, 'lm_head_packed', None))  # bf16

    @torch.no_grad()
    def pack_decode_weights(self, dense_quant: str = ""):
        """Pre-shuffle every projection weight into the fragment-linear
        layout the skinny decode GEMM streams (ops.pack_weight). Keeps the
        natural-layout Parameters for the prefill hipBLASLt path — the
        packed copies double weight memory, well inside 288 GB HBM3E.

        dense_quant: "" (bf16, the default/headline dtype), "fp8"
        (per-channel e4m3), or "mxfp4" (e2m1 + e8m0 block-32 scales) —
        opt-in reduced-precision DECODE projections (prefill stays bf16)."""
        for layer in self.layers:
            if dense_quant == "mxfp4":
                for mod_, wname, qname in (
                        (layer.attn, "qkv_proj", "qkv_q"),
                        (layer.attn, "o_proj", "o_q"),
                        (layer.mlp, "gate_up", "gate_up_q"),
                        (layer.mlp, "down", "down_q")):
                    w = getattr(mod_, wname).data
                    p, sc = ops.quantize_weight_mxfp4(w)
                    setattr(mod_, qname, ("mxfp4", p, sc, w.size(0)))
            elif dense_quant == "fp8":
                for mod_, wname, qname in (
                        (layer.attn, "qkv_proj", "qkv_q"),
                        (layer.attn, "o_proj", "o_q"),
                        (layer.mlp, "gate_up", "gate_up_q"),
                        (layer.mlp, "down", "down_q")):
                    w = getattr(mod_, wname).data
                    p, sw = ops.quantize_weight_fp8(w)
                    setattr(mod_, qname, ("fp8", p, sw, w.size(0)))
            else:
                layer.attn.qkv_packed = ops.pack_weight(layer.attn.qkv_proj.data)
                layer.attn.o_packed = ops.pack_weight(layer.attn.o_proj.data)
                layer.mlp.gate_up_packed = ops.pack_weight(layer.mlp.gate_up.data)
                layer.mlp.down_packed = ops.pack_weight(layer.mlp.down.data)
        self.lm_head_packed = ops.pack_weight(
            self.lm_head.data if isinstance(self.lm_head, torch.nn.Parameter)
            else self.lm_head)

    @torch.no_grad()
    def load_safetensors(self, path: str):
        """Load HF-format Llama weights (weights-path deploys)."""
        from safetensors.torch import load_file
        import glob, os

        files = sorted(glob.glob(os.path.join(path, "*.safetensors")))
        if not files:
            raise FileNotFoundError(f"no safetensors under {path}")
        sd = {}
        for f in files:
            sd.update(load_file(f))
        self._load_hf_state_dict(sd)

    @torch.no_grad()
    def _load_hf_state_dict(self, sd: Dict[str, torch.Tensor]):
        """Copy full HF tensors into this rank's (possibly TP-sharded)
        parameters: column-parallel q/k/v/gate/up slice output rows by
        rank, row-parallel o/down slice input columns; embeddings, norms
        and lm_head are replicated. TP=1 reduces to the identity slices,
        so weights-path deploys shard inside the loader for any degree
        (the round-1 loader only accepted full shapes)."""
        cfg = self.cfg
        r, s = self.tp_rank, self.tp_size
        qh = cfg.n_heads // s
        kvh = max(cfg.n_kv_heads // s, 1)
        hd = cfg.head_dim
        inter = cfg.intermediate_size // s

        def t(name):
            return sd[name].to(torch.bfloat16)

        def rows(x, n):  # column-parallel: rank's slice of output rows
            return x[r * n:(r + 1) * n]

        self.embed.copy_(t("model.embed_tokens.weight"))
        if not cfg.tie_embeddings and "lm_head.weight" in sd:
            self.lm_head.copy_(t("lm_head.weight"))
        self.final_ln.copy_(t("model.norm.weight"))
        for i, layer in enumerate(self.layers):
            pfx = f"model.layers.{i}."
            q = rows(t(pfx + "self_attn.q_proj.weight"), qh * hd)
            k = rows(t(pfx + "self_attn.k_proj.weight"), kvh * hd)
            v = rows(t(pfx + "self_attn.v_proj.weight"), kvh * hd)
            layer.attn.qkv_proj.copy_(torch.cat([q, k, v], dim=0))
            if layer.attn.qkv_bias is not None:
                layer.attn.qkv_bias.copy_(torch.cat(
                    [rows(t(pfx + "self_attn.q_proj.bias"), qh * hd),
                     rows(t(pfx + "self_attn.k_proj.bias"), kvh * hd),
                     rows(t(pfx + "self_attn.v_proj.bias"), kvh * hd)], dim=0))
            layer.attn.o_proj.copy_(
                t(pfx + "self_attn.o_proj.weight")[:, r * qh * hd:(r + 1) * qh * hd])
            g = rows(t(pfx + "mlp.gate_proj.weight"), inter)
            u = rows(t(pfx + "mlp.up_proj.weight"), inter)
            layer.mlp.gate_up.copy_(torch.cat([g, u], dim=0))
            layer.mlp.down.copy_(
                t(pfx + "mlp.down_proj.weight")[:, r * inter:(r + 1) * inter])
            layer.input_ln.copy_(t(pfx + "input_layernorm.weight"))
            layer.post_ln.copy_(t(pfx + "post_attention_layernorm.weight"))


def config_from_hf(path: str) -> LlamaConfig:
    """Build a LlamaConfig from an HF checkpoint directory's config.json
    (weights-path deploys — the reference's "image" analog is a local
    model directory, SURVEY.md §2.1 docker-client row)."""
    import json
    import os

    with open(os.path.join(path, "config.json")) as f:
        hf = json.load(f)
    mt = hf.get("model_type", "")
    archs = hf.get("architectures", [])
    known = mt in ("llama", "qwen2", "") or any(
        ("Llama" in a or "Qwen2" in a) for a in archs)
    if not known:
        raise ValueError(f"unsupported model_type {mt!r} at {path} "
                         "(llama/qwen2 families only)")
    n_heads = int(hf["num_attention_heads"])
    hidden = int(hf["hidden_size"])
    head_dim = int(hf.get("head_dim") or hidden // n_heads)
    if head_dim != 128:
        raise ValueError(f"head_dim {head_dim} unsupported (kernels are "
                         "specialized for 128)")
    return LlamaConfig(
        name=path,
        vocab_size=int(hf["vocab_size"]),
        hidden_size=hidden,
        n_layers=int(hf["num_hidden_layers"]),
        n_heads=n_heads,
        n_kv_heads=int(hf.get("num_key_value_heads") or n_heads),
        head_dim=head_dim,
        intermediate_size=int(hf["intermediate_size"]),
        rope_theta=float(hf.get("rope_theta", 500000.0)),
        max_position=int(hf.get("max_position_embeddings", 8192)),
        norm_eps=float(hf.get("rms_norm_eps", 1e-5)),
        tie_embeddings=bool(hf.get("tie_word_embeddings", False)),
        qkv_bias=(mt == "qwen2" or any("Qwen2" in a for a in archs)
                  or bool(hf.get("attention_bias", False))),
    )


# ---------------- model registry ----------------

LLAMA_CONFIGS: Dict[str, LlamaConfig] = {
    # hd=128 + GQA ratio 4 so the tiny model exercises the real kernels
    "tiny-llama": LlamaConfig(name="tiny-llama", vocab_size=512, hidden_size=512,
                              n_layers=2, n_heads=4, n_kv_heads=1,
                              intermediate_size=1024, max_position=4096),
    # 2-way-shardable tiny model for the gloo TP tests (n_kv=2)
    "tiny-llama-tp": LlamaConfig(name="tiny-llama-tp", vocab_size=512,
                                 hidden_size=1024, n_layers=2, n_heads=8,
                                 n_kv_heads=2, intermediate_size=1024,
                                 max_position=4096),
    "llama3-8b": LlamaConfig(name="llama3-8b", vocab_size=128256,
                             hidden_size=4096, n_layers=32, n_heads=32,
                             n_kv_heads=8, intermediate_size=14336,
                             max_position=8192, tie_embeddings=False),
    "llama3-70b": LlamaConfig(name="llama3-70b", vocab_size=128256,
                              hidden_size=8192, n_layers=80, n_heads=64,
                              n_kv_heads=8, intermediate_size=28672,
                              max_position=8192, tie_embeddings=False),
}
