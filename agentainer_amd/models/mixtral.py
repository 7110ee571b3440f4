"""Mixtral (sparse MoE) family — MI355X-first serving forward.

BASELINE.json config 5: Mixtral 8x7B MoE agents. Architecture = the Llama
attention stack (same HIP kernels: RMSNorm, RoPE, paged attention) with
the dense MLP replaced by a top-2-of-8 expert MoE.

MoE execution model, chosen for the MI355X serving regime:

  * Decode (small batch): DENSE-ROUTED — every (local) expert runs on the
    whole batch, scaled by the router's top-k-masked weight (0 for
    unrouted tokens). At decode batch sizes every expert is hit anyway,
    so the op is expert-WEIGHT-streaming-bound and dense routing costs
    the same HBM traffic as gather/scatter — while keeping the step free
    of host syncs (hipGraph-capturable) .
  * Expert parallelism (EP over RCCL/xGMI): experts are PARTITIONED
    across ranks. Decode keeps activations replicated (the attention
    all-reduce already paid for that), each rank computes its local
    experts and one all-reduce combines — latency-optimal for tiny
    batches on xGMI. Prefill-scale batches switch to TOKEN-SHUFFLE EP:
    each rank routes its token shard, ships tokens to their experts'
    owner ranks with all_to_all over xGMI, computes, ships results back
    and all-gathers the output shards (forward_a2a).
  * fp8 MFMA expert GEMMs (config 5's decode dtype): expert weights are
    quantized offline to OCP e4m3 with per-output-channel scales,
    activations dynamically per token (ops.quantize_weight_fp8 /
    linear_fp8, v_mfma_f32_16x16x32_fp8_fp8) — half the weight bytes of
    bf16 on the streaming-bound decode path. Enable with
    engine.expert_fp8: true (weights are shared per model instance, so
    the quantization choice is engine-level, not per-agent).
  * MXFP4 expert GEMMs (engine.expert_fp4): e2m1 weights with e8m0
    block-32 scales on the gfx950 block-scaled MFMA
    (v_mfma_scale_f32_16x16x128_f8f6f4, operand maps probed in
    tools/mx_probe.py) — QUARTER the expert bytes; measured 17.1 req/s
    vs 14.2 bf16 at the config-5 bench.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional

import torch
import torch.nn.functional as F

from .. import ops
from .llama import AttnMetadata, LlamaAttention, LlamaConfig


@dataclass
class MixtralConfig(LlamaConfig):
    n_experts: int = 8
    top_k: int = 2


class MixtralMoE(torch.nn.Module):
    """Top-k router + SwiGLU experts, expert-partitioned across EP ranks."""

    def __init__(self, cfg: MixtralConfig, ep_rank: int = 0, ep_size: int = 1):
        super().__init__()
        assert cfg.n_experts % ep_size == 0
        self.cfg = cfg
        self.n_local = cfg.n_experts // ep_size
        self.e0 = ep_rank * self.n_local  # first local expert id
        self.top_k = cfg.top_k
        inter = cfg.intermediate_size
        self.router = torch.nn.Parameter(
            torch.empty(cfg.n_experts, cfg.hidden_size, dtype=torch.bfloat16))
        self.gate_up = torch.nn.ParameterList([
            torch.nn.Parameter(torch.empty(2 * inter, cfg.hidden_size,
                                           dtype=torch.bfloat16))
            for _ in range(self.n_local)])
        self.down = torch.nn.ParameterList([
            torch.nn.Parameter(torch.empty(cfg.hidden_size, inter,
                                           dtype=torch.bfloat16))
            for _ in range(self.n_local)])
        self.inter = inter
        self.gate_up_packed: List[Optional[torch.Tensor]] = [None] * self.n_local
        self.down_packed: List[Optional[torch.Tensor]] = [None] * self.n_local
        # fp8 expert path (config 5): (packed bytes, scales) per expert
        self.gate_up_fp8: List[Optional[tuple]] = [None] * self.n_local
        self.down_fp8: List[Optional[tuple]] = [None] * self.n_local
        # MXFP4 expert path (block-scaled fp4 MFMA): quarter the bytes
        self.gate_up_fp4: List[Optional[tuple]] = [None] * self.n_local
        self.down_fp4: List[Optional[tuple]] = [None] * self.n_local

    # batches at least this big use token-shuffle all-to-all EP by default
    A2A_MIN_TOKENS = 128
    # ep_mode: "auto" = token-count threshold (a2a_min_tokens), "a2a" =
    # always token-shuffle (config 5's decode all-to-all, forceable for
    # crossover measurement), "dense" = always dense-routed + all-reduce.
    # Set per engine via engine.moe_ep_mode / engine.moe_a2a_min_tokens
    # (make_mixtral_instance threads them onto every layer's MoE).
    ep_mode = "auto"
    a2a_min_tokens = A2A_MIN_TOKENS

    # prefill-scale batches route SPARSELY (gather per expert): dense
    # routing computes every expert over the WHOLE batch — fine for tiny
    # decode batches where every expert is hit anyway (and required for
    # hipGraph capture), but at prefill it wastes (E - top_k)/E of the
    # expert FLOPs. Crossover left at the decode/prefill boundary.
    SPARSE_MIN_TOKENS = 65

    def forward(self, h: torch.Tensor, ep_group=None) -> torch.Tensor:
        T = h.size(0)
        if ep_group is not None and torch.distributed.get_world_size(ep_group) > 1:
            use_a2a = (self.ep_mode == "a2a"
                       or (self.ep_mode == "auto" and T >= self.a2a_min_tokens))
            if use_a2a:
                return self.forward_a2a(h, ep_group)
        logits = F.linear(h, self.router).float()           # [T, E]
        probs = torch.softmax(logits, dim=-1)
        topv, topi = probs.topk(self.top_k, dim=-1)         # [T, k]
        topv = topv / topv.sum(dim=-1, keepdim=True)        # renormalize
        if T >= self.SPARSE_MIN_TOKENS:
            out = self._experts_sparse(h, topi, topv)
            if ep_group is not None:
                torch.distributed.all_reduce(out, group=ep_group)
            return out
        # dense top-k mask: weight[t, e] (0 if e not in token t's top-k)
        weight = torch.zeros_like(probs)
        weight.scatter_(1, topi, topv)
        out = torch.zeros_like(h, dtype=torch.float32)
        use_fp8 = (h.is_cuda and T <= 64 and self.gate_up_fp8[0] is not None)
        use_fp4 = (h.is_cuda and T <= 64 and self.gate_up_fp4[0] is not None)
        for i in range(self.n_local):
            e = self.e0 + i
            if use_fp4:
                wp, sw = self.gate_up_fp4[i]
                gu = ops.linear_mxfp4(h, wp, sw, 2 * self.inter)
            elif use_fp8:
                wp, sw = self.gate_up_fp8[i]
                gu = ops.linear_fp8(h, wp, sw, 2 * self.inter)
            else:
                gu = ops.linear(h, self.gate_up[i], self.gate_up_packed[i])
            gate, up = gu[:, :self.inter], gu[:, self.inter:]
            act = torch.empty(T, self.inter, dtype=gu.dtype, device=gu.device)
            ops.silu_mul(act, gate, up)
            if use_fp4:
                wp, sw = self.down_fp4[i]
                eo = ops.linear_mxfp4(act, wp, sw, h.size(1))
            elif use_fp8:
                wp, sw = self.down_fp8[i]
                eo = ops.linear_fp8(act, wp, sw, h.size(1))
            else:
                eo = ops.linear(act, self.down[i], self.down_packed[i])
            out += weight[:, e].unsqueeze(1) * eo.float()
        out = out.to(h.dtype)
        if ep_group is not None:
            torch.distributed.all_reduce(out, group=ep_group)
        return out

    def _experts_sparse(self, h: torch.Tensor, topi: torch.Tensor,
                        topv: torch.Tensor) -> torch.Tensor:
        """Gather-per-expert execution for prefill-scale batches: each
        local expert runs only over the tokens that routed to it (top_k/E
        of the batch on average), with a weighted index_add_ scatter back.
        Deterministic (fp32 accumulation per token, fixed expert order),
        so greedy replay stays exact."""
        T, H = h.shape
        out = torch.zeros(T, H, dtype=torch.float32, device=h.device)
        for i in range(self.n_local):
            e = self.e0 + i
            hit = (topi == e)                          # [T, k]
            rows = hit.any(dim=-1).nonzero(as_tuple=True)[0]
            if rows.numel() == 0:
                continue
            w = (topv * hit.float()).sum(dim=-1)[rows]  # [n] routing weight
            x = h.index_select(0, rows)
            gu = ops.linear(x, self.gate_up[i], None)
            gate, up = gu[:, :self.inter], gu[:, self.inter:]
            act = torch.empty(x.size(0), self.inter, dtype=gu.dtype,
                              device=gu.device)
            ops.silu_mul(act, gate, up)
            eo = ops.linear(act, self.down[i], None)
            out.index_add_(0, rows, w.unsqueeze(1).float() * eo.float())
        return out.to(h.dtype)

    # ---------- token-shuffle EP (prefill) ----------

    @staticmethod
    def _exchange(send, group):
        """Variable-size tensor all-to-all: send[d] goes to rank d; returns
        what every rank sent here. NCCL/RCCL uses all_to_all; gloo (CPU CI)
        falls back to all_gather_object."""
        import torch.distributed as dist

        world = dist.get_world_size(group)
        me = dist.get_rank(group)
        backend = dist.get_backend(group)
        if backend == "nccl":
            counts = torch.tensor([t.shape[0] for t in send], device="cuda")
            all_counts = torch.zeros(world, world, dtype=counts.dtype,
                                     device="cuda")
            dist.all_gather_into_tensor(all_counts.view(-1), counts, group=group)
            cols = send[0].shape[1]
            recv = [torch.empty(int(all_counts[d, me]), cols,
                                dtype=send[0].dtype, device="cuda")
                    for d in range(world)]
            dist.all_to_all(recv, list(send), group=group)
            return recv
        holder = [None] * world
        dist.all_gather_object(holder, [t.cpu() for t in send], group=group)
        return [holder[d][me].to(send[0].device) for d in range(world)]

    def forward_a2a(self, h: torch.Tensor, ep_group) -> torch.Tensor:
        """Token-shuffle expert parallelism: route my token shard, ship
        each (token, expert) pair to the expert owner rank, compute there,
        ship weighted results back, all-gather output shards."""
        import torch.distributed as dist

        world = dist.get_world_size(ep_group)
        me = dist.get_rank(ep_group)
        T, H = h.shape
        # contiguous token shard for this rank
        base, rem = divmod(T, world)
        t0 = me * base + min(me, rem)
        ts = base + (1 if me < rem else 0)
        hs = h[t0:t0 + ts]
        logits = F.linear(hs, self.router).float()
        probs = torch.softmax(logits, dim=-1)
        topv, topi = probs.topk(self.top_k, dim=-1)
        topv = topv / topv.sum(dim=-1, keepdim=True)
        # flatten (token, k) pairs and bucket by owner rank
        rows = torch.arange(ts, device=h.device).repeat_interleave(self.top_k)
        experts = topi.reshape(-1)
        weights = topv.reshape(-1)
        owner = experts // self.n_local
        send, send_meta = [], []
        for d in range(world):
            sel = (owner == d).nonzero(as_tuple=True)[0]
            send.append(hs[rows[sel]].contiguous())
            meta = torch.stack([
                rows[sel].float(),
                (experts[sel] % self.n_local).float(),
                weights[sel].float()], dim=1)
            send_meta.append(meta.contiguous())
        recv = self._exchange(send, ep_group)
        recv_meta = self._exchange(send_meta, ep_group)
        # compute local experts on the received tokens, weighted
        results = []
        for d in range(world):
            xd, md = recv[d], recv_meta[d]
            yd = torch.zeros(xd.shape[0], H, dtype=torch.float32,
                             device=h.device)
            le = md[:, 1].long()
            for i in range(self.n_local):
                sel = (le == i).nonzero(as_tuple=True)[0]
                if sel.numel() == 0:
                    continue
                xi = xd[sel]
                gu = ops.linear(xi, self.gate_up[i], None)
                gate, up = gu[:, :self.inter], gu[:, self.inter:]
                act = torch.empty(xi.shape[0], self.inter, dtype=gu.dtype,
                                  device=gu.device)
                ops.silu_mul(act, gate, up)
                eo = ops.linear(act, self.down[i], None)
                yd[sel] = md[sel, 2].unsqueeze(1) * eo.float()
            results.append(yd.to(h.dtype).contiguous())
        returned = self._exchange(results, ep_group)
        out_shard = torch.zeros(ts, H, dtype=torch.float32, device=h.device)
        for d in range(world):
            md = send_meta[d]
            if md.shape[0]:
                out_shard.index_add_(0, md[:, 0].long(),
                                     returned[d].float())
        # all-gather the output shards back to the replicated layout
        shards = [None] * world
        sizes = [base + (1 if r < rem else 0) for r in range(world)]
        out_shard = out_shard.to(h.dtype)
        if dist.get_backend(ep_group) == "nccl" and len(set(sizes)) == 1:
            full = torch.empty(T, H, dtype=h.dtype, device=h.device)
            dist.all_gather_into_tensor(full.view(world, -1).view(-1),
                                        out_shard.contiguous(), group=ep_group)
            return full
        dist.all_gather_object(shards, out_shard.cpu(), group=ep_group)
        return torch.cat([s.to(h.device) for s in shards], dim=0)


class MixtralLayer(torch.nn.Module):
    def __init__(self, cfg: MixtralConfig, tp_rank: int = 0, tp_size: int = 1):
        super().__init__()
        # attention is TP-sharded like Llama; experts are EP-partitioned
        # over the same ranks
        self.attn = LlamaAttention(cfg, tp_rank, tp_size)
        self.moe = MixtralMoE(cfg, ep_rank=tp_rank, ep_size=tp_size)
        self.input_ln = torch.nn.Parameter(
            torch.empty(cfg.hidden_size, dtype=torch.bfloat16))
        self.post_ln = torch.nn.Parameter(
            torch.empty(cfg.hidden_size, dtype=torch.bfloat16))
        self.eps = cfg.norm_eps

    def forward(self, hidden, residual, k_cache, v_cache, md, cos_sin,
                tp_group=None):
        normed = torch.empty_like(hidden)
        if residual is None:
            residual = hidden.clone()
            ops.rmsnorm(normed, hidden, self.input_ln, self.eps)
        else:
            ops.fused_add_rmsnorm(normed, hidden, residual, self.input_ln, self.eps)
        attn_out = self.attn(normed, k_cache, v_cache, md, cos_sin, tp_group)
        normed2 = torch.empty_like(hidden)
        ops.fused_add_rmsnorm(normed2, attn_out, residual, self.post_ln, self.eps)
        moe_out = self.moe(normed2, ep_group=tp_group)
        return moe_out, residual


class MixtralForCausalLM(torch.nn.Module):
    def __init__(self, cfg: MixtralConfig, device="cpu", tp_rank: int = 0,
                 tp_size: int = 1, seed: int = 0):
        super().__init__()
        self.cfg = cfg
        self.tp_rank = tp_rank
        self.tp_size = tp_size
        self.tp_group = None
        with torch.device(device):
            self.embed = torch.nn.Parameter(
                torch.empty(cfg.vocab_size, cfg.hidden_size, dtype=torch.bfloat16))
            self.layers = torch.nn.ModuleList(
                [MixtralLayer(cfg, tp_rank, tp_size) for _ in range(cfg.n_layers)])
            self.final_ln = torch.nn.Parameter(
                torch.empty(cfg.hidden_size, dtype=torch.bfloat16))
            self.lm_head = torch.nn.Parameter(
                torch.empty(cfg.vocab_size, cfg.hidden_size, dtype=torch.bfloat16))
        self.register_buffer(
            "cos_sin",
            ops.make_cos_sin_table(cfg.max_position, cfg.head_dim,
                                   cfg.rope_theta, device=device),
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
                p.fill_(1.0)

    @torch.no_grad()
    def pack_decode_weights(self, expert_fp8: bool = False,
                            expert_fp4: bool = False):
        for layer in self.layers:
            layer.attn.qkv_packed = ops.pack_weight(layer.attn.qkv_proj.data)
            layer.attn.o_packed = ops.pack_weight(layer.attn.o_proj.data)
            for i in range(layer.moe.n_local):
                if expert_fp4:
                    layer.moe.gate_up_fp4[i] = ops.quantize_weight_mxfp4(
                        layer.moe.gate_up[i].data)
                    layer.moe.down_fp4[i] = ops.quantize_weight_mxfp4(
                        layer.moe.down[i].data)
                elif expert_fp8:
                    layer.moe.gate_up_fp8[i] = ops.quantize_weight_fp8(
                        layer.moe.gate_up[i].data)
                    layer.moe.down_fp8[i] = ops.quantize_weight_fp8(
                        layer.moe.down[i].data)
                else:
                    layer.moe.gate_up_packed[i] = ops.pack_weight(
                        layer.moe.gate_up[i].data)
                    layer.moe.down_packed[i] = ops.pack_weight(
                        layer.moe.down[i].data)
        self.lm_head_packed = ops.pack_weight(self.lm_head.data)

    @torch.no_grad()
    def forward(self, input_ids, md: AttnMetadata, kv_caches,
                last_rows=None) -> torch.Tensor:
        hidden = F.embedding(input_ids, self.embed)
        residual = None
        for i, layer in enumerate(self.layers):
            k_cache, v_cache = kv_caches[i]
            hidden, residual = layer(hidden, residual, k_cache, v_cache, md,
                                     self.cos_sin, self.tp_group)
        final = torch.empty_like(hidden)
        ops.fused_add_rmsnorm(final, hidden, residual, self.final_ln,
                              self.cfg.norm_eps)
        if last_rows is not None:
            final = final[last_rows]
        return ops.linear(final, self.lm_head,
                          getattr(self, "lm_head_packed", None))


MIXTRAL_CONFIGS: Dict[str, MixtralConfig] = {
    "tiny-mixtral": MixtralConfig(
        name="tiny-mixtral", vocab_size=512, hidden_size=512, n_layers=2,
        n_heads=4, n_kv_heads=1, intermediate_size=512, max_position=4096,
        n_experts=4, top_k=2, tie_embeddings=False),
    # 2-shardable tiny MoE for the gloo TP/EP tests (n_kv=2, 4 experts)
    "tiny-mixtral-tp": MixtralConfig(
        name="tiny-mixtral-tp", vocab_size=512, hidden_size=1024, n_layers=2,
        n_heads=8, n_kv_heads=2, intermediate_size=512, max_position=4096,
        n_experts=4, top_k=2, tie_embeddings=False),
    "mixtral-8x7b": MixtralConfig(
        name="mixtral-8x7b", vocab_size=32064, hidden_size=4096, n_layers=32,
        n_heads=32, n_kv_heads=8, intermediate_size=14336, max_position=8192,
        rope_theta=1e6, n_experts=8, top_k=2, tie_embeddings=False),
}


def make_mixtral_instance(name: str, device: str, engine_cfg):
    """ModelInstance over a Mixtral model (engine factory hook)."""
    from ..engine.llm import ModelInstance

    inst = ModelInstance(name, MIXTRAL_CONFIGS[name], device, engine_cfg,
                         model_cls=MixtralForCausalLM)
    mode = str(engine_cfg.get("moe_ep_mode", "auto"))
    if mode not in ("auto", "a2a", "dense"):
        raise ValueError(f"unsupported moe_ep_mode {mode!r}")
    thr = int(engine_cfg.get("moe_a2a_min_tokens", MixtralMoE.A2A_MIN_TOKENS))
    for layer in inst.model.layers:
        layer.moe.ep_mode = mode
        layer.moe.a2a_min_tokens = thr
    return inst
