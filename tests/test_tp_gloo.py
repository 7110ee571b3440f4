"""Tensor-parallel engine over gloo (world_size 2, CPU).

Covers the distributed path the driver's 8-GPU round-end run exercises
(SURVEY.md §4 implication (4)): TP sharding math, the SPMD worker loop
(plan broadcast + deterministic page allocation), and TP KV
checkpoint/restore — all with the same code that runs RCCL on MI355X.

Correctness oracle: a FULL (tp=1) tiny model whose weights are sharded
by hand onto the two ranks; greedy generation must match token-for-token.
"""

import os
import socket
import tempfile

import pytest
import torch
import torch.multiprocessing as mp

TP_MODEL = "tiny-llama-tp"
PROMPT = list(range(3, 43))
MAX_NEW = 6


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _shard_weights(full_sd, model, rank, size):
    """Slice a full model state dict onto one TP rank's model."""
    import re

    cfg = model.cfg
    qh = cfg.n_heads // size
    kvh = cfg.n_kv_heads // size
    hd = cfg.head_dim
    inter = cfg.intermediate_size // size
    with torch.no_grad():
        model.embed.copy_(full_sd["embed"])
        model.final_ln.copy_(full_sd["final_ln"])
        if not cfg.tie_embeddings:
            model.lm_head.copy_(full_sd["lm_head"])
        for i, layer in enumerate(model.layers):
            pfx = f"layers.{i}."
            fq = full_sd[pfx + "attn.qkv_proj"]
            q_full, kv_full = cfg.n_heads * hd, cfg.n_kv_heads * hd
            q = fq[rank * qh * hd:(rank + 1) * qh * hd]
            k = fq[q_full + rank * kvh * hd: q_full + (rank + 1) * kvh * hd]
            v = fq[q_full + kv_full + rank * kvh * hd:
                   q_full + kv_full + (rank + 1) * kvh * hd]
            layer.attn.qkv_proj.copy_(torch.cat([q, k, v]))
            layer.attn.o_proj.copy_(
                full_sd[pfx + "attn.o_proj"][:, rank * qh * hd:(rank + 1) * qh * hd])
            fgu = full_sd[pfx + "mlp.gate_up"]
            fi = cfg.intermediate_size
            gate = fgu[rank * inter:(rank + 1) * inter]
            up = fgu[fi + rank * inter: fi + (rank + 1) * inter]
            layer.mlp.gate_up.copy_(torch.cat([gate, up]))
            layer.mlp.down.copy_(
                full_sd[pfx + "mlp.down"][:, rank * inter:(rank + 1) * inter])
            layer.input_ln.copy_(full_sd[pfx + "input_ln"])
            layer.post_ln.copy_(full_sd[pfx + "post_ln"])


def _reference_tokens(full_sd, tmpdir):
    """Greedy generation on the full (tp=1) model."""
    from agentainer_amd.config import load_config
    from agentainer_amd.engine.llm import GenRequest, LLMEngine
    from agentainer_amd.registry import Manager
    from agentainer_amd.store import Store

    cfg = load_config(path="/nonexistent.yaml", env={})
    cfg.data["engine"]["sync_mode"] = True
    cfg.data["engine"]["kv_pool_gb"] = 0.02
    store = Store(os.path.join(tmpdir, "ref-state"), sync="never")
    eng = LLMEngine(store, cfg, device="cpu", state_root=tmpdir + "/ref")
    man = Manager(store, eng, cfg)
    a = man.deploy(name="ref", model=TP_MODEL, sampling={"max_tokens": MAX_NEW})
    man.start(a.id)
    inst = eng._instances[TP_MODEL]
    inst.model.load_state_dict(
        {k: v for k, v in full_sd.items()}, strict=False)
    # state dict keys: convert plain names used in _shard_weights
    with torch.no_grad():
        inst.model.embed.copy_(full_sd["embed"])
        inst.model.final_ln.copy_(full_sd["final_ln"])
        for i, layer in enumerate(inst.model.layers):
            pfx = f"layers.{i}."
            layer.attn.qkv_proj.copy_(full_sd[pfx + "attn.qkv_proj"])
            layer.attn.o_proj.copy_(full_sd[pfx + "attn.o_proj"])
            layer.mlp.gate_up.copy_(full_sd[pfx + "mlp.gate_up"])
            layer.mlp.down.copy_(full_sd[pfx + "mlp.down"])
            layer.input_ln.copy_(full_sd[pfx + "input_ln"])
            layer.post_ln.copy_(full_sd[pfx + "post_ln"])
    req = GenRequest(agent_id=a.id, prompt_tokens=PROMPT, max_new=MAX_NEW,
                     temperature=0.0, top_p=1.0, seed=0)
    b = inst.binding(a.id)
    with inst._lock:
        b.queue.put(req)
        inst._pump_agent(b)
    for _ in range(MAX_NEW + 4):
        inst.step()
        if req.done.is_set():
            break
    assert req.done.is_set() and not req.error
    eng.shutdown()
    store.close()
    return req.generated


def _tp_worker(rank, world, port, tmpdir, result_file):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world), "LOCAL_RANK": str(rank),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist

    from agentainer_amd import parallel as par
    from agentainer_amd.config import load_config
    from agentainer_amd.engine.llm import GenRequest, LLMEngine
    from agentainer_amd.registry import Manager
    from agentainer_amd.store import Store

    par.init_distributed(backend="gloo")
    full_sd = torch.load(os.path.join(tmpdir, "full_sd.pt"), weights_only=True)
    cfg = load_config(path="/nonexistent.yaml", env={})
    cfg.data["engine"]["sync_mode"] = True
    cfg.data["engine"]["kv_pool_gb"] = 0.02
    cfg.data["engine"]["tp_degree"] = world
    store = Store(os.path.join(tmpdir, f"state-{rank}"), sync="never")
    eng = LLMEngine(store, cfg, device="cpu", state_root=f"{tmpdir}/r{rank}")
    if rank != 0:
        # build the instance when rank0 broadcasts, then serve plans;
        # shard weights when the instance appears
        import threading

        def patch_weights():
            # run_worker creates the instance on the "instance" cmd; patch
            # by wrapping _make_instance
            orig = eng._make_instance
            def wrapped(model):
                inst = orig(model)
                _shard_weights(full_sd, inst.model, rank, world)
                return inst
            eng._make_instance = wrapped
        patch_weights()
        eng.run_worker()
        dist.barrier()
        return
    man = Manager(store, eng, cfg)
    a = man.deploy(name="tp", model=TP_MODEL, sampling={"max_tokens": MAX_NEW})
    man.start(a.id)
    inst = eng._instances[TP_MODEL]
    _shard_weights(full_sd, inst.model, 0, world)
    req = GenRequest(agent_id=a.id, prompt_tokens=PROMPT, max_new=MAX_NEW,
                     temperature=0.0, top_p=1.0, seed=0)
    b = inst.binding(a.id)
    with inst._lock:
        b.queue.put(req)
        inst._pump_agent(b)
    for _ in range(MAX_NEW + 4):
        inst.step()
        if req.done.is_set():
            break
    # KV checkpoint roundtrip under TP: stop + resume, then one more turn
    man.stop(a.id)
    man.resume(a.id)
    req2 = GenRequest(agent_id=a.id, prompt_tokens=list(range(50, 70)),
                      max_new=MAX_NEW, temperature=0.0, top_p=1.0, seed=0)
    b = inst.binding(a.id)
    with inst._lock:
        b.queue.put(req2)
        inst._pump_agent(b)
    for _ in range(MAX_NEW + 4):
        inst.step()
        if req2.done.is_set():
            break
    # COW prefix sharing under TP: adoption is plan-driven, so both ranks
    # must adopt the same pages in lockstep; a divergence would corrupt
    # the all-reduced activations and show up as a token mismatch
    sp = "shared tp system prompt driving prefix adoption across ranks"
    pf_toks = []
    pags = []
    for name in ("tp-p1", "tp-p2"):
        ag = man.deploy(name=name, model=TP_MODEL, system_prompt=sp,
                        sampling={"max_tokens": MAX_NEW})
        man.start(ag.id)
        pags.append(ag)
    pk = inst.binding(pags[0].id).prefix_tokens or []
    for ag in pags:
        rq = GenRequest(agent_id=ag.id, prompt_tokens=list(pk) + PROMPT,
                        max_new=MAX_NEW, temperature=0.0, top_p=1.0, seed=0)
        bb = inst.binding(ag.id)
        with inst._lock:
            bb.queue.put(rq)
            inst._pump_agent(bb)
        for _ in range(MAX_NEW + 6):
            inst.step()
            if rq.done.is_set():
                break
        pf_toks.append(list(rq.generated))
    torch.save({"tokens": req.generated, "err": req.error,
                "tokens2": req2.generated, "err2": req2.error,
                "len_after": inst.kvm.seq_len(a.id),
                "pfx_tokens": pf_toks, "pfx_len": len(pk),
                "pfx_refs": bool(inst.kvm._refs)}, result_file)
    eng.shutdown()  # broadcasts shutdown to the worker
    dist.barrier()


def _write_tp_checkpoint(d):
    """2-shardable HF checkpoint (heads 8 / kv 2 / head_dim 128)."""
    import json
    import os

    from safetensors.torch import save_file

    from agentainer_amd.models.llama import LlamaForCausalLM, config_from_hf

    os.makedirs(d, exist_ok=True)
    cfg = {
        "model_type": "llama", "architectures": ["LlamaForCausalLM"],
        "vocab_size": 512, "hidden_size": 1024, "num_hidden_layers": 2,
        "num_attention_heads": 8, "num_key_value_heads": 2, "head_dim": 128,
        "intermediate_size": 1024, "rope_theta": 500000.0,
        "max_position_embeddings": 4096, "rms_norm_eps": 1e-5,
        "tie_word_embeddings": True, "eos_token_id": 0,
    }
    with open(os.path.join(d, "config.json"), "w") as f:
        json.dump(cfg, f)
    lc = config_from_hf(d)
    m = LlamaForCausalLM(lc, device="cpu", seed=11)
    sd = {"model.embed_tokens.weight": m.embed.data.clone(),
          "model.norm.weight": m.final_ln.data.clone()}
    q_sz, kv_sz = lc.q_size, lc.kv_size
    for i, layer in enumerate(m.layers):
        pfx = f"model.layers.{i}."
        qkv = layer.attn.qkv_proj.data
        sd[pfx + "self_attn.q_proj.weight"] = qkv[:q_sz].clone()
        sd[pfx + "self_attn.k_proj.weight"] = qkv[q_sz:q_sz + kv_sz].clone()
        sd[pfx + "self_attn.v_proj.weight"] = qkv[q_sz + kv_sz:].clone()
        sd[pfx + "self_attn.o_proj.weight"] = layer.attn.o_proj.data.clone()
        gu = layer.mlp.gate_up.data
        sd[pfx + "mlp.gate_proj.weight"] = gu[:lc.intermediate_size].clone()
        sd[pfx + "mlp.up_proj.weight"] = gu[lc.intermediate_size:].clone()
        sd[pfx + "mlp.down_proj.weight"] = layer.mlp.down.data.clone()
        sd[pfx + "input_layernorm.weight"] = layer.input_ln.data.clone()
        sd[pfx + "post_attention_layernorm.weight"] = layer.post_ln.data.clone()
    save_file(sd, os.path.join(d, "model.safetensors"))


def _gen_once(eng, man, model, prompt, max_new):
    from agentainer_amd.engine.llm import GenRequest

    a = man.deploy(name="wp", model=model, sampling={"max_tokens": max_new})
    man.start(a.id)
    inst = eng._instances[model]
    req = GenRequest(agent_id=a.id, prompt_tokens=prompt, max_new=max_new,
                     temperature=0.0, top_p=1.0, seed=0)
    b = inst.binding(a.id)
    with inst._lock:
        b.queue.put(req)
        inst._pump_agent(b)
    for _ in range(max_new + 4):
        inst.step()
        if req.done.is_set():
            break
    assert req.done.is_set() and not req.error, req.error
    return req.generated


def _wp_worker(rank, world, port, tmpdir, ckpt_dir, result_file):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world), "LOCAL_RANK": str(rank),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist

    from agentainer_amd import parallel as par
    from agentainer_amd.config import load_config
    from agentainer_amd.engine.llm import LLMEngine
    from agentainer_amd.registry import Manager
    from agentainer_amd.store import Store

    par.init_distributed(backend="gloo")
    cfg = load_config(path="/nonexistent.yaml", env={})
    cfg.data["engine"]["sync_mode"] = True
    cfg.data["engine"]["kv_pool_gb"] = 0.02
    cfg.data["engine"]["tp_degree"] = world
    store = Store(os.path.join(tmpdir, f"wp-state-{rank}"), sync="never")
    eng = LLMEngine(store, cfg, device="cpu", state_root=f"{tmpdir}/wp-r{rank}")
    if rank != 0:
        # NO weight patching: load_safetensors shards by this rank itself
        eng.run_worker()
        dist.barrier()
        return
    man = Manager(store, eng, cfg)
    toks = _gen_once(eng, man, ckpt_dir, PROMPT, MAX_NEW)
    torch.save({"tokens": toks}, result_file)
    eng.shutdown()
    dist.barrier()


@pytest.mark.timeout(300)
def test_tp2_weights_path_shards_in_loader(tmp_path):
    """Weights-path deploy with tp_degree=2: the safetensors loader
    slices q/k/v/o/gate_up/down by rank (VERDICT r1 #4 — round 1 required
    hand-sharding in test code and threw shape errors on real deploys).
    Greedy tokens must match a full (tp=1) engine on the same checkpoint."""
    from agentainer_amd.config import load_config
    from agentainer_amd.engine.llm import LLMEngine
    from agentainer_amd.registry import Manager
    from agentainer_amd.store import Store

    tmpdir = str(tmp_path)
    ckpt = os.path.join(tmpdir, "ckpt")
    _write_tp_checkpoint(ckpt)

    # reference: full model, tp=1
    cfg = load_config(path="/nonexistent.yaml", env={})
    cfg.data["engine"]["sync_mode"] = True
    cfg.data["engine"]["kv_pool_gb"] = 0.02
    store = Store(os.path.join(tmpdir, "ref-state"), sync="never")
    eng = LLMEngine(store, cfg, device="cpu", state_root=tmpdir + "/ref")
    man = Manager(store, eng, cfg)
    want = _gen_once(eng, man, ckpt, PROMPT, MAX_NEW)
    eng.shutdown()
    store.close()
    assert len(want) == MAX_NEW

    result_file = os.path.join(tmpdir, "wp-result.pt")
    ctx = mp.get_context("spawn")
    for attempt in range(2):
        port = _free_port()
        procs = [ctx.Process(target=_wp_worker,
                             args=(r, 2, port, tmpdir, ckpt, result_file))
                 for r in range(2)]
        for p in procs:
            p.start()
        codes = []
        for p in procs:
            p.join(timeout=240)
            codes.append(p.exitcode)
        for p in procs:
            if p.is_alive():
                p.terminate()
        if all(c == 0 for c in codes):
            break
        assert attempt == 0, f"worker exits {codes} (after retry)"
    res = torch.load(result_file, weights_only=True)
    assert res["tokens"] == want, f"TP tokens {res['tokens']} != full {want}"


@pytest.mark.timeout(300)
def test_tp2_matches_full_model(tmp_path):
    from agentainer_amd.models.llama import LLAMA_CONFIGS, LlamaForCausalLM

    tmpdir = str(tmp_path)
    # build the full model once; save a plain-named state dict
    full = LlamaForCausalLM(LLAMA_CONFIGS[TP_MODEL], device="cpu", seed=3)
    sd = {"embed": full.embed.data, "final_ln": full.final_ln.data}
    for i, layer in enumerate(full.layers):
        pfx = f"layers.{i}."
        sd[pfx + "attn.qkv_proj"] = layer.attn.qkv_proj.data
        sd[pfx + "attn.o_proj"] = layer.attn.o_proj.data
        sd[pfx + "mlp.gate_up"] = layer.mlp.gate_up.data
        sd[pfx + "mlp.down"] = layer.mlp.down.data
        sd[pfx + "input_ln"] = layer.input_ln.data
        sd[pfx + "post_ln"] = layer.post_ln.data
    torch.save(sd, os.path.join(tmpdir, "full_sd.pt"))

    want = _reference_tokens(sd, tmpdir)
    assert len(want) == MAX_NEW

    result_file = os.path.join(tmpdir, "result.pt")
    ctx = mp.get_context("spawn")
    # one retry with a fresh port: the gloo rendezvous can lose the port to
    # another process between _free_port() and rank 0's bind. Token
    # mismatches below stay strict — only infra failures are retried.
    for attempt in range(2):
        port = _free_port()
        procs = [ctx.Process(target=_tp_worker,
                             args=(r, 2, port, tmpdir, result_file))
                 for r in range(2)]
        for p in procs:
            p.start()
        codes = []
        for p in procs:
            p.join(timeout=240)
            codes.append(p.exitcode)
        for p in procs:
            if p.is_alive():
                p.terminate()
        if all(c == 0 for c in codes):
            break
        assert attempt == 0, f"worker exits {codes} (after retry)"
    res = torch.load(result_file, weights_only=True)
    assert res["err"] is None and res["err2"] is None
    assert res["tokens"] == want, f"TP tokens {res['tokens']} != full {want}"
    assert len(res["tokens2"]) == MAX_NEW  # post-restore turn completed
    assert res["len_after"] > len(PROMPT) + MAX_NEW  # KV restored + extended
    # prefix sharing stayed in lockstep across ranks
    assert res["pfx_len"] >= 16  # at least one whole page shared
    assert res["pfx_refs"] is True
    assert (res["pfx_tokens"][0] == res["pfx_tokens"][1]
            and len(res["pfx_tokens"][0]) == MAX_NEW), res["pfx_tokens"]
