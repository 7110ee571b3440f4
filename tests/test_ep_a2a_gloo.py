"""Token-shuffle EP all-to-all over gloo (world 2, CPU): the sharded MoE
must reproduce the single-rank dense-routed MoE on the same weights."""

import os
import socket

import pytest
import torch
import torch.multiprocessing as mp

from agentainer_amd.models.mixtral import MIXTRAL_CONFIGS, MixtralMoE

T, SEED = 160, 7  # >= A2A_MIN_TOKENS so the a2a path engages


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _full_moe():
    torch.manual_seed(SEED)
    cfg = MIXTRAL_CONFIGS["tiny-mixtral"]
    moe = MixtralMoE(cfg)
    for p in moe.parameters():
        p.data.normal_(0, 0.05, generator=torch.Generator().manual_seed(SEED))
    # distinct per-expert weights
    g = torch.Generator().manual_seed(SEED + 1)
    for i in range(moe.n_local):
        moe.gate_up[i].data.normal_(0, 0.05, generator=g)
        moe.down[i].data.normal_(0, 0.05, generator=g)
    return cfg, moe


def _worker(rank, world, port, tmpdir):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port)})
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    cfg, full = _full_moe()
    shard = MixtralMoE(cfg, ep_rank=rank, ep_size=world)
    with torch.no_grad():
        shard.router.copy_(full.router)
        for i in range(shard.n_local):
            shard.gate_up[i].copy_(full.gate_up[shard.e0 + i])
            shard.down[i].copy_(full.down[shard.e0 + i])
    torch.manual_seed(SEED + 2)
    h = torch.randn(T, cfg.hidden_size, dtype=torch.bfloat16)
    want = full(h.clone())                       # dense single-rank oracle
    got = shard.forward_a2a(h.clone(), dist.group.WORLD)
    diff = (got.float() - want.float()).abs().max().item()
    assert diff < 0.05, f"rank {rank}: a2a EP diff {diff}"
    # the small-batch path (dense + all_reduce) must agree too
    got2 = shard(h[:8].clone(), ep_group=dist.group.WORLD)
    diff2 = (got2.float() - want[:8].float()).abs().max().item()
    assert diff2 < 0.05, f"rank {rank}: dense EP diff {diff2}"
    # DECODE-shaped token a2a (config 5: expert all-to-all for decode):
    # force ep_mode="a2a" so tiny batches take the token-shuffle path and
    # must match the dense oracle token-for-token
    shard.ep_mode = "a2a"
    for Td in (1, 4, 8):
        got3 = shard(h[:Td].clone(), ep_group=dist.group.WORLD)
        diff3 = (got3.float() - want[:Td].float()).abs().max().item()
        assert diff3 < 0.05, f"rank {rank}: decode a2a T={Td} diff {diff3}"
    # and ep_mode="dense" pins the latency path regardless of batch size
    shard.ep_mode = "dense"
    got4 = shard(h.clone(), ep_group=dist.group.WORLD)
    diff4 = (got4.float() - want.float()).abs().max().item()
    assert diff4 < 0.05, f"rank {rank}: forced-dense diff {diff4}"
    shard.ep_mode = "auto"
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(240)
def test_ep_a2a_matches_dense(tmp_path):
    port = _free_port()
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, 2, port, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=200)
        assert p.exitcode == 0, f"exit {p.exitcode}"


# ------------------- engine-level decode a2a (full plan machinery) --------

def _engine_worker(rank, world, port, tmpdir, result_file):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world), "LOCAL_RANK": str(rank),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port)})
    import torch.distributed as dist

    from agentainer_amd import parallel as par
    from agentainer_amd.config import load_config
    from agentainer_amd.engine.llm import GenRequest, LLMEngine
    from agentainer_amd.registry import Manager
    from agentainer_amd.store import Store

    par.init_distributed(backend="gloo")
    results = {}
    for mode in ("dense", "a2a"):
        cfg = load_config(path="/nonexistent.yaml", env={})
        cfg.data["engine"]["sync_mode"] = True
        cfg.data["engine"]["kv_pool_gb"] = 0.02
        cfg.data["engine"]["tp_degree"] = world
        cfg.data["engine"]["moe_ep_mode"] = mode
        cfg.data["engine"]["moe_a2a_min_tokens"] = 1
        store = Store(os.path.join(tmpdir, f"ep-{mode}-{rank}"), sync="never")
        eng = LLMEngine(store, cfg, device="cpu",
                        state_root=f"{tmpdir}/ep-{mode}-r{rank}")
        if rank != 0:
            eng.run_worker()
            continue
        man = Manager(store, eng, cfg)
        a = man.deploy(name=f"moe-{mode}", model="tiny-mixtral-tp",
                       sampling={"max_tokens": 6})
        man.start(a.id)
        inst = eng._instances["tiny-mixtral-tp"]
        req = GenRequest(agent_id=a.id, prompt_tokens=list(range(3, 23)),
                         max_new=6, temperature=0.0, top_p=1.0, seed=0)
        b = inst.binding(a.id)
        with inst._lock:
            b.queue.put(req)
            inst._pump_agent(b)
        for _ in range(12):
            inst.step()
            if req.done.is_set():
                break
        assert req.done.is_set() and not req.error, req.error
        results[mode] = list(req.generated)
        eng.shutdown()
    if rank == 0:
        torch.save(results, result_file)
    dist.barrier()


@pytest.mark.timeout(240)
def test_engine_decode_a2a_matches_dense(tmp_path):
    """Config 5's decode expert all-to-all end-to-end: a TP/EP-2 Mixtral
    engine generating with moe_ep_mode=a2a (forced down to decode batch
    sizes) must emit exactly the tokens the dense+all-reduce mode does."""
    result_file = os.path.join(str(tmp_path), "ep-engine.pt")
    for attempt in range(2):
        port = _free_port()
        ctx = mp.get_context("spawn")
        procs = [ctx.Process(target=_engine_worker,
                             args=(r, 2, port, str(tmp_path), result_file))
                 for r in range(2)]
        for p in procs:
            p.start()
        codes = []
        for p in procs:
            p.join(timeout=200)
            codes.append(p.exitcode)
        for p in procs:
            if p.is_alive():
                p.terminate()
        if all(c == 0 for c in codes):
            break
        assert attempt == 0, f"worker exits {codes} (after retry)"
    res = torch.load(result_file, weights_only=True)
    assert len(res["dense"]) == 6
    assert res["a2a"] == res["dense"], res
