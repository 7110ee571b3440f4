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
