"""Structured/audit logging + backup/restore."""

import json
import os


def test_log_sinks_and_query(runtime):
    runtime.logger.info("hello", component="test", agent_id="a-1")
    runtime.logger.error("bad thing", component="test")
    logs = runtime.logger.get_logs(component="test")
    assert len(logs) == 2
    only_err = runtime.logger.get_logs(level="error")
    assert len(only_err) == 1 and only_err[0]["message"] == "bad thing"
    # file sink exists and is JSONL
    log_file = os.path.join(runtime.logger.log_dir, "agentainer.log")
    with open(log_file) as f:
        lines = [json.loads(line) for line in f]
    assert len(lines) == 2


def test_log_level_filtering(runtime):
    runtime.logger.level = "warn"
    assert runtime.logger.info("suppressed") == {}
    assert runtime.logger.warn("kept")["message"] == "kept"
    runtime.logger.level = "info"


def test_audit_trail(runtime):
    runtime.logger.audit("cli", "deploy", "agent-1", "success", ip="1.2.3.4")
    runtime.logger.audit("cli", "remove", "agent-1", "failure")
    entries = runtime.logger.get_audit_logs(action="deploy")
    assert len(entries) == 1
    assert entries[0]["user"] == "cli" and entries[0]["ip"] == "1.2.3.4"
    assert len(runtime.logger.get_audit_logs()) == 2


def test_tail_stream(runtime):
    seen = []
    unsub = runtime.logger.tail(seen.append)
    runtime.logger.info("streamed")
    assert len(seen) == 1 and json.loads(seen[0])["message"] == "streamed"
    unsub()


def test_per_agent_log(runtime):
    a = runtime.agents.deploy(name="la", model="echo")
    runtime.logger.info("agent event", agent_id=a.id)
    entries = runtime.agents.get_logs(a.id)
    assert len(entries) == 1 and entries[0]["message"] == "agent event"


def test_backup_restore_roundtrip(runtime):
    a = runtime.agents.deploy(name="ba", model="echo", auto_restart=True,
                              system_prompt="be nice")
    runtime.agents.start(a.id)
    runtime.agent_request(a.id, "POST", "/chat", body={"message": "remember me"})
    b = runtime.backups.create("snap1", "test backup")
    assert len(b["agents"]) == 1
    listed = runtime.backups.list()
    assert listed and listed[0]["id"] == b["id"]
    restored = runtime.backups.restore(b["id"])
    assert len(restored) == 1
    r = restored[0]
    assert r.name == "ba-restored"  # {name}-restored (manager.go:132-191)
    assert r.system_prompt == "be nice"
    hist = runtime.store.lrange(f"agent:{r.id}:conversations")
    assert len(hist) == 1 and hist[0]["user"] == "remember me"


def test_backup_export_import(runtime, tmp_path):
    runtime.agents.deploy(name="ex", model="echo")
    b = runtime.backups.create("exported")
    out = str(tmp_path / "bundle.tar.gz")
    runtime.backups.export(b["id"], out)
    runtime.backups.delete(b["id"])
    assert runtime.backups.list() == []
    ids = runtime.backups.import_(out)
    assert ids == [b["id"]]
    assert runtime.backups.load(b["id"])["name"] == "exported"


def test_backup_restores_kv_exact(tmp_path):
    """Backups snapshot the conversation KV (live copy-on-read offload):
    a restored agent continues from the SAME context it had at backup
    time, byte-exact — not a replay regeneration."""
    import torch

    from agentainer_amd.config import load_config
    from agentainer_amd.engine.llm import LLMEngine
    from agentainer_amd.service import Runtime
    from agentainer_amd.store import Store

    cfg = load_config(path="/nonexistent.yaml", env={})
    root = str(tmp_path / "root")
    cfg.data["store"]["path"] = root
    cfg.data["engine"]["kv_pool_gb"] = 0.01
    s = Store(root + "/state", sync="interval")
    torch.manual_seed(0)
    rt = Runtime(cfg, engine=LLMEngine(s, cfg, device="cpu", state_root=root),
                 store=s, state_root=root)
    try:
        a = rt.agents.deploy(name="bk", model="tiny-llama",
                             sampling={"max_tokens": 6})
        rt.agents.start(a.id)
        ctl = rt.agents.deploy(name="ctl", model="tiny-llama",
                               sampling={"max_tokens": 6})
        rt.agents.start(ctl.id)
        rt.agent_request(a.id, "POST", "/chat", body={"message": "alpha"})
        rt.agent_request(ctl.id, "POST", "/chat", body={"message": "alpha"})
        bk = rt.backups.create("kv-test", agent_ids=[a.id])
        assert bk["agents"][0].get("kv"), "live KV not snapshotted"
        # export/import round trip carries the checkpoint payload
        tar = str(tmp_path / "bundle.tar.gz")
        rt.backups.export(bk["id"], tar)
        rt.backups.delete(bk["id"])
        import shutil
        shutil.rmtree(f"{rt.backups.backup_dir}/{bk['id']}.kv")
        ids = rt.backups.import_(tar)
        assert ids == [bk["id"]]
        restored = rt.backups.restore(bk["id"])
        assert len(restored) == 1
        r = restored[0]
        rt.agents.start(r.id)
        inst = rt.engine._instances["tiny-llama"]
        assert inst.kvm.seq_len(r.id) > 0  # KV restored, not empty
        # continuation equals the uninterrupted control's continuation
        p_r = rt.agent_request(r.id, "POST", "/chat", body={"message": "beta"})[1]
        p_c = rt.agent_request(ctl.id, "POST", "/chat", body={"message": "beta"})[1]
        assert p_r["response"] == p_c["response"]
    finally:
        rt.shutdown()


def test_backup_roundtrip_property(tmp_path):
    """Property-style backup/restore: for a randomized set of agents with
    randomized histories/metrics, create -> restore must reproduce every
    agent config and conversation exactly (as {name}-restored)."""
    import random

    from agentainer_amd.backup import BackupManager
    from agentainer_amd.config import load_config
    from agentainer_amd.engine.echo import EchoEngine
    from agentainer_amd.registry import Manager
    from agentainer_amd.store import Store

    rng = random.Random(7)
    cfg = load_config(path="/nonexistent.yaml", env={})
    store = Store(str(tmp_path / "state"), sync="never")
    eng = EchoEngine(store)
    man = Manager(store, eng, cfg)
    bm = BackupManager(store, man, str(tmp_path / "bk"))
    want = {}
    for i in range(rng.randint(3, 6)):
        sp = f"sys-{rng.randint(0, 99)}" if rng.random() < 0.7 else ""
        a = man.deploy(name=f"pb{i}", model="echo", system_prompt=sp,
                       auto_restart=rng.random() < 0.5,
                       sampling={"max_tokens": rng.randint(1, 64)})
        if rng.random() < 0.8:
            man.start(a.id)
        hist = [{"user": f"u{j}", "assistant": f"r{j}"}
                for j in range(rng.randint(0, 5))]
        for h in hist:
            store.rpush(f"agent:{a.id}:conversations", h)
        want[a.name] = (sp, a.auto_restart, dict(a.sampling), hist)
    b = bm.create(name="prop", description="fuzz")
    # wipe everything, then restore
    for a in list(man.list()):
        if a.status == "running":
            man.stop(a.id)
        man.remove(a.id)
    assert not man.list()
    bm.restore(b["id"] if isinstance(b, dict) else b.id)
    got = {a.name: a for a in man.list()}
    assert set(got) == {f"{n}-restored" for n in want}, got
    for name, (sp, auto, sampling, hist) in want.items():
        a = got[f"{name}-restored"]
        assert (a.system_prompt or "") == sp
        assert a.auto_restart == auto
        assert dict(a.sampling) == sampling
        assert store.lrange(f"agent:{a.id}:conversations") == hist
    store.close()
