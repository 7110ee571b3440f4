"""Store: structures, TTL, AOF crash-recovery, pub/sub patterns."""

import time

from agentainer_amd.store import Store


def test_strings_and_ttl(store):
    store.set("a", {"x": 1})
    assert store.get("a") == {"x": 1}
    assert store.exists("a")
    store.set("b", "v", ttl=0.05)
    assert store.get("b") == "v"
    time.sleep(0.06)
    assert store.get("b") is None
    assert not store.exists("b")
    store.delete("a")
    assert store.get("a") is None


def test_lists(store):
    store.rpush("q", "1", "2")
    store.rpush("q", "3")
    assert store.lrange("q") == ["1", "2", "3"]
    assert store.llen("q") == 3
    assert store.lrange("q", -2, -1) == ["2", "3"]
    store.lrem("q", "2")
    assert store.lrange("q") == ["1", "3"]
    store.ltrim("q", 1, -1)
    assert store.lrange("q") == ["3"]
    assert store.lpop("q") == "3"
    assert store.lpop("q") is None


def test_zsets(store):
    store.zadd("z", 1.0, "a")
    store.zadd("z", 2.0, "b")
    store.zadd("z", 3.0, "c")
    assert [m for m, _ in store.zrangebyscore("z", 1.5, 3.0)] == ["b", "c"]
    assert store.zremrangebyscore("z", 0, 1.5) == 1
    assert store.zcard("z") == 2


def test_hash_set_incr(store):
    store.hset("h", "f", 10)
    assert store.hget("h", "f") == 10
    assert store.hgetall("h") == {"f": 10}
    store.sadd("s", "x", "y")
    store.srem("s", "x")
    assert store.smembers("s") == ["y"]
    assert store.incr("cnt") == 1
    assert store.incr("cnt", 5) == 6


def test_keys_pattern(store):
    store.set("agent:1:requests:a", 1)
    store.set("agent:1:requests:b", 2)
    store.set("agent:2:requests:c", 3)
    assert sorted(store.keys("agent:1:requests:*")) == [
        "agent:1:requests:a", "agent:1:requests:b"]


def test_aof_recovery(tmp_path):
    path = str(tmp_path / "kv")
    s = Store(path, sync="always")
    s.set("k", "v")
    s.rpush("lst", "a", "b")
    s.zadd("z", 1.0, "m")
    s.sadd("set", "x")
    s.hset("h", "f", "v")
    # simulate crash: no close()
    s2 = Store(path)
    assert s2.get("k") == "v"
    assert s2.lrange("lst") == ["a", "b"]
    assert s2.zcard("z") == 1
    assert s2.smembers("set") == ["x"]
    assert s2.hget("h", "f") == "v"
    s2.close()


def test_compaction_roundtrip(tmp_path):
    path = str(tmp_path / "kv")
    s = Store(path)
    for i in range(100):
        s.set(f"k{i}", i)
    s.rpush("l", *range(10))
    s.compact()
    s.set("after", "snap")
    s.close()
    s2 = Store(path)
    assert s2.get("k99") == 99
    assert s2.get("after") == "snap"
    assert len(s2.lrange("l")) == 10
    s2.close()


def test_torn_aof_tail(tmp_path):
    path = str(tmp_path / "kv")
    s = Store(path, sync="always")
    s.set("good", 1)
    s.close()
    with open(path + ".aof", "a") as f:
        f.write('["set", "bad", tr')  # torn write
    s2 = Store(path)
    assert s2.get("good") == 1
    assert s2.get("bad") is None
    s2.close()


def test_pubsub_glob_patterns(store):
    """Pattern subscribe must glob-match (the reference's monitor.go:301 bug
    — plain Subscribe on 'agent:status:*' — is fixed by design here)."""
    seen = []
    unsub = store.subscribe("agent:status:*", lambda ch, msg: seen.append((ch, msg)))
    store.publish("agent:status:agent-1", "running")
    store.publish("other:channel", "x")
    assert seen == [("agent:status:agent-1", "running")]
    unsub()
    store.publish("agent:status:agent-1", "stopped")
    assert len(seen) == 1
