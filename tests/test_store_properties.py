"""Property-based crash-safety of the embedded store (hypothesis).

The store is the registry + WAL + metrics substrate; its durability
contract is "any committed op sequence survives close/reopen (AOF
replay) and compaction (snapshot + tail)". A stateful fuzz drives a
random op mix against a SHADOW model (plain dicts) and, at random
points, closes and reopens the store — every readable piece of state
must match the shadow exactly afterwards. TTL ops are exercised with
far-future expiries so time never mutates state mid-sequence.
"""

import shutil
import tempfile

import hypothesis.strategies as st
from hypothesis import HealthCheck, settings
from hypothesis.stateful import RuleBasedStateMachine, invariant, rule

from agentainer_amd.store import Store

KEYS = [f"k{i}" for i in range(6)]
FIELDS = ["a", "b", "c"]
VALS = st.one_of(st.integers(-5, 5), st.text("xy", max_size=3),
                 st.dictionaries(st.sampled_from(FIELDS),
                                 st.integers(0, 3), max_size=2))


class StoreMachine(RuleBasedStateMachine):
    def __init__(self):
        super().__init__()
        self.dir = tempfile.mkdtemp(prefix="store-fuzz-")
        self.store = Store(self.dir + "/s", sync="interval")
        # shadow: type-tagged per key
        self.kv = {}
        self.lists = {}
        self.zsets = {}
        self.hashes = {}
        self.sets = {}

    def teardown(self):
        self.store.close()
        shutil.rmtree(self.dir, ignore_errors=True)

    # ---------- string / scalar ----------

    @rule(k=st.sampled_from(KEYS), v=VALS)
    def set_(self, k, v):
        self.store.set(k, v)
        self.kv[k] = v
        for m in (self.lists, self.zsets, self.hashes, self.sets):
            m.pop(k, None)

    @rule(k=st.sampled_from(KEYS))
    def delete(self, k):
        self.store.delete(k)
        for m in (self.kv, self.lists, self.zsets, self.hashes, self.sets):
            m.pop(k, None)

    @rule(k=st.sampled_from(KEYS), by=st.integers(-3, 3))
    def incr(self, k, by):
        if k in self.kv and not isinstance(self.kv[k], int):
            return
        if any(k in m for m in (self.lists, self.zsets, self.hashes, self.sets)):
            return
        self.kv[k] = self.kv.get(k, 0) + by
        self.store.incr(k, by)

    # ---------- list ----------

    @rule(k=st.sampled_from(KEYS), v=st.integers(0, 9))
    def rpush(self, k, v):
        if k in self.kv or any(k in m for m in (self.zsets, self.hashes, self.sets)):
            return
        self.store.rpush(k, v)
        self.lists.setdefault(k, []).append(v)

    @rule(k=st.sampled_from(KEYS))
    def lpop(self, k):
        if k not in self.lists:
            return
        got = self.store.lpop(k)
        want = self.lists[k].pop(0) if self.lists[k] else None
        if not self.lists.get(k):
            self.lists.pop(k, None)
        assert got == want

    @rule(k=st.sampled_from(KEYS), start=st.integers(-4, 4),
          stop=st.integers(-4, 4))
    def ltrim(self, k, start, stop):
        if k not in self.lists:
            return
        self.store.ltrim(k, start, stop)
        lst = self.lists[k]
        n = len(lst)
        s = start if start >= 0 else max(0, n + start)
        e = stop + 1 if stop >= 0 else n + stop + 1
        trimmed = lst[s:max(e, 0)] if e is not None else lst[s:]
        if trimmed:
            self.lists[k] = trimmed
        else:
            self.lists.pop(k, None)

    # ---------- zset / hash / set ----------

    @rule(k=st.sampled_from(KEYS), m=st.sampled_from(FIELDS),
          score=st.integers(0, 9))
    def zadd(self, k, m, score):
        if k in self.kv or any(k in d for d in (self.lists, self.hashes, self.sets)):
            return
        self.store.zadd(k, float(score), m)
        self.zsets.setdefault(k, {})[m] = float(score)

    @rule(k=st.sampled_from(KEYS), lo=st.integers(0, 9), hi=st.integers(0, 9))
    def zrem_range(self, k, lo, hi):
        if k not in self.zsets:
            return
        self.store.zremrangebyscore(k, float(lo), float(hi))
        z = {m: s for m, s in self.zsets[k].items() if not (lo <= s <= hi)}
        if z:
            self.zsets[k] = z
        else:
            self.zsets.pop(k, None)

    @rule(k=st.sampled_from(KEYS), f=st.sampled_from(FIELDS), v=VALS)
    def hset(self, k, f, v):
        if k in self.kv or any(k in d for d in (self.lists, self.zsets, self.sets)):
            return
        self.store.hset(k, f, v)
        self.hashes.setdefault(k, {})[f] = v

    @rule(k=st.sampled_from(KEYS), m=st.sampled_from(FIELDS))
    def sadd(self, k, m):
        if k in self.kv or any(k in d for d in (self.lists, self.zsets, self.hashes)):
            return
        self.store.sadd(k, m)
        self.sets.setdefault(k, set()).add(m)

    @rule(k=st.sampled_from(KEYS), m=st.sampled_from(FIELDS))
    def srem(self, k, m):
        if k not in self.sets:
            return
        self.store.srem(k, m)
        self.sets[k].discard(m)
        if not self.sets[k]:
            self.sets.pop(k, None)

    # ---------- durability events ----------

    @rule()
    def reopen(self):
        """SIGKILL-analog: drop the in-memory image, replay AOF."""
        self.store.close()
        self.store = Store(self.dir + "/s", sync="interval")

    @rule()
    def compact(self):
        self.store.compact()

    # ---------- the contract ----------

    @invariant()
    def matches_shadow(self):
        st_ = self.store
        for k, v in self.kv.items():
            assert st_.get(k) == v, (k, st_.get(k), v)
        for k, lst in self.lists.items():
            assert st_.lrange(k) == lst, (k, st_.lrange(k), lst)
        for k, z in self.zsets.items():
            got = dict(st_.zrangebyscore(k, float("-inf"), float("inf")))
            assert got == z, (k, got, z)
        for k, h in self.hashes.items():
            assert st_.hgetall(k) == h, (k, st_.hgetall(k), h)
        for k, s in self.sets.items():
            assert set(st_.smembers(k)) == s, (k, st_.smembers(k), s)
        shadow_keys = (set(self.kv) | set(self.lists) | set(self.zsets)
                       | set(self.hashes) | set(self.sets))
        assert set(st_.keys()) == shadow_keys, (set(st_.keys()), shadow_keys)


TestStoreProperties = StoreMachine.TestCase
TestStoreProperties.settings = settings(
    max_examples=40, stateful_step_count=30, deadline=None,
    suppress_health_check=[HealthCheck.too_slow])
