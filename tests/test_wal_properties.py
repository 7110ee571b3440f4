"""Property-based WAL state machine fuzz (hypothesis).

Random interleavings of store_request / store_response / mark_failed
(+ AOF close/reopen crashes) against a shadow, pinning the reference's
request contract (requests.go:63-275): pending -> completed on
response; pending -> retry_count++ on failure until max_retries, then
the dead-letter queue; the pending index set tracks exactly the agents
with live pending work; every record survives a crash.
"""

import shutil
import tempfile

import hypothesis.strategies as st
from hypothesis import HealthCheck, settings
from hypothesis.stateful import Bundle, RuleBasedStateMachine, invariant, rule

from agentainer_amd.store import Store
from agentainer_amd.wal import COMPLETED, FAILED, PENDING, RequestManager

AGENTS = ["a0", "a1"]
MAX_RETRIES = 2


class WalMachine(RuleBasedStateMachine):
    def __init__(self):
        super().__init__()
        self.dir = tempfile.mkdtemp(prefix="wal-fuzz-")
        self.store = Store(self.dir + "/s", sync="never")
        self.wal = RequestManager(self.store, max_retries=MAX_RETRIES)
        # shadow: req_id -> (agent, status, retries)
        self.shadow = {}

    def teardown(self):
        self.store.close()
        shutil.rmtree(self.dir, ignore_errors=True)

    reqs = Bundle("reqs")

    @rule(target=reqs, agent=st.sampled_from(AGENTS))
    def submit(self, agent):
        r = self.wal.store_request(agent, "POST", "/chat", body={"m": 1})
        self.shadow[r.id] = [agent, PENDING, 0]
        return r.id

    @rule(rid=reqs)
    def respond(self, rid):
        agent, status, retries = self.shadow[rid]
        got = self.wal.store_response(agent, rid, {"ok": True})
        if status == PENDING:
            self.shadow[rid][1] = COMPLETED
            assert got is not None and got.status == COMPLETED
        # responding to a completed/failed request is an idempotent no-op
        # or refreshes the record; it must never resurrect a dead one
        cur = self.wal.get(agent, rid)
        assert cur.status in (COMPLETED, FAILED)

    @rule(rid=reqs)
    def fail(self, rid):
        agent, status, retries = self.shadow[rid]
        got = self.wal.mark_failed(agent, rid, "boom")
        if status != PENDING:
            return
        if retries + 1 >= MAX_RETRIES:
            self.shadow[rid][1] = FAILED
            assert got.status == FAILED
        else:
            self.shadow[rid][2] += 1
            assert got.status == PENDING and got.retry_count == retries + 1

    @rule()
    def crash(self):
        self.store.close()
        self.store = Store(self.dir + "/s", sync="never")
        self.wal = RequestManager(self.store, max_retries=MAX_RETRIES)

    @invariant()
    def records_match_shadow(self):
        for rid, (agent, status, retries) in self.shadow.items():
            rec = self.wal.get(agent, rid)
            assert rec is not None, f"{rid} lost"
            assert rec.status == status, (rid, rec.status, status)
            if status == PENDING:
                assert rec.retry_count == retries

    @invariant()
    def queues_partition_requests(self):
        for agent in AGENTS:
            want = {s: sorted(r for r, (a, st_, _n) in self.shadow.items()
                              if a == agent and st_ == s)
                    for s in (PENDING, COMPLETED, FAILED)}
            for s in (PENDING, COMPLETED, FAILED):
                got = sorted(r.id for r in self.wal.by_queue(agent, s))
                assert got == want[s], (agent, s, got, want[s])

    @invariant()
    def pending_index_exact(self):
        want = sorted({a for a, st_, _n in self.shadow.values()
                       if st_ == PENDING})
        got = sorted(self.wal.agents_with_pending())
        assert got == want, (got, want)


TestWalProperties = WalMachine.TestCase
TestWalProperties.settings = settings(
    max_examples=40, stateful_step_count=30, deadline=None,
    suppress_health_check=[HealthCheck.too_slow])
