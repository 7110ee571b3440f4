"""Property-based lifecycle FSM fuzz (hypothesis).

Random interleavings of every lifecycle verb over a pool of agents,
with engine-crash injection (detach behind the registry's back) and
reconciler passes — the control-plane analog of the KV/store fuzzes.
Standing invariants mirror the reference's contract (agent.go FSM +
state_sync convergence):

  * status is always a legal FSM state;
  * RUNNING/PAUSED implies engine-attached AFTER a reconcile (and a
    crashed attachment converges to STOPPED or auto-restarts);
  * removed agents leave no registry record, engine attachment, or
    store keys behind.
"""

import shutil
import tempfile

import hypothesis.strategies as st
from hypothesis import HealthCheck, settings
from hypothesis.stateful import RuleBasedStateMachine, invariant, rule

from agentainer_amd.config import load_config
from agentainer_amd.engine.echo import EchoEngine
from agentainer_amd.registry import (CREATED, FAILED, PAUSED, RUNNING,
                                     STOPPED, AgentError, AgentNotFound,
                                     Manager)
from agentainer_amd.registry.reconciler import Reconciler
from agentainer_amd.store import Store

N_AGENTS = 4


class LifecycleMachine(RuleBasedStateMachine):
    def __init__(self):
        super().__init__()
        self.dir = tempfile.mkdtemp(prefix="fsm-fuzz-")
        cfg = load_config(path="/nonexistent.yaml", env={})
        self.store = Store(self.dir + "/s", sync="never")
        self.engine = EchoEngine(self.store)
        self.man = Manager(self.store, self.engine, cfg)
        self.rec = Reconciler(self.man, interval_s=3600.0)
        self.ids = {}       # slot -> agent_id (None = removed/never)
        self.removed = set()

    def teardown(self):
        self.store.close()
        shutil.rmtree(self.dir, ignore_errors=True)

    slots = st.integers(0, N_AGENTS - 1)

    def _id(self, slot):
        return self.ids.get(slot)

    @rule(slot=slots, auto=st.booleans())
    def deploy(self, slot, auto):
        if self._id(slot) is not None:
            return
        a = self.man.deploy(name=f"fsm-{slot}-{len(self.removed)}",
                            model="echo", auto_restart=auto)
        self.ids[slot] = a.id

    @rule(slot=slots, verb=st.sampled_from(
        ["start", "stop", "pause", "resume", "restart"]))
    def lifecycle(self, slot, verb):
        aid = self._id(slot)
        if aid is None:
            return
        try:
            getattr(self.man, verb)(aid)
        except (AgentError, AgentNotFound):
            pass  # illegal transitions must raise, never corrupt

    @rule(slot=slots)
    def remove(self, slot):
        aid = self._id(slot)
        if aid is None:
            return
        try:
            self.man.remove(aid)
        except AgentError:
            return  # e.g. running agents may require stop first
        self.ids[slot] = None
        self.removed.add(aid)

    @rule(slot=slots)
    def crash_engine_side(self, slot):
        """The container-died analog: the engine attachment vanishes
        without the registry hearing about it."""
        aid = self._id(slot)
        if aid is None:
            return
        if self.engine.is_attached(aid):
            self.engine.detach(aid, offload_kv=False)

    @rule()
    def reconcile(self):
        self.rec.sync_all()

    # ---------- invariants ----------

    @invariant()
    def legal_states(self):
        for slot, aid in self.ids.items():
            if aid is None:
                continue
            a = self.man.get(aid)
            assert a.status in (CREATED, RUNNING, STOPPED, PAUSED, FAILED), \
                a.status

    @invariant()
    def removed_leave_nothing(self):
        for aid in self.removed:
            try:
                self.man.get(aid)
                raise AssertionError(f"removed agent {aid} still in registry")
            except AgentNotFound:
                pass
            assert not self.engine.is_attached(aid)
            leaked = [k for k in self.store.keys(f"agent:{aid}:requests:*")]
            assert not leaked, f"removed agent {aid} leaked WAL keys {leaked}"

    @invariant()
    def converges_after_reconcile(self):
        """After an explicit reconcile, RUNNING/PAUSED <=> attached
        (auto-restart may legitimately flip a crashed agent back to
        RUNNING with a fresh attachment)."""
        self.rec.sync_all()
        for slot, aid in self.ids.items():
            if aid is None:
                continue
            a = self.man.get(aid)
            attached = self.engine.is_attached(aid)
            if a.status in (RUNNING, PAUSED):
                assert attached, (aid, a.status)
            else:
                assert not attached, (aid, a.status)


TestLifecycleProperties = LifecycleMachine.TestCase
TestLifecycleProperties.settings = settings(
    max_examples=30, stateful_step_count=30, deadline=None,
    suppress_health_check=[HealthCheck.too_slow])
