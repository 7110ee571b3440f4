"""Property-based invariants of the paged KV allocator (hypothesis).

The KV pool is the correctness core of the multi-tenant engine: every
page must always be in exactly one place (free list, exactly one owner,
or shared with an exact refcount), across any interleaving of
create/append/reserve/reset/adopt/offload/restore/free. A stateful
fuzz over those operations pins the invariants the unit tests only
sample.
"""

import hypothesis.strategies as st
import pytest
import torch
from hypothesis import HealthCheck, settings
from hypothesis.stateful import (Bundle, RuleBasedStateMachine, invariant,
                                 rule)

from agentainer_amd.engine.kvcache import KVCacheManager, OutOfPages

N_PAGES = 24
PAGE = 4
MAX_SLOTS = 12


class KVMachine(RuleBasedStateMachine):
    def __init__(self):
        super().__init__()
        self.kvm = KVCacheManager(n_layers=1, n_kv=1, head_dim=8,
                                  page_size=PAGE, n_pages=N_PAGES,
                                  device="cpu", dtype=torch.bfloat16,
                                  max_slots=MAX_SLOTS)
        self.next_id = 0
        self.ckpts = {}

    seqs = Bundle("seqs")

    @rule(target=seqs)
    def create(self):
        sid = f"s{self.next_id}"
        self.next_id += 1
        try:
            self.kvm.create_seq(sid)
        except OutOfPages:  # slot exhaustion is a legal outcome
            return "dead"
        return sid

    @rule(s=seqs, n=st.integers(1, 3 * PAGE))
    def append(self, s, n):
        if s == "dead" or not self.kvm.has_seq(s):
            return
        try:
            self.kvm.append_slots(s, n)
        except OutOfPages:
            pass

    @rule(s=seqs, n=st.integers(1, 4 * PAGE))
    def reserve(self, s, n):
        if s == "dead" or not self.kvm.has_seq(s):
            return
        try:
            self.kvm.reserve(s, n)
        except OutOfPages:
            pass

    @rule(s=seqs)
    def reset(self, s):
        if s == "dead" or not self.kvm.has_seq(s):
            return
        self.kvm.reset_seq(s)

    @rule(s=seqs)
    def free(self, s):
        if s == "dead" or not self.kvm.has_seq(s):
            return
        self.kvm.free_seq(s)

    @rule(src=seqs, dst=seqs)
    def adopt(self, src, dst):
        if "dead" in (src, dst) or src == dst:
            return
        if not (self.kvm.has_seq(src) and self.kvm.has_seq(dst)):
            return
        n_pg = len(self.kvm._seqs[src].pages)
        if n_pg == 0 or self.kvm.seq_len(src) < PAGE:
            return
        if self.kvm.seq_len(dst) != 0:
            return
        n_tok = (min(self.kvm.seq_len(src), n_pg * PAGE) // PAGE) * PAGE
        if n_tok <= 0:
            return
        self.kvm.adopt_prefix(dst, src, n_tok)

    @rule(s=seqs)
    def offload_restore(self, s):
        if s == "dead" or not self.kvm.has_seq(s):
            return
        ckpt = self.kvm.offload(s, free=True)
        try:
            self.kvm.restore(s, ckpt)
        except OutOfPages:
            # restore may legitimately fail under pressure; the seq was
            # recreated empty by restore() before the page grab — check
            if not self.kvm.has_seq(s):
                return

    @invariant()
    def conservation(self):
        """Every page is free, solely owned, or shared with a correct
        refcount — and the counts add up to the pool size."""
        kvm = self.kvm
        with kvm._lock:
            free = list(kvm._free)
            assert len(free) == len(set(free)), "free list duplicates"
            owners = {}
            for sid, s in kvm._seqs.items():
                for p in s.pages:
                    owners.setdefault(p, []).append(sid)
            for p in free:
                assert p not in owners, f"page {p} free AND owned"
                assert 1 <= p < N_PAGES  # page 0 is the reserved scratch
            for p, sids in owners.items():
                if len(sids) > 1:
                    assert kvm._refs.get(p) == len(sids), \
                        f"page {p}: {len(sids)} owners, refs {kvm._refs.get(p)}"
                else:
                    assert kvm._refs.get(p) in (None,), \
                        f"page {p}: sole owner but refcounted"
            # accounting: free + owned-distinct + scratch == pool
            assert len(free) + len(owners) + 1 == N_PAGES

    @invariant()
    def slots_consistent(self):
        kvm = self.kvm
        with kvm._lock:
            assert set(kvm._slot_of) == set(kvm._seqs)
            assert len(set(kvm._slot_of.values())) == len(kvm._slot_of)
            for sid, slot in kvm._slot_of.items():
                assert kvm._seq_of_slot[slot] == sid

    @invariant()
    def lengths_fit_pages(self):
        kvm = self.kvm
        with kvm._lock:
            for sid, s in kvm._seqs.items():
                assert 0 <= s.length <= len(s.pages) * PAGE, \
                    f"{sid}: length {s.length} vs {len(s.pages)} pages"


TestKVProperties = KVMachine.TestCase
TestKVProperties.settings = settings(
    max_examples=60, stateful_step_count=40, deadline=None,
    suppress_health_check=[HealthCheck.too_slow])


@pytest.mark.timeout(300)
def test_marker():  # keeps pytest collection obvious in -q output
    assert True
