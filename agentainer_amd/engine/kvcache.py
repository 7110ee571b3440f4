"""Paged KV-cache manager — HBM3E page pool + pinned-host checkpoints.

The MI355X replacement for the reference's "container keeps its
filesystem + Redis state" durability (SURVEY.md §2.3 "KV-cache manager"
row, §5 Checkpoint/resume):

  * One page pool per model layer in HBM (layouts match the attention
    kernels: k [P, n_kv, D/8, PS, 8], v [P, n_kv, PS, D]).
  * Sequences (= agent conversations) own page lists via a page table.
  * stop/offload gathers a sequence's pages into a contiguous GPU staging
    buffer (ops.gather_kv_pages) and streams it to PINNED host memory with
    a non-blocking copy (hipMemcpyAsync under torch); resume streams it
    back and rebuilds the page table. Checkpoints can also be serialized
    to disk for backup/export.

Sized for 288 GB HBM3E: by default the pool takes the configured fraction
of free device memory after weights.
"""

from __future__ import annotations

import threading
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import torch

from .. import ops


class OutOfPages(RuntimeError):
    pass


@dataclass
class Sequence:
    seq_id: str
    pages: List[int] = field(default_factory=list)
    length: int = 0  # tokens written


@dataclass
class KVCheckpoint:
    """Host-side image of one sequence's KV state."""

    length: int
    n_pages: int
    data: torch.Tensor  # pinned host [n_layers, n_pages, page_shorts] bf16


class KVCacheManager:
    def __init__(self, n_layers: int, n_kv: int, head_dim: int, page_size: int,
                 n_pages: int, device="cpu", dtype=torch.bfloat16,
                 max_slots: int = 1024, max_pages_per_seq: int = 512,
                 mirrors: Optional[bool] = None):
        assert head_dim % 8 == 0
        self.n_layers = n_layers
        self.n_kv = n_kv
        self.head_dim = head_dim
        self.page_size = page_size
        self.n_pages = n_pages
        self.device = device
        # dtype: bf16 (default) or float8_e4m3fn — fp8 halves KV bytes per
        # token (2x the agents per GPU) with OCP e4m3 storage, vLLM-style
        # scale-free semantics (e4m3 range +-448 covers post-norm K/V)
        self.dtype = dtype
        self.k_caches: List[torch.Tensor] = []
        self.v_caches: List[torch.Tensor] = []
        for _ in range(n_layers):
            self.k_caches.append(torch.zeros(
                n_pages, n_kv, head_dim // 8, page_size, 8, dtype=dtype, device=device))
            self.v_caches.append(torch.zeros(
                n_pages, n_kv, page_size, head_dim, dtype=dtype, device=device))
        # page 0 is reserved as the scratch page: hipGraph-padded decode rows
        # write their garbage K/V there (see llm.ModelInstance._device_decode)
        self._free: List[int] = list(range(n_pages - 1, 0, -1))
        self._seqs: Dict[str, Sequence] = {}
        # copy-on-write refcounts for SHARED pages (prefix sharing). A page
        # absent from the map has exactly one owner; adopt_prefix bumps the
        # count, release decrements and only returns a page to the pool when
        # its last owner lets go. Shared pages are never written: adopters
        # take only FULL pages, and appends always land at slot >= length,
        # which is past the shared region by construction.
        self._refs: Dict[int, int] = {}
        self._lock = threading.RLock()
        # shorts per page per layer: K plane + V plane
        self.page_shorts = 2 * n_kv * head_dim * page_size
        # ---- device mirrors for the in-graph decode step ----
        # dev_page_table[slot] / dev_seq_lens[slot] are the ground truth the
        # captured decode graph reads; host bookkeeping mirrors them. On GPU
        # they live in HBM; `mirrors=True` on CPU materializes them as CPU
        # tensors so the device-decode path (and the async-TP protocol on
        # top of it) is testable without hardware.
        self.is_gpu = str(device) not in ("cpu",)
        self.mirrors = self.is_gpu if mirrors is None else bool(mirrors)
        self.max_slots = max_slots
        self.max_pages_per_seq = max_pages_per_seq
        if self.mirrors:
            self.dev_page_table = torch.zeros(max_slots, max_pages_per_seq,
                                              dtype=torch.int32, device=device)
            self.dev_seq_lens = torch.full((max_slots,), -1, dtype=torch.int32,
                                           device=device)
        else:
            self.dev_page_table = None
            self.dev_seq_lens = None
        # slots are assigned on CPU too: slot allocation is deterministic
        # across TP ranks (same create/free order), so the slot id doubles
        # as the wire identity of a sequence in binary step plans
        self._free_slots: List[int] = list(range(max_slots - 1, -1, -1))
        self._slot_of: Dict[str, int] = {}
        self._seq_of_slot: Dict[int, str] = {}

    # ---------- queries ----------

    @property
    def free_pages(self) -> int:
        with self._lock:
            return len(self._free)

    @property
    def used_pages(self) -> int:
        return self.n_pages - self.free_pages

    def kv_caches(self) -> List[Tuple[torch.Tensor, torch.Tensor]]:
        return list(zip(self.k_caches, self.v_caches))

    def has_seq(self, seq_id: str) -> bool:
        with self._lock:
            return seq_id in self._seqs

    def seq_len(self, seq_id: str) -> int:
        with self._lock:
            s = self._seqs.get(seq_id)
            return s.length if s else 0

    def seq_bytes(self, seq_id: str) -> int:
        with self._lock:
            s = self._seqs.get(seq_id)
            if not s:
                return 0
            return (len(s.pages) * self.page_shorts * self.n_layers *
                    self.k_caches[0].element_size())

    # ---------- allocation ----------

    def create_seq(self, seq_id: str) -> Sequence:
        with self._lock:
            if seq_id in self._seqs:
                raise ValueError(f"sequence {seq_id} exists")
            if not self._free_slots:
                # check BEFORE registering: raising after the _seqs insert
                # left a zombie slotless entry behind (found by the
                # stateful fuzz in tests/test_kvcache_properties.py)
                raise OutOfPages("no free sequence slots")
            s = Sequence(seq_id)
            self._seqs[seq_id] = s
            slot = self._free_slots.pop()
            self._slot_of[seq_id] = slot
            self._seq_of_slot[slot] = seq_id
            if self.mirrors:
                self.dev_seq_lens[slot] = 0
            return s

    def slot(self, seq_id: str) -> int:
        return self._slot_of[seq_id]

    def seq_of_slot(self, slot: int) -> str:
        """Inverse of slot(): wire identity -> sequence (TP worker plans)."""
        with self._lock:
            return self._seq_of_slot[slot]

    def push_dev(self, seq_id: str) -> None:
        """Sync a sequence's page row + length to the device mirrors."""
        if not self.mirrors:
            return
        with self._lock:
            s = self._seqs[seq_id]
            slot = self._slot_of[seq_id]
            if s.pages:
                self.dev_page_table[slot, :len(s.pages)] = torch.tensor(
                    s.pages, dtype=torch.int32, device=self.device)
            self.dev_seq_lens[slot] = s.length

    def ensure_decode_page(self, seq_id: str) -> None:
        """Host half of the in-graph decode append: guarantee the page for
        the NEXT token exists (the graph computes the slot on device)."""
        with self._lock:
            s = self._seqs[seq_id]
            page_idx = s.length // self.page_size
            if page_idx >= self.max_pages_per_seq:
                raise OutOfPages("sequence exceeded max_pages_per_seq")
            if page_idx >= len(s.pages):
                if not self._free:
                    raise OutOfPages(f"KV pool exhausted ({self.n_pages} pages)")
                page = self._free.pop()
                s.pages.append(page)
                if self.mirrors:
                    self.dev_page_table[self._slot_of[seq_id], page_idx] = page

    def advance_host(self, seq_id: str) -> None:
        """Host length += 1 after an in-graph decode step incremented the
        device length."""
        with self._lock:
            self._seqs[seq_id].length += 1

    def decode_batch_prepare(self, seq_ids) -> list:
        """One-lock batch form of ensure_decode_page: guarantees each
        sequence's next-token page exists; returns the slot indices."""
        rows = []
        with self._lock:
            for seq_id in seq_ids:
                s = self._seqs[seq_id]
                page_idx = s.length // self.page_size
                if page_idx >= self.max_pages_per_seq:
                    raise OutOfPages("sequence exceeded max_pages_per_seq")
                if page_idx >= len(s.pages):
                    if not self._free:
                        raise OutOfPages(
                            f"KV pool exhausted ({self.n_pages} pages)")
                    page = self._free.pop()
                    s.pages.append(page)
                    if self.mirrors:
                        self.dev_page_table[self._slot_of[seq_id],
                                            page_idx] = page
                rows.append(self._slot_of[seq_id])
        return rows

    def advance_many(self, seq_ids) -> None:
        with self._lock:
            for seq_id in seq_ids:
                self._seqs[seq_id].length += 1

    def rollback_many(self, seq_ids) -> None:
        """Undo one speculative decode append per sequence (async decode:
        a row that turned out finished had a garbage token appended by the
        in-flight step). Device decrement is stream-ordered after that
        step's graph, so it lands exactly once the append has happened."""
        if not seq_ids:
            return
        with self._lock:
            slots = []
            for seq_id in seq_ids:
                self._seqs[seq_id].length -= 1
                if self.mirrors:
                    slots.append(self._slot_of[seq_id])
        if self.mirrors and slots:
            idx = torch.tensor(slots, dtype=torch.long, device=self.device)
            self.dev_seq_lens.index_add_(
                0, idx, torch.full((len(slots),), -1, dtype=torch.int32,
                                   device=self.device))

    def adopt_prefix(self, seq_id: str, src_seq_id: str, n_tokens: int) -> None:
        """Share the first n_tokens (a whole number of pages) of src with an
        EMPTY sequence — copy-on-write prefix sharing for agents with a
        common system prompt. The adopter's page table points at the shared
        pages; its subsequent appends land past the shared region (length
        starts at n_tokens, a page boundary), so the shared pages stay
        read-only. Refcounts keep them alive until the last owner frees."""
        with self._lock:
            src = self._seqs[src_seq_id]
            dst = self._seqs[seq_id]
            if dst.length != 0:
                raise ValueError(f"adopt_prefix: {seq_id} is not empty")
            if n_tokens % self.page_size != 0 or n_tokens <= 0:
                raise ValueError("adopt_prefix: n_tokens must be whole pages")
            n_pg = n_tokens // self.page_size
            if src.length < n_tokens or len(src.pages) < n_pg:
                raise ValueError("adopt_prefix: source prefix too short")
            shared = src.pages[:n_pg]
            for p in shared:
                self._refs[p] = self._refs.get(p, 1) + 1
            # pages already attached at length 0 are admission reservations
            # (reserve()) sized for the FULL prompt; the shared prefix now
            # supplies the first n_pg pages' worth, so release exactly that
            # surplus and keep the rest as the post-prefix tail
            tail = dst.pages
            for _ in range(min(n_pg, len(tail))):
                self._free.append(tail.pop())
            dst.pages = list(shared) + tail
            dst.length = n_tokens
            if self.mirrors:
                slot = self._slot_of[seq_id]
                self.dev_page_table[slot, :len(dst.pages)] = torch.tensor(
                    dst.pages, dtype=torch.int32, device=self.device)
                self.dev_seq_lens[slot] = n_tokens

    def _release_pages(self, pages: List[int]) -> None:
        """Return pages to the pool, honoring shared-page refcounts. Caller
        holds the lock."""
        for p in reversed(pages):
            c = self._refs.get(p)
            if c is None:
                self._free.append(p)       # sole owner — really free
            elif c <= 2:
                del self._refs[p]          # one owner left; unshared again
            else:
                self._refs[p] = c - 1

    def free_seq(self, seq_id: str) -> None:
        with self._lock:
            s = self._seqs.pop(seq_id, None)
            if s:
                self._release_pages(s.pages)
            slot = self._slot_of.pop(seq_id, None)
            if slot is not None:
                self._seq_of_slot.pop(slot, None)
                if self.mirrors:
                    self.dev_seq_lens[slot] = -1
                self._free_slots.append(slot)

    def reset_seq(self, seq_id: str) -> None:
        """Drop a sequence's KV but keep it registered (context truncation)."""
        with self._lock:
            s = self._seqs[seq_id]
            self._release_pages(s.pages)
            s.pages = []
            s.length = 0
            if self.mirrors:
                self.dev_seq_lens[self._slot_of[seq_id]] = 0

    def reserve(self, seq_id: str, n_tokens: int) -> None:
        """Pre-allocate pages covering n_tokens future appends (chunked
        prefill admission: room for the whole prompt+generation is claimed
        up front, so other agents admitted between chunks cannot starve a
        half-prefilled sequence into OutOfPages mid-plan). Raises
        OutOfPages atomically — no pages are taken on failure."""
        with self._lock:
            s = self._seqs[seq_id]
            need_pages = -(-(s.length + n_tokens) // self.page_size)
            if need_pages > self.max_pages_per_seq:
                raise OutOfPages("sequence would exceed max_pages_per_seq")
            short = need_pages - len(s.pages)
            if short <= 0:
                return
            if short > len(self._free):
                raise OutOfPages(f"KV pool exhausted ({self.n_pages} pages)")
            start = len(s.pages)
            taken = self._free[-short:][::-1]  # same order as a pop() loop
            del self._free[-short:]
            s.pages.extend(taken)
            if self.mirrors:
                # one batched mirror write — per-page scalar assigns cost
                # a tiny H2D launch EACH (visible in the admit phase)
                self.dev_page_table[self._slot_of[seq_id],
                                    start:start + short] = torch.tensor(
                    taken, dtype=torch.int32, device=self.device)

    def can_append(self, seq_id: str, n_tokens: int) -> bool:
        with self._lock:
            s = self._seqs[seq_id]
            have = len(s.pages) * self.page_size - s.length
            need_pages = max(0, -(-(n_tokens - have) // self.page_size))
            return need_pages <= len(self._free)

    def can_append_after_reset(self, seq_id: str, n_tokens: int) -> bool:
        """Would n_tokens fit if this sequence's pages were freed first?
        Shared (refcounted) pages don't return to the pool on reset."""
        with self._lock:
            s = self._seqs[seq_id]
            need_pages = -(-n_tokens // self.page_size)
            freeable = sum(1 for p in s.pages if p not in self._refs)
            return need_pages <= len(self._free) + freeable

    def append_slots(self, seq_id: str, n_tokens: int) -> List[int]:
        """Reserve slots for n_tokens new tokens; allocates pages as needed.
        Returns global slot ids (page * page_size + offset)."""
        with self._lock:
            s = self._seqs[seq_id]
            slots = []
            for _ in range(n_tokens):
                if s.length == len(s.pages) * self.page_size:
                    if not self._free:
                        raise OutOfPages(
                            f"KV pool exhausted ({self.n_pages} pages)")
                    s.pages.append(self._free.pop())
                page = s.pages[s.length // self.page_size]
                off = s.length % self.page_size
                slots.append(page * self.page_size + off)
                s.length += 1
            return slots

    def page_table(self, seq_ids: List[str], device=None) -> torch.Tensor:
        """int32 [B, max_pages_in_batch] (padded with 0)."""
        with self._lock:
            rows = [self._seqs[sid].pages for sid in seq_ids]
        width = max((len(r) for r in rows), default=1) or 1
        t = torch.zeros(len(rows), width, dtype=torch.int32)
        for i, r in enumerate(rows):
            if r:
                t[i, :len(r)] = torch.tensor(r, dtype=torch.int32)
        return t.to(device or self.device)

    def seq_lens(self, seq_ids: List[str], device=None) -> torch.Tensor:
        with self._lock:
            lens = [self._seqs[sid].length for sid in seq_ids]
        return torch.tensor(lens, dtype=torch.int32, device=device or self.device)

    # ---------- checkpoint (stop/resume, crash recovery) ----------

    def offload(self, seq_id: str, free: bool = True) -> Optional[KVCheckpoint]:
        """Gather the sequence's pages and stream them to pinned host memory."""
        with self._lock:
            s = self._seqs.get(seq_id)
            if s is None or not s.pages:
                if free and s is not None:
                    self.free_seq(seq_id)
                return None
            pages = list(s.pages)
            length = s.length
        n = len(pages)
        page_ids = torch.tensor(pages, dtype=torch.int32, device=self.device)
        is_gpu = str(self.device) not in ("cpu",)
        host = torch.empty(self.n_layers, n, self.page_shorts, dtype=self.dtype,
                           pin_memory=is_gpu)
        if is_gpu:
            staging = torch.empty(n, self.page_shorts, dtype=self.dtype,
                                  device=self.device)
            for li in range(self.n_layers):
                ops.gather_kv_pages(staging, self.k_caches[li], self.v_caches[li],
                                    page_ids)
                host[li].copy_(staging, non_blocking=True)
            torch.cuda.synchronize()
        else:
            staging = torch.empty(n, self.page_shorts, dtype=self.dtype)
            for li in range(self.n_layers):
                ops.gather_kv_pages(staging, self.k_caches[li], self.v_caches[li],
                                    page_ids)
                host[li].copy_(staging)
        if free:
            self.free_seq(seq_id)
        return KVCheckpoint(length=length, n_pages=n, data=host)

    def restore(self, seq_id: str, ckpt: Optional[KVCheckpoint]) -> None:
        """Upload a checkpoint into freshly allocated pages."""
        with self._lock:
            if seq_id in self._seqs:
                self.free_seq(seq_id)
            s = self.create_seq(seq_id)
            if ckpt is None or ckpt.n_pages == 0:
                return
            if len(self._free) < ckpt.n_pages:
                raise OutOfPages("not enough pages to restore checkpoint")
            s.pages = [self._free.pop() for _ in range(ckpt.n_pages)]
            s.length = ckpt.length
            pages = list(s.pages)
        page_ids = torch.tensor(pages, dtype=torch.int32, device=self.device)
        is_gpu = str(self.device) not in ("cpu",)
        for li in range(self.n_layers):
            staging = ckpt.data[li].to(self.device, non_blocking=is_gpu)
            ops.scatter_kv_pages(self.k_caches[li], self.v_caches[li],
                                 staging, page_ids)
        self.push_dev(seq_id)
        if is_gpu:
            torch.cuda.synchronize()
