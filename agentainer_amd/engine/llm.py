"""LLMEngine — the in-process multi-tenant inference engine.

This is what replaces the reference's Docker data plane (SURVEY.md north
star): each attached agent is a (shared) model binding plus a private
conversation KV sequence in the paged HBM pool. The continuous-batching
scheduler below is the analog of the reference's request queue + replay
worker feeding containers (replay_worker.go:57-117): admission comes from
the per-model request queue, prefill is batched under a token budget,
decode runs one batched step for every running sequence.

Structure per model (agents SHARE one weight set — BASELINE config 3):
  ModelInstance: weights + KVCacheManager + waiting/running queues +
  one engine thread running steps.

Lifecycle mapping:
  attach   = bind shard (refcounted load) + create/restore KV sequence
  pause    = close admission; KV stays resident
  detach   = drain + offload KV to pinned host (+ disk for restarts)
  chat     = enqueue -> scheduler -> response (greedy => deterministic, so
             at-least-once crash replay regenerates identical output)
"""

from __future__ import annotations

import os
import queue
import threading
import time
import traceback
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

import torch

from .. import ops
from .. import parallel as par
from ..store import Store
from .base import ModelNotFound
from .kvcache import KVCacheManager, KVCheckpoint, OutOfPages
from .tokenizer import ByteTokenizer
from ..models.llama import (AttnMetadata, LLAMA_CONFIGS, LlamaConfig,
                            LlamaForCausalLM)

DEFAULT_MAX_NEW = 64

# int-vector op codes on the TP plan channel (parallel.dist.send_ints);
# 0 is reserved for pickled commands
OP_DECODE = 1   # [OP, model_idx, B, slot*B, token*B] — the per-step hot path
OP_BARRIER = 2  # [OP] — timing fence (bench.py --tp)
# async (speculative) decode: rollbacks from the previous resolve are
# prepended, then per-row (slot, token|-1, prev_idx, eff_seed, temp_bits,
# top_p_bits) — workers feed carried-over rows from their OWN previous
# device sample (logits are bitwise-identical across ranks after the
# all-reduce, and sampling is seed-deterministic, so every rank samples
# the same token without any extra communication)
OP_DECODE_ASYNC = 3  # [OP, model_idx, B, nrb, rb_slot*nrb, row*6*B]
OP_ROLLBACK = 4      # [OP, model_idx, n, slot*n] — flush before other cmds


def _f2i(x: float) -> int:
    """float32 bit pattern as int (plan-channel encoding)."""
    import struct
    return struct.unpack("<I", struct.pack("<f", float(x)))[0]


def _i2f(x: int) -> float:
    import struct
    return struct.unpack("<f", struct.pack("<I", int(x) & 0xFFFFFFFF))[0]


class EngineDead(RuntimeError):
    pass


@dataclass(eq=False)
class GenRequest:
    agent_id: str
    prompt_tokens: List[int]
    max_new: int
    temperature: float
    top_p: float
    seed: int
    done: threading.Event = field(default_factory=threading.Event)
    generated: List[int] = field(default_factory=list)
    error: Optional[str] = None
    enq_t: float = 0.0
    first_token_t: float = 0.0
    fin_t: float = 0.0
    trace_id: str = ""  # WAL request id — end-to-end tracing
    # streaming: the scheduler puts each sampled token id here as it is
    # produced, then None when the request finishes (SSE /chat path)
    stream_q: Optional[Any] = None
    # chunked prefill: tokens of the prompt already written to KV, and the
    # length of the slice admitted for the CURRENT step (set by _admit)
    prefill_pos: int = 0
    slice_len: int = 0


@dataclass
class AgentBinding:
    agent: Any
    model_name: str
    seq_id: str
    paused: bool = False
    requests: int = 0
    tokens: int = 0
    # /clear: the next admitted request prefills onto a reset KV sequence
    # (plan-mirrored to TP workers via needs_reset)
    pending_reset: bool = False
    queue: "queue.Queue[GenRequest]" = field(default_factory=queue.Queue)
    active: Optional[GenRequest] = None
    # whole-page token prefix shared with other agents (system prompt),
    # None when the system prompt is shorter than one KV page
    prefix_tokens: Optional[List[int]] = None


class ModelInstance:
    """One loaded model + KV pool + scheduler thread."""

    def __init__(self, name: str, cfg: LlamaConfig, device: str,
                 engine_cfg: Dict[str, Any], weights_path: Optional[str] = None,
                 model_cls=None):
        self.name = name
        self.cfg = cfg
        self.device = device
        tok_json = (os.path.join(weights_path, "tokenizer.json")
                    if weights_path else None)
        if tok_json and os.path.exists(tok_json):
            from .tokenizer import HFTokenizer
            self.tokenizer = HFTokenizer.from_checkpoint(weights_path)
        else:
            self.tokenizer = ByteTokenizer(cfg.vocab_size)
        # tensor parallelism: engine-level degree must match the launched
        # world size; rank 0 schedules, plans broadcast (parallel.dist)
        self.tp_size = int(engine_cfg.get("tp_degree", 1))
        self.tp_rank = par.tp_rank() if self.tp_size > 1 else 0
        if self.tp_size > 1:
            assert par.is_tp() and par.tp_size() == self.tp_size, (
                f"tp_degree={self.tp_size} requires torchrun world of that size")
        model_cls = model_cls or LlamaForCausalLM
        self.model = model_cls(cfg, device=device,
                               tp_rank=self.tp_rank, tp_size=self.tp_size)
        if self.tp_size > 1:
            self.model.tp_group = par.tp_group()
        if weights_path:
            self.model.load_safetensors(weights_path)
        if device.startswith("cuda"):
            # packed decode copies add memory on top of the bf16 weights:
            # x1 for a bf16 repack, x0.5 for fp8 experts, x0.25 for MXFP4 —
            # size the guard by what is actually being packed (the old
            # flat x2 guard silently skipped quantization for Mixtral-8x7B)
            param_bytes = sum(p.numel() * p.element_size()
                              for p in self.model.parameters())
            fp8 = bool(engine_cfg.get("expert_fp8", False))
            fp4 = bool(engine_cfg.get("expert_fp4", False))
            dq = str(engine_cfg.get("dense_quant", "") or "")
            if dq not in ("", "fp8", "mxfp4"):
                raise ValueError(f"unsupported dense_quant {dq!r} "
                                 "(expected fp8 or mxfp4)")
            factor = (0.25 if (fp4 or dq == "mxfp4")
                      else 0.5 if (fp8 or dq == "fp8") else 1.0)
            free, total = torch.cuda.mem_get_info()
            if param_bytes * (1.0 + factor) < 0.75 * total:
                import inspect
                sig = inspect.signature(self.model.pack_decode_weights)
                if "expert_fp8" in sig.parameters:
                    self.model.pack_decode_weights(expert_fp8=fp8,
                                                   expert_fp4=fp4)
                elif "dense_quant" in sig.parameters:
                    self.model.pack_decode_weights(dense_quant=dq)
                else:
                    self.model.pack_decode_weights()
        page_size = int(engine_cfg.get("kv_page_size", 16))
        kv_dtype_s = str(engine_cfg.get("kv_dtype", "bf16"))
        if kv_dtype_s in ("fp8", "fp8_e4m3", "e4m3"):
            kv_dtype = torch.float8_e4m3fn  # halves KV bytes: 2x the agents
        elif kv_dtype_s in ("bf16", "bfloat16"):
            kv_dtype = torch.bfloat16
        else:
            raise ValueError(f"unsupported kv_dtype {kv_dtype_s!r}")
        n_pages = self._pool_pages(cfg, page_size, device, engine_cfg,
                                   self.tp_size)
        # sequence-slot budget: device page-table/length mirrors are tiny
        # (2048 slots x 512 pages x 4 B = 4 MB), so default well above the
        # largest measured concurrency (1k+ agents per GPU with fp8 KV)
        max_seqs = int(engine_cfg.get("max_seqs", 2048))
        self.kvm = KVCacheManager(cfg.n_layers,
                                  max(cfg.n_kv_heads // self.tp_size, 1),
                                  cfg.head_dim, page_size, n_pages,
                                  device=device, dtype=kv_dtype,
                                  max_slots=max_seqs,
                                  mirrors=(device.startswith("cuda") or bool(
                                      engine_cfg.get("async_decode_emulate",
                                                     False))))
        self.max_batch_tokens = int(engine_cfg.get("max_batch_tokens", 8192))
        self.max_decode_batch = int(engine_cfg.get("max_decode_batch", 256))
        self.refcount = 0
        # prefix sharing (copy-on-write system prompts): token-tuple ->
        # shared-prefix sequence id; populated lazily by the first prefill
        # that carries a registered prefix (plan-driven, so TP ranks stay
        # in lockstep)
        self.prefix_sharing = bool(engine_cfg.get("prefix_sharing", True))
        self._prefixes: Dict[tuple, str] = {}
        self.waiting: "queue.Queue[GenRequest]" = queue.Queue()
        # long prompts mid-chunked-prefill: continue before new admissions
        self._chunking: List[GenRequest] = []
        self.running: List[GenRequest] = []
        self._running_set = set()
        self._bindings: Dict[str, AgentBinding] = {}
        self._lock = threading.RLock()
        # serializes whole engine steps against lifecycle ops (bind/unbind
        # drain the speculative step and mutate the KV pool — they must
        # never interleave with a step on the engine thread)
        self._step_mutex = threading.Lock()
        self._stop = threading.Event()
        self._wake = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.steps = 0
        self.decode_tokens = 0
        self.prefill_tokens = 0
        self.occupancy_acc = 0.0
        self.sync_mode = bool(engine_cfg.get("sync_mode", False))
        # engine watchdog (SURVEY.md §5 failure detection): a step that has
        # been in flight longer than this marks the instance unhealthy
        # (stuck kernel / GPU fell off the bus) so the health monitor's
        # auto-restart path fires
        self.watchdog_timeout_s = float(engine_cfg.get("watchdog_timeout_s", 120.0))
        self._step_started: Optional[float] = None
        self.last_step_t = time.time()
        # hipGraph-captured decode step (SURVEY.md north star): the whole
        # batched decode forward — page-table gather, slot computation,
        # RoPE/append/attention/GEMMs, device-side length increment — is
        # captured once per batch bucket and replayed with two small H2D
        # copies (row slots + last tokens) per step.
        self.is_gpu = device.startswith("cuda")
        # hipGraph capture now includes the TP case: the captured decode
        # body contains the per-sublayer RCCL all-reduces (SURVEY.md §7.3
        # names exactly this); capture failure falls back to eager replay
        # (_get_graph) identically on every rank.
        self.use_graph = (self.is_gpu
                          and bool(engine_cfg.get("graph_capture", True)))
        # test-only: run the device-decode machinery (mirrors, slot-fed
        # decode, async protocol) on CPU tensors so gloo world-2 tests can
        # validate the exact code path the GPU runs
        self._emulate_dev = (not self.is_gpu
                             and bool(engine_cfg.get("async_decode_emulate",
                                                     False)))
        self.dev_decode = self.is_gpu or self._emulate_dev
        # async decode (speculative one-step lag): the next decode step is
        # launched with device-fed tokens BEFORE the previous step's tokens
        # reach the host, so per-step host bookkeeping overlaps GPU work.
        # Rows that resolve as finished get their speculative append rolled
        # back (kvm.rollback_many, stream-ordered). Under TP, rank 0
        # broadcasts OP_DECODE_ASYNC plans; every rank samples on-device
        # from its own (identical) logits, so the carried-over token feed
        # needs no cross-rank traffic.
        self.async_decode = ((self.use_graph or self._emulate_dev)
                             and bool(engine_cfg.get("async_decode", True)))
        # index of this model in the engine's creation order — the wire
        # identity in binary step plans (identical on every TP rank)
        self._model_idx = 0
        # rollback slots applied locally but not yet mirrored to workers
        # (flushed at the head of the next broadcast, any kind)
        self._pending_rb: List[int] = []
        self._phase_ms: Dict[str, float] = {}
        # deferred prefill sample (async engines): (logits, final_reqs)
        self._pf_pending: Optional[tuple] = None
        # worker-side speculative mirror: previous step's device sample
        self._spec_worker: Optional[Dict[str, Any]] = None
        self._spec: Optional[Dict[str, Any]] = None
        self._graphs: Dict[int, Dict[str, Any]] = {}
        # prefill runs on its own HIP stream so a prefill batch overlaps
        # the in-flight decode graph replay (prefill is MFMA-bound, decode
        # is HBM-bound — they share the chip well). Safe by construction:
        # a sequence is either prefilling or decoding, never both, so the
        # two streams touch disjoint pages/page-table rows; the only
        # cross-stream readers are speculatively-launched rows that are
        # already invalidated (their outputs are discarded).
        self._prefill_stream = (torch.cuda.Stream()
                                if self.is_gpu
                                and bool(engine_cfg.get("prefill_stream", True))
                                else None)
        # set whenever pages were freed (reset/rollback) while a decode
        # step may still be in flight: the next side-stream prefill must
        # order itself after that decode so a speculative K/V write can't
        # land in a page the prefill just re-allocated
        self._pages_freed = False
        self._pad_slot = -1
        if self.dev_decode:
            # reserve one sequence slot as the graph's pad row target
            self._pad_slot = self.kvm._free_slots.pop()
            self.kvm.dev_seq_lens[self._pad_slot] = -1
            self._pad_slot_t = torch.tensor([self._pad_slot], dtype=torch.long,
                                            device=device)

    @staticmethod
    def _pool_pages(cfg: LlamaConfig, page_size: int, device: str,
                    engine_cfg: Dict[str, Any], tp_size: int = 1) -> int:
        eb = 1 if str(engine_cfg.get("kv_dtype", "bf16")).startswith(("fp8", "e4m3")) else 2
        bytes_per_page = cfg.kv_bytes_per_token(eb) * page_size // max(tp_size, 1)
        pool_gb = float(engine_cfg.get("kv_pool_gb", 0.0))
        if device.startswith("cuda") and pool_gb <= 0:
            free, _total = torch.cuda.mem_get_info()
            pool_bytes = int(free * 0.80)  # leave headroom for activations
        elif pool_gb > 0:
            pool_bytes = int(pool_gb * (1 << 30))
        else:
            pool_bytes = 64 << 20  # CPU tests: 64 MB
        return max(8, pool_bytes // bytes_per_page)

    # ---------- scheduler thread ----------

    def start(self):
        if self.sync_mode:
            return  # bench/test harnesses drive step() directly
        if self._thread is None:
            self._stop.clear()
            self._thread = threading.Thread(target=self._loop,
                                            name=f"engine-{self.name}", daemon=True)
            self._thread.start()

    def stop(self):
        if self.async_decode:
            self.drain_async()
        self._stop.set()
        self._wake.set()
        if self._thread is not None:
            self._thread.join(timeout=10.0)
            self._thread = None

    def alive(self) -> bool:
        if self.stuck():
            return False
        if self.sync_mode:
            return True
        return self._thread is not None and self._thread.is_alive()

    def stuck(self) -> bool:
        """Watchdog verdict: one engine step exceeding the timeout."""
        st = self._step_started
        return st is not None and (time.time() - st) > self.watchdog_timeout_s

    def _loop(self):
        # AGENTAINER_ENGINE_PROFILE=/path.txt: cProfile the first ~800
        # engine steps and dump cumulative stats (diagnosing host-side
        # step cost under serving load)
        prof_path = os.environ.get("AGENTAINER_ENGINE_PROFILE", "")
        prof = None
        prof_steps = 0
        if prof_path:
            import cProfile
            prof = cProfile.Profile()
        while not self._stop.is_set():
            try:
                with self._step_mutex:
                    if prof is not None:
                        prof.enable()
                    did = self.step()
                    if prof is not None:
                        prof.disable()
                        prof_steps += 1 if did else 0
                        if prof_steps == 800:
                            import io
                            import pstats
                            s = io.StringIO()
                            pstats.Stats(prof, stream=s).sort_stats(
                                "cumulative").print_stats(40)
                            with open(prof_path, "w") as f:
                                f.write(s.getvalue())
                            prof = None
            except Exception:
                traceback.print_exc()
                did = False
            if not did:
                self._wake.wait(timeout=0.005)
                self._wake.clear()

    # ---------- agent-side API ----------

    def submit(self, req: GenRequest):
        req.enq_t = time.time()
        self.waiting.put(req)
        self._wake.set()

    # ---------- one engine step ----------

    def step(self) -> bool:
        """Admit + prefill, then one decode step. Returns True if work ran."""
        self._step_started = time.time()
        if self.async_decode:
            ran = self._step_async()
            self.steps += 1 if ran else 0
            if ran:  # EMA of wall step time (engine-side diagnostics)
                dt = (time.time() - self._step_started) * 1000.0
                self._step_ms_ema = (0.05 * dt
                                     + 0.95 * getattr(self, "_step_ms_ema", dt))
            self._step_started = None
            self.last_step_t = time.time()
            return ran
        if self._spec is not None:
            # async was disabled mid-flight (graph-capture fallback):
            # resolve the orphaned speculative step before going eager
            self.drain_async()
        admitted = self._admit()
        if admitted:
            try:
                self._prefill(admitted)
            except OutOfPages as e:
                self._fail_prefill(admitted, str(e))
        ran_decode = False
        with self._lock:
            batch = [r for r in self.running if not r.done.is_set()]
        if batch:
            if len(batch) > self.max_decode_batch:
                batch = batch[: self.max_decode_batch]
                self._rotate_running(batch)
            self._decode(batch)
            ran_decode = True
        self.steps += 1 if (admitted or ran_decode) else 0
        self._step_started = None
        self.last_step_t = time.time()
        return bool(admitted) or ran_decode

    # ---------- async (speculative) decode step ----------

    def _phase_mark(self, key: str, t0: float) -> float:
        """EMA per-phase wall times (stats diagnostics); returns now."""
        now = time.time()
        ema = self._phase_ms.get(key, 0.0)
        self._phase_ms[key] = 0.05 * (now - t0) * 1000.0 + 0.95 * ema
        return now

    def _step_async(self) -> bool:
        kvm = self.kvm
        dev = self.device
        t0 = time.time()
        # phase 0: publish last step's deferred prefill sample, so those
        # rows are in `running` for this step's snapshot (same step they
        # would have joined under synchronous sampling)
        had_pf = self._pf_pending is not None
        self._resolve_prefill()
        t0 = self._phase_mark("p0_pf_resolve_ms", t0)
        # phase 1: LAUNCH the next decode for the current running set; old
        # rows' input tokens come straight from the previous step's device
        # sample buffer (their host values are not resolved yet)
        with self._lock:
            batch = [r for r in self.running if not r.done.is_set()]
        launched = None
        if batch:
            bucket = min(self._bucket(len(batch)), max(self.max_decode_batch, 1))
            if len(batch) > bucket:
                batch = batch[:bucket]
                self._rotate_running(batch)
            B = len(batch)
            seq_ids = [self._bindings[r.agent_id].seq_id for r in batch]
            rows = kvm.decode_batch_prepare(seq_ids)
            prev = self._spec
            pos = {id(r): i for i, r in enumerate(prev["reqs"])} if prev else {}
            idx = [pos.get(id(r), -1) for r in batch]
            # effective sampling seed: a carried-over row has ONE unresolved
            # token in flight ahead of this sample (generated is appended at
            # resolve, one step later), so its counter runs one ahead of
            # len(generated) — this keeps the sampled stream identical to
            # the synchronous path's seed+len(generated) sequence
            seeds_eff = [(r.seed + len(r.generated)
                          + (1 if idx[i] >= 0 else 0)) & 0x7FFFFFFFFFFFFFFF
                         for i, r in enumerate(batch)]
            if self.tp_size > 1 and self.tp_rank == 0:
                # async plan: pending rollbacks + per-row feed/sample spec;
                # workers replicate the launch and sample on-device from
                # their own identical logits (no return traffic)
                rb, self._pending_rb = self._pending_rb, []
                vec = [OP_DECODE_ASYNC, self._model_idx, B, len(rb)] + rb
                for i, r in enumerate(batch):
                    vec += [rows[i],
                            r.generated[-1] if idx[i] < 0 else -1,
                            idx[i],
                            seeds_eff[i],
                            _f2i(r.temperature), _f2i(r.top_p)]
                par.send_ints(vec)
            entry = self._get_graph(bucket)
            # the PREVIOUS step's non-blocking H2D copies read these pinned
            # staging buffers; wait for that DMA before rewriting them (a
            # torn read would feed wrong slots/tokens into the graph)
            ev = entry.get("h2d_ev") if self.is_gpu else None
            if ev is not None:
                ev.synchronize()
            entry["rows_pin"][:B] = torch.tensor(rows, dtype=torch.long)
            entry["rows_pin"][B:] = self._pad_slot
            entry["rows"].copy_(entry["rows_pin"], non_blocking=True)
            entry["ids_pin"][:B] = torch.tensor(
                [r.generated[-1] if idx[i] < 0 else 0
                 for i, r in enumerate(batch)], dtype=torch.long)
            entry["ids"][:B].copy_(entry["ids_pin"][:B], non_blocking=True)
            if self.is_gpu:
                if ev is None:
                    ev = entry["h2d_ev"] = torch.cuda.Event()
                ev.record()
            if prev is not None and any(x >= 0 for x in idx):
                gidx = torch.tensor([max(x, 0) for x in idx], dtype=torch.long,
                                    device=dev)
                mask = torch.tensor([x >= 0 for x in idx], dtype=torch.bool,
                                    device=dev)
                gathered = prev["sampled"].index_select(0, gidx)
                entry["ids"][:B].copy_(
                    torch.where(mask, gathered, entry["ids"][:B]))
            if entry["graph"] is not None:
                entry["graph"].replay()
                logits = entry["logits"][:B]
            else:  # eager fallback (capture failure / CPU emulation)
                logits = self._device_decode_fwd(entry["rows"], entry["ids"],
                                                 entry["inc"])[:B]
            sampled = self._sample_device_params(
                logits, [r.temperature for r in batch],
                [r.top_p for r in batch], seeds_eff)
            kvm.advance_many(seq_ids)
            self.decode_tokens += B
            self.occupancy_acc += B / max(1, self.max_decode_batch)
            launched = {"reqs": batch, "sampled": sampled, "invalid": set()}
        t0 = self._phase_mark("p1_launch_ms", t0)
        prev = self._spec
        self._spec = launched

        # phase 2: RESOLVE the previous step's tokens on the host while the
        # GPU runs the step launched above
        self._resolve_spec(prev, launched)
        t0 = self._phase_mark("p2_resolve_ms", t0)

        # phase 3: admission + prefill, overlapping the decode replay
        # launched in phase 1 (side stream; fully host-synced by the
        # prefill sampling before this returns)
        admitted = self._admit()
        t0 = self._phase_mark("p3_admit_ms", t0)
        if admitted:
            if self._prefill_stream is not None:
                if self._pages_freed:
                    # pages freed by a reset/rollback may be re-allocated by
                    # this prefill while the in-flight decode still writes
                    # its (rolled-back) speculative K/V there — serialize
                    # the streams for this one step
                    self._prefill_stream.wait_stream(torch.cuda.current_stream())
                    self._pages_freed = False
                try:
                    with torch.cuda.stream(self._prefill_stream):
                        self._prefill(admitted)
                except OutOfPages as e:
                    self._fail_prefill(admitted, str(e))
            else:
                try:
                    self._prefill(admitted)
                except OutOfPages as e:
                    self._fail_prefill(admitted, str(e))
            self._phase_mark("p3_prefill_ms", t0)
        return bool(batch) or bool(admitted) or prev is not None or had_pf

    def _rotate_running(self, served: List[GenRequest]) -> None:
        """Round-robin fairness when the running set exceeds the decode
        batch cap: move the rows just served to the tail so oversubscribed
        pools don't starve the tail indefinitely (measured p99 34.6 s at
        512 agents against the 256-row default before this)."""
        with self._lock:
            sids = set(map(id, served))
            self.running = ([r for r in self.running if id(r) not in sids]
                            + [r for r in self.running if id(r) in sids])

    def _fail_prefill(self, reqs: List[GenRequest], msg: str) -> None:
        """Admission reserves full KV room, so a prefill-time OutOfPages
        should be impossible — but if one ever escapes, fail the batch's
        requests EXPLICITLY (clients see the error instead of hanging to
        timeout) and reset their sequences to a consistent state."""
        with self._lock:
            for r in reqs:
                if r in self._chunking:
                    self._chunking.remove(r)
                if r in self._running_set:
                    self.running.remove(r)
                    self._running_set.discard(r)
                if not r.done.is_set():
                    r.error = f"KV pool exhausted: {msg}"
                    if r.stream_q is not None:
                        r.stream_q.put(None)
                    r.done.set()
                b = self._bindings.get(r.agent_id)
                if b is not None:
                    if b.active is r:
                        b.active = None
                    if self.kvm.has_seq(b.seq_id):
                        self.kvm.reset_seq(b.seq_id)
                        self._pages_freed = True
                    self._pump_agent(b)

    def _resolve_spec(self, prev, launched) -> None:
        if prev is None:
            return
        toks = prev["sampled"].tolist()
        rollback_seqs = []
        with self._lock:
            lpos = ({id(r): i for i, r in enumerate(launched["reqs"])}
                    if launched else {})
            for i, r in enumerate(prev["reqs"]):
                if i in prev["invalid"] or r.done.is_set():
                    # invalidated row, or the request died (detach) mid-flight
                    continue
                t = int(toks[i])
                r.generated.append(t)
                self._finish_or_run(r, t)
                if r.done.is_set():
                    # the just-launched step speculatively appended a token
                    # for this row: mark invalid + roll back that append
                    j = lpos.get(id(r))
                    if j is not None:
                        launched["invalid"].add(j)
                        b = self._bindings.get(r.agent_id)
                        if b is not None:
                            rollback_seqs.append(b.seq_id)
        if rollback_seqs:
            self._pages_freed = True  # see _step_async phase 3
            if self.tp_size > 1 and self.tp_rank == 0:
                # applied locally now; workers get them at the head of the
                # NEXT broadcast (async plan or any other command)
                self._pending_rb.extend(self.kvm.slot(s)
                                        for s in rollback_seqs)
        self.kvm.rollback_many(rollback_seqs)

    def drain_async(self) -> None:
        """Resolve the in-flight speculative step AND any deferred prefill
        WITHOUT launching new work (called before detach/offload and at
        engine stop, so checkpoints never capture an unresolved token)."""
        self._resolve_prefill()
        prev = self._spec
        self._spec = None
        self._resolve_spec(prev, None)

    def _admit(self) -> List[GenRequest]:
        out: List[GenRequest] = []
        budget = self.max_batch_tokens
        # chunked-prefill continuations first: their KV room is already
        # reserved and decode is waiting on them
        while self._chunking and budget > 0:
            r = self._chunking.pop(0)
            r.slice_len = min(len(r.prompt_tokens) - r.prefill_pos, budget)
            budget -= r.slice_len
            out.append(r)
        while budget > 0:
            try:
                req = self.waiting.get_nowait()
            except queue.Empty:
                break
            b = self._bindings.get(req.agent_id)
            if b is None:
                req.error = "agent detached"
                req.done.set()
                continue
            need = len(req.prompt_tokens)
            cap = min(self.kvm.max_pages_per_seq * self.kvm.page_size,
                      self.cfg.max_position)
            if need + req.max_new > cap:
                req.error = f"prompt too long ({need} tokens; cap {cap})"
                req.done.set()
                continue
            if need > budget and out:
                # put back; try next step
                self.waiting.put(req)
                break
            # a prompt above the whole step budget is admitted CHUNKED:
            # this step prefills the first max_batch_tokens of it, the
            # remainder continues next step (decode in between — long
            # prompts no longer stall the whole batch)
            req.slice_len = min(need, budget)
            # KV room: prompt + generation. Context truncation fires on
            # pool pressure, on the model's rope/position horizon (a
            # conversation crossing max_position would step the rope
            # table out of bounds), and on an explicit /clear
            req.needs_reset = (
                b.pending_reset
                or self.kvm.seq_len(b.seq_id) + need + req.max_new
                > self.cfg.max_position
                or not self.kvm.can_append(b.seq_id, need + req.max_new))
            if req.needs_reset:
                # truncate conversation NOW (rank 0); the prefill plan
                # carries the flag so TP workers mirror the reset
                self.kvm.reset_seq(b.seq_id)
                self._pages_freed = True  # in-flight-step hazard, see _step_async
            try:
                # claim pages for the whole prompt + generation (+1 for a
                # speculative async append) up front: admission is the ONLY
                # place the KV pool can reject work, so a half-prefilled
                # chunked prompt can never die OutOfPages mid-plan and the
                # decode path never allocates under pressure
                self.kvm.reserve(b.seq_id, need + req.max_new + 1)
            except OutOfPages:
                req.error = "KV pool exhausted"
                req.done.set()
                continue
            b.pending_reset = False
            out.append(req)
            budget -= req.slice_len
        return out

    def _sample(self, logits: torch.Tensor, reqs: List[GenRequest]) -> List[int]:
        return self._sample_device(logits, reqs).tolist()

    def _sample_device(self, logits: torch.Tensor,
                       reqs: List[GenRequest]) -> torch.Tensor:
        return self._sample_device_params(
            logits,
            [r.temperature for r in reqs],
            [r.top_p for r in reqs],
            [(r.seed + len(r.generated)) & 0x7FFFFFFFFFFFFFFF for r in reqs])

    def _sample_device_params(self, logits: torch.Tensor,
                              temps_l: List[float], tops_l: List[float],
                              seeds_l: List[int]) -> torch.Tensor:
        """Param-vector core shared by rank 0 (from GenRequests) and TP
        workers (from the async plan): given bitwise-identical logits and
        the same (temp, top_p, effective-seed) rows, every rank samples
        the same tokens."""
        B = logits.size(0)
        out = torch.empty(B, dtype=torch.long, device=logits.device)
        greedy_rows = [i for i in range(B) if temps_l[i] <= 0.0]
        samp_rows = [i for i in range(B) if temps_l[i] > 0.0]
        lb = logits.to(torch.bfloat16).contiguous()
        if greedy_rows:
            if len(greedy_rows) == B:
                ops.greedy_sample(out, lb)
            else:
                sub = torch.empty(len(greedy_rows), dtype=torch.long,
                                  device=logits.device)
                ops.greedy_sample(sub, lb[greedy_rows].contiguous())
                out[greedy_rows] = sub
        if samp_rows:
            sub = torch.empty(len(samp_rows), dtype=torch.long, device=logits.device)
            temps = torch.tensor([temps_l[i] for i in samp_rows],
                                 dtype=torch.float32, device=logits.device)
            tps = torch.tensor([tops_l[i] for i in samp_rows],
                               dtype=torch.float32, device=logits.device)
            seeds = torch.tensor([seeds_l[i] for i in samp_rows],
                                 dtype=torch.int64, device=logits.device)
            ops.topp_sample(sub, lb[samp_rows].contiguous(), temps, tps, seeds)
            out[samp_rows] = sub
        return out

    def _bcast(self, cmd):
        if self.tp_size > 1 and self.tp_rank == 0:
            if self._pending_rb:
                # flush speculative rollbacks FIRST: the command below may
                # read KV state (prefill positions, unbind) that is only
                # consistent across ranks once workers applied them
                rb, self._pending_rb = self._pending_rb, []
                par.send_ints([OP_ROLLBACK, self._model_idx, len(rb)] + rb)
            par.broadcast_obj(cmd)

    def _prefill(self, reqs: List[GenRequest]):
        t0 = time.time()
        c0 = time.thread_time()
        plan = []
        n_extra = 0  # shared-prefix rows prepended to the plan (no GenRequest)
        final = []   # reqs whose slice completes the prompt -> sample
        for r in reqs:
            b = self._bindings[r.agent_id]
            first = r.prefill_pos == 0
            needs_reset = first and bool(getattr(r, "needs_reset", False))
            sl = r.slice_len or (len(r.prompt_tokens) - r.prefill_pos)
            tokens = r.prompt_tokens[r.prefill_pos:r.prefill_pos + sl]
            is_final = (r.prefill_pos + sl) == len(r.prompt_tokens)
            adopt = None
            pk = b.prefix_tokens
            if (pk is not None and first and is_final and not needs_reset
                    and self.kvm.seq_len(b.seq_id) == 0
                    and len(tokens) > len(pk) and tokens[:len(pk)] == pk):
                key = tuple(pk)
                sid = self._prefixes.get(key)
                if sid is None:
                    # first user pays: prefill the shared-prefix sequence as
                    # an extra plan row ahead of its adopter (its K/V append
                    # kernel runs before any attention in the same forward)
                    n_pg = len(pk) // self.kvm.page_size
                    if self.kvm.free_pages >= n_pg + 8:
                        sid = f"\x00pfx:{len(self._prefixes)}"
                        self._prefixes[key] = sid
                        plan.insert(n_extra, (sid, list(pk), False, None))
                        n_extra += 1
                if sid is not None:
                    adopt = (sid, len(pk))
                    tokens = tokens[len(pk):]
            plan.append((b.seq_id, tokens, needs_reset, adopt))
            r.prefill_pos += sl
            final.append(is_final)
        self._bcast(("prefill", self.name, plan))
        t0 = self._phase_mark("pf_plan_ms", t0)
        logits = self._prefill_exec(plan)
        t0 = self._phase_mark("pf_exec_ms", t0)  # launch-side (no sync)
        logits = logits[n_extra:]
        f_rows = [i for i, f in enumerate(final) if f]
        f_reqs = [r for r, f in zip(reqs, final) if f]
        with self._lock:
            for r, f in zip(reqs, final):
                if not f:
                    self._chunking.append(r)  # next slice next step
        if not f_reqs:
            return
        if len(f_rows) < len(reqs):
            logits = logits[f_rows]
        if self.is_gpu and os.environ.get("AGENTAINER_PF_SYNC_PROBE"):
            torch.cuda.synchronize()  # diagnose GPU-wait vs sample cost
            t0 = self._phase_mark("pf_sync_ms", t0)
        if self.async_decode:
            # DEFER the sample: a prefilled row can only join the decode
            # batch at the NEXT step's snapshot anyway, so resolving at
            # the head of the next _step_async costs no latency and lets
            # the prefill's GPU tail overlap a whole decode step instead
            # of host-blocking here
            self._pf_pending = (logits, f_reqs)
            self._phase_mark("pf_sample_ms", t0)
            return
        toks = self._sample(logits, f_reqs)  # host sync point
        self._phase_mark("pf_sample_ms", t0)
        ema = self._phase_ms.get("pf_cpu_ms", 0.0)
        self._phase_ms["pf_cpu_ms"] = (0.05 * (time.thread_time() - c0) * 1000.0
                                       + 0.95 * ema)
        now = time.time()
        with self._lock:
            for r, t in zip(f_reqs, toks):
                r.generated.append(int(t))
                r.first_token_t = now
                self._finish_or_run(r, int(t))

    def _resolve_prefill(self) -> None:
        """Sample + publish a deferred prefill (see _prefill). Runs at the
        head of the next step and at every drain point, so no checkpoint
        or unbind ever observes an unsampled prefill."""
        pend = self._pf_pending
        if pend is None:
            return
        self._pf_pending = None
        logits, f_reqs = pend
        if self._prefill_stream is not None:
            # order the sampling kernels after the side-stream producer
            torch.cuda.current_stream().wait_stream(self._prefill_stream)
        toks = self._sample(logits, f_reqs)
        now = time.time()
        with self._lock:
            for r, t in zip(f_reqs, toks):
                if r.done.is_set():
                    continue  # failed/detached while deferred
                r.generated.append(int(t))
                r.first_token_t = now
                self._finish_or_run(r, int(t))

    @staticmethod
    def _prefill_bucket(n: int) -> int:
        """Round a prefill token count up to a small set of GEMM shapes.

        hipBLASLt runs its (CPU-expensive) algorithm heuristic per NEW
        problem shape; real-text prompts make nearly every prefill batch
        a never-seen M, which measured as ~100 ms of pure CPU per prefill
        under HTTP serving (tools/server_load.py engine_mode.pf_cpu_ms).
        Bucketing M to powers of two keeps the library cache warm after
        one occurrence of each bucket."""
        b = 32
        while b < n:
            b <<= 1
        return b

    def _prefill_exec(self, plan) -> torch.Tensor:
        """Runs identically on every TP rank (plan = seq ids + tokens)."""
        dev = self.device
        ids: List[int] = []
        positions: List[int] = []
        slots: List[int] = []
        q_starts, q_lens, seq_ids = [], [], []
        for seq_id, tokens, needs_reset, adopt in plan:
            if not self.kvm.has_seq(seq_id):
                self.kvm.create_seq(seq_id)  # lazily-created shared-prefix seq
            if needs_reset and self.tp_rank != 0:
                # rank 0 already reset at admission (before reserving KV
                # room); workers mirror the truncation via the plan flag
                self.kvm.reset_seq(seq_id)
            if adopt is not None and self.kvm.seq_len(seq_id) == 0:
                self.kvm.adopt_prefix(seq_id, adopt[0], adopt[1])
            prev = self.kvm.seq_len(seq_id)
            n = len(tokens)
            q_starts.append(len(ids))
            q_lens.append(n)
            seq_ids.append(seq_id)
            ids.extend(tokens)
            positions.extend(range(prev, prev + n))
            slots.extend(self.kvm.append_slots(seq_id, n))
        if self.dev_decode:
            for sid in seq_ids:
                self.kvm.push_dev(sid)
        self.prefill_tokens += len(ids)
        page_table = self.kvm.page_table(seq_ids, device="cpu")
        seq_lens = self.kvm.seq_lens(seq_ids, device="cpu")
        n_real = len(seq_ids)
        # pad the token dimension to a bucketed GEMM shape: extra tokens
        # form one fake row whose K/V land in the reserved scratch page 0
        # and whose logits are never read (last_rows covers real rows only)
        n_pad = (self._prefill_bucket(len(ids)) - len(ids)
                 if self.is_gpu else 0)
        if n_pad > 0:
            PS = self.kvm.page_size
            q_starts.append(len(ids))
            q_lens.append(n_pad)
            ids.extend([0] * n_pad)
            positions.extend(i % PS for i in range(n_pad))
            slots.extend(i % PS for i in range(n_pad))  # page 0 scratch
            # the attention kernel requires seq_len >= qlen (causal window
            # math); the pad row's "pages" are ceil(n_pad/PS) references to
            # scratch page 0, so its K/V reads stay in-bounds
            pad_pages = -(-n_pad // PS)
            width = max(page_table.size(1), pad_pages)
            if width > page_table.size(1):
                page_table = torch.cat(
                    [page_table,
                     torch.zeros(page_table.size(0),
                                 width - page_table.size(1),
                                 dtype=torch.int32)], dim=1)
            pad_row = torch.zeros(1, width, dtype=torch.int32)
            page_table = torch.cat([page_table, pad_row], dim=0)
            seq_lens = torch.cat([seq_lens, torch.tensor(
                [n_pad], dtype=torch.int32)])
        md = AttnMetadata(
            page_table=page_table.to(dev),
            seq_lens=seq_lens.to(dev),
            slot_mapping=torch.tensor(slots, dtype=torch.int64, device=dev),
            positions=torch.tensor(positions, dtype=torch.int32, device=dev),
            is_prefill=True,
            query_starts=torch.tensor(q_starts, dtype=torch.int32, device=dev),
            query_lens=torch.tensor(q_lens, dtype=torch.int32, device=dev),
            max_qlen=max(q_lens),
        )
        input_ids = torch.tensor(ids, dtype=torch.long, device=dev)
        last_rows = torch.tensor(
            [s + l - 1 for s, l in zip(q_starts[:n_real], q_lens[:n_real])],
            dtype=torch.long, device=dev)
        c0 = time.thread_time()
        out = self.model(input_ids, md, self.kvm.kv_caches(), last_rows)
        fwd_cpu = (time.thread_time() - c0) * 1000.0
        ema = self._phase_ms.get("pf_fwd_cpu_ms", 0.0)
        self._phase_ms["pf_fwd_cpu_ms"] = 0.05 * fwd_cpu + 0.95 * ema
        mn = self._phase_ms.get("pf_fwd_cpu_min_ms")
        if mn is None or fwd_cpu < mn:
            self._phase_ms["pf_fwd_cpu_min_ms"] = fwd_cpu
        return out

    # ---------- device-side decode (hipGraph) ----------

    def _device_decode_fwd(self, rows: torch.Tensor, input_ids: torch.Tensor,
                           inc: torch.Tensor) -> torch.Tensor:
        """The captured body: everything reads device state only."""
        kvm = self.kvm
        PS = kvm.page_size
        lens = kvm.dev_seq_lens.index_select(0, rows)      # int32 [Bk]
        pos = lens.clamp(min=0)
        page_idx = torch.div(pos, PS, rounding_mode="floor").long()
        pt = kvm.dev_page_table.index_select(0, rows)      # [Bk, MP] int32
        pages = pt.gather(1, page_idx.unsqueeze(1)).squeeze(1).long()
        slots = pages * PS + (pos % PS).long()             # pad rows -> page 0
        md = AttnMetadata(page_table=pt, seq_lens=lens + 1,
                          slot_mapping=slots, positions=pos, is_prefill=False)
        logits = self.model(input_ids, md, self.kvm.kv_caches(), None)
        kvm.dev_seq_lens.index_add_(0, rows, inc)
        kvm.dev_seq_lens.index_fill_(0, self._pad_slot_t, -1)
        return logits

    @staticmethod
    def _bucket(n: int) -> int:
        b = 1
        while b < n:
            b <<= 1
        return b

    def _get_graph(self, bucket: int) -> Dict[str, Any]:
        entry = self._graphs.get(bucket)
        if entry is not None:
            return entry
        dev = self.device
        rows = torch.full((bucket,), self._pad_slot, dtype=torch.long, device=dev)
        ids = torch.zeros(bucket, dtype=torch.long, device=dev)
        inc = torch.ones(bucket, dtype=torch.int32, device=dev)
        rows_pin = torch.full((bucket,), self._pad_slot, dtype=torch.long,
                              pin_memory=self.is_gpu)
        ids_pin = torch.zeros(bucket, dtype=torch.long, pin_memory=self.is_gpu)
        entry = {"rows": rows, "ids": ids, "inc": inc, "rows_pin": rows_pin,
                 "ids_pin": ids_pin, "graph": None, "logits": None}
        if self.use_graph:
            try:
                for _ in range(2):  # warmup before capture (also exercises
                    self._device_decode_fwd(rows, ids, inc)  # RCCL comms)
                torch.cuda.synchronize()
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    entry["logits"] = self._device_decode_fwd(rows, ids, inc)
                entry["graph"] = g
            except Exception:
                # e.g. an RCCL build that rejects capture: run eager. Every
                # rank executes identical code, so the fallback is lockstep.
                traceback.print_exc()
                self.use_graph = False
                self.async_decode = False
                entry["graph"] = None
                entry["logits"] = None
                torch.cuda.synchronize()
        self._graphs[bucket] = entry
        return entry

    def _decode(self, reqs: List[GenRequest]):
        B0 = len(reqs)
        if self.dev_decode:
            bucket = min(self._bucket(B0), max(self.max_decode_batch, 1))
            if bucket < B0:
                reqs = reqs[:bucket]
        plan = [(self._bindings[r.agent_id].seq_id, r.generated[-1])
                for r in reqs]
        if self.tp_size > 1 and self.tp_rank == 0:
            if self._pending_rb:  # async->eager fallback edge
                rb, self._pending_rb = self._pending_rb, []
                par.send_ints([OP_ROLLBACK, self._model_idx, len(rb)] + rb)
            # binary hot path: ONE int-tensor broadcast on the gloo plan
            # channel (slot ids are deterministic across ranks), replacing
            # the round-1 per-step pickled broadcast_object_list
            vec = [OP_DECODE, self._model_idx, len(plan)]
            vec.extend(self.kvm.slot(s) for s, _t in plan)
            vec.extend(t for _s, t in plan)
            par.send_ints(vec)
        logits = self._decode_exec(plan)
        toks = self._sample(logits, reqs)
        self.decode_tokens += len(reqs)
        self.occupancy_acc += len(reqs) / max(1, self.max_decode_batch)
        with self._lock:
            for r, t in zip(reqs, toks):
                r.generated.append(int(t))
                self._finish_or_run(r, int(t))

    def _decode_exec(self, plan) -> torch.Tensor:
        """One batched decode step; runs identically on every TP rank."""
        if self.dev_decode:
            return self._decode_exec_gpu(plan)
        dev = self.device
        ids, positions, slots, seq_ids = [], [], [], []
        for seq_id, tok in plan:
            prev = self.kvm.seq_len(seq_id)
            ids.append(tok)
            positions.append(prev)
            slots.extend(self.kvm.append_slots(seq_id, 1))
            seq_ids.append(seq_id)
        md = AttnMetadata(
            page_table=self.kvm.page_table(seq_ids, device=dev),
            seq_lens=self.kvm.seq_lens(seq_ids, device=dev),
            slot_mapping=torch.tensor(slots, dtype=torch.int64, device=dev),
            positions=torch.tensor(positions, dtype=torch.int32, device=dev),
            is_prefill=False,
        )
        input_ids = torch.tensor(ids, dtype=torch.long, device=dev)
        return self.model(input_ids, md, self.kvm.kv_caches(), None)

    def _decode_exec_gpu(self, plan) -> torch.Tensor:
        kvm = self.kvm
        B = len(plan)
        bucket = min(self._bucket(B), max(self.max_decode_batch, 1))
        # host half of the append, one lock for the whole batch
        row_ids = kvm.decode_batch_prepare([s for s, _t in plan])
        entry = self._get_graph(bucket)
        # TP workers run ahead with no host sync per step: wait out the
        # previous step's H2D DMA before rewriting the pinned staging
        ev = entry.get("h2d_ev") if self.is_gpu else None
        if ev is not None:
            ev.synchronize()
        entry["rows_pin"][:B] = torch.tensor(row_ids, dtype=torch.long)
        entry["rows_pin"][B:] = self._pad_slot
        entry["ids_pin"][:B] = torch.tensor([t for _s, t in plan],
                                            dtype=torch.long)
        entry["rows"].copy_(entry["rows_pin"], non_blocking=True)
        entry["ids"].copy_(entry["ids_pin"], non_blocking=True)
        if self.is_gpu:
            if ev is None:
                ev = entry["h2d_ev"] = torch.cuda.Event()
            ev.record()
        if entry["graph"] is not None:
            entry["graph"].replay()
            logits = entry["logits"][:B]
        else:
            logits = self._device_decode_fwd(entry["rows"], entry["ids"],
                                             entry["inc"])[:B]
        kvm.advance_many([s for s, _t in plan])
        return logits

    def _decode_async_worker(self, cmd: List[int]) -> None:
        """Worker half of the speculative decode step (OP_DECODE_ASYNC).

        Mirrors rank 0's phase-1 launch exactly: apply the prepended
        rollbacks, prepare pages, feed carried-over rows from this rank's
        OWN previous device sample (identical logits => identical sample),
        replay the graph, sample on-device with the plan's per-row
        (temp, top_p, effective-seed), advance. No host sync, no return
        traffic — the worker runs ahead of its GPU."""
        kvm = self.kvm
        B = cmd[2]
        nrb = cmd[3]
        off = 4
        rb = cmd[off:off + nrb]
        off += nrb
        if rb:
            kvm.rollback_many([kvm.seq_of_slot(s) for s in rb])
        rows_l: List[int] = []
        toks_l: List[int] = []
        idx_l: List[int] = []
        seeds_l: List[int] = []
        temps_l: List[float] = []
        tops_l: List[float] = []
        for i in range(B):
            slot, tok, pidx, seed, tbits, pbits = cmd[off + 6 * i:off + 6 * i + 6]
            rows_l.append(slot)
            toks_l.append(tok if tok >= 0 else 0)
            idx_l.append(pidx)
            seeds_l.append(seed)
            temps_l.append(_i2f(tbits))
            tops_l.append(_i2f(pbits))
        seq_ids = [kvm.seq_of_slot(s) for s in rows_l]
        kvm.decode_batch_prepare(seq_ids)  # same pages, deterministic
        bucket = min(self._bucket(B), max(self.max_decode_batch, 1))
        entry = self._get_graph(bucket)
        ev = entry.get("h2d_ev") if self.is_gpu else None
        if ev is not None:
            ev.synchronize()
        entry["rows_pin"][:B] = torch.tensor(rows_l, dtype=torch.long)
        entry["rows_pin"][B:] = self._pad_slot
        entry["rows"].copy_(entry["rows_pin"], non_blocking=True)
        entry["ids_pin"][:B] = torch.tensor(toks_l, dtype=torch.long)
        entry["ids"][:B].copy_(entry["ids_pin"][:B], non_blocking=True)
        if self.is_gpu:
            if ev is None:
                ev = entry["h2d_ev"] = torch.cuda.Event()
            ev.record()
        prev = self._spec_worker
        if prev is not None and any(x >= 0 for x in idx_l):
            dev = self.device
            gidx = torch.tensor([max(x, 0) for x in idx_l], dtype=torch.long,
                                device=dev)
            mask = torch.tensor([x >= 0 for x in idx_l], dtype=torch.bool,
                                device=dev)
            gathered = prev["sampled"].index_select(0, gidx)
            entry["ids"][:B].copy_(torch.where(mask, gathered,
                                               entry["ids"][:B]))
        if entry["graph"] is not None:
            entry["graph"].replay()
            logits = entry["logits"][:B]
        else:
            logits = self._device_decode_fwd(entry["rows"], entry["ids"],
                                             entry["inc"])[:B]
        sampled = self._sample_device_params(logits, temps_l, tops_l, seeds_l)
        kvm.advance_many(seq_ids)
        self.decode_tokens += B
        self._spec_worker = {"sampled": sampled}

    def _finish_or_run(self, r: GenRequest, tok: int):
        """Called with lock held, after appending tok."""
        finished = (tok == self.tokenizer.eos_id or
                    len(r.generated) >= r.max_new)
        if r.stream_q is not None:
            r.stream_q.put(tok)
            if finished:
                r.stream_q.put(None)
        if r in self._running_set:
            if finished:
                self.running.remove(r)
                self._running_set.discard(r)
        elif not finished:
            self.running.append(r)
            self._running_set.add(r)
        if finished:
            b = self._bindings.get(r.agent_id)
            if b is not None:
                b.requests += 1
                b.tokens += len(r.generated)
                b.active = None
            r.fin_t = time.time()
            r.done.set()
            self._pump_agent(b)

    def _pump_agent(self, b: Optional[AgentBinding]):
        """Admit the agent's next queued request (per-agent FIFO — one
        in-flight request per conversation sequence)."""
        if b is None or b.paused or b.active is not None:
            return
        try:
            nxt = b.queue.get_nowait()
        except queue.Empty:
            return
        b.active = nxt
        self.submit(nxt)

    # ---------- binding management ----------

    def _bind_seq(self, seq_id: str, ckpt: Optional[KVCheckpoint]):
        if ckpt is not None:
            self.kvm.restore(seq_id, ckpt)
        elif not self.kvm.has_seq(seq_id):
            self.kvm.create_seq(seq_id)

    def _acquire_step_mutex_unless_stuck(self) -> bool:
        """Blocking acquire that gives up when the engine thread is wedged
        mid-step (watchdog fired): a stuck HIP stream holds _step_mutex
        forever, and the health monitor's restart path must not deadlock
        behind it (it IS the recovery path for exactly that state)."""
        while True:
            if self._step_mutex.acquire(timeout=2.0):
                return True
            if self.stuck():
                return False

    def bind(self, agent, seq_id: str, ckpt: Optional[KVCheckpoint],
             worker_has_ckpt: bool = False):
        locked = False
        if not self.sync_mode:
            locked = self._acquire_step_mutex_unless_stuck()
            if not locked:
                raise EngineDead(f"engine {self.name} is stuck (watchdog)")
        try:
            self._bind_locked(agent, seq_id, ckpt, worker_has_ckpt)
        finally:
            if locked:
                self._step_mutex.release()

    def _bind_locked(self, agent, seq_id: str, ckpt: Optional[KVCheckpoint],
                     worker_has_ckpt: bool = False):
        with self._lock:
            self._bcast(("bind", self.name, seq_id,
                         ckpt is not None or worker_has_ckpt))
            self._bind_seq(seq_id, ckpt)
            b = AgentBinding(agent=agent, model_name=self.name, seq_id=seq_id)
            sp = getattr(agent, "system_prompt", "") or ""
            if self.prefix_sharing and sp:
                # the whole-page head of this agent's first-turn prompt
                # (matches _build_prompt's "[system] ...\n" framing)
                pk = self.tokenizer.encode(f"[system] {sp}\n")
                n_full = (len(pk) // self.kvm.page_size) * self.kvm.page_size
                if n_full >= self.kvm.page_size:
                    b.prefix_tokens = pk[:n_full]
            self._bindings[agent.id] = b
            self.refcount += 1

    def unbind(self, agent_id: str, offload: bool) -> Optional[KVCheckpoint]:
        locked = True
        if not self.sync_mode:
            locked = self._acquire_step_mutex_unless_stuck()
        try:
            # stuck engine: force-detach — skip the drain/offload (any GPU
            # work would hang on the wedged stream) so the health monitor's
            # stop->start restart completes; the agent resumes with a fresh
            # KV sequence and its pending WAL replays (at-least-once)
            return self._unbind_locked(agent_id, offload and locked,
                                       force=not locked)
        finally:
            if locked and not self.sync_mode:
                self._step_mutex.release()

    def _unbind_locked(self, agent_id: str, offload: bool,
                       force: bool = False) -> Optional[KVCheckpoint]:
        if self.async_decode and not force:
            self.drain_async()
        with self._lock:
            b = self._bindings.pop(agent_id, None)
            if b is None:
                return None
            self.refcount -= 1
            # drain: fail queued requests (they stay pending in the WAL and
            # will be replayed after resume — at-least-once contract)
            if b.active is not None and not b.active.done.is_set():
                b.active.error = "agent detached"
                b.active.done.set()
                if b.active in self._running_set:
                    self.running.remove(b.active)
                    self._running_set.discard(b.active)
                if b.active in self._chunking:
                    self._chunking.remove(b.active)  # mid-chunked-prefill
            while True:
                try:
                    r = b.queue.get_nowait()
                except queue.Empty:
                    break
                r.error = "agent detached"
                r.done.set()
        self._bcast(("unbind", self.name, b.seq_id, offload))
        ckpt = self.kvm.offload(b.seq_id, free=True) if offload else None
        if not offload:
            self.kvm.free_seq(b.seq_id)
        return ckpt

    def binding(self, agent_id: str) -> Optional[AgentBinding]:
        with self._lock:
            return self._bindings.get(agent_id)


class LLMEngine:
    """EngineBackend over ModelInstances. Also serves the echo model so the
    whole control plane runs against one backend."""

    def __init__(self, store: Store, config=None, device: str = "cpu",
                 state_root: Optional[str] = None):
        self.store = store
        self.device = device
        self.engine_cfg = dict(config.get("engine")) if config else {}
        self.state_root = state_root or os.path.expanduser("~/.agentainer_amd")
        self._instances: Dict[str, ModelInstance] = {}
        # creation order == binary-plan model index (lockstep on all ranks)
        self._instance_order: List[str] = []
        self._agent_model: Dict[str, str] = {}
        self._ckpts: Dict[str, KVCheckpoint] = {}
        self._lock = threading.RLock()
        self.tp_size = int(self.engine_cfg.get("tp_degree", 1))
        self.tp_rank = par.tp_rank() if self.tp_size > 1 else 0
        if device == "cuda" and torch.cuda.is_available():
            if not ops.hip_available():
                raise RuntimeError(
                    "GPU engine requires the gfx950 HIP extension "
                    "(python -m agentainer_amd.ops.build)")

    # ---------- model registry ----------

    def validate_model(self, model: str) -> None:
        from ..models.mixtral import MIXTRAL_CONFIGS
        if model in LLAMA_CONFIGS or model in MIXTRAL_CONFIGS:
            return
        from .echo import ECHO_MODELS
        if model in ECHO_MODELS:
            return
        if os.path.isdir(os.path.expanduser(model)):
            return  # weights directory
        from ..models.mixtral import MIXTRAL_CONFIGS as _MC
        raise ModelNotFound(
            f"unknown model {model!r}; known: "
            f"{sorted(LLAMA_CONFIGS) + sorted(_MC)} + echo")

    def _make_instance(self, model: str) -> ModelInstance:
        from ..models.mixtral import MIXTRAL_CONFIGS, make_mixtral_instance
        if model in LLAMA_CONFIGS:
            return ModelInstance(model, LLAMA_CONFIGS[model], self.device,
                                 self.engine_cfg)
        if model in MIXTRAL_CONFIGS:
            return make_mixtral_instance(model, self.device, self.engine_cfg)
        path = os.path.expanduser(model)
        if os.path.isdir(path) and os.path.exists(
                os.path.join(path, "config.json")):
            # weights-path deploy: HF checkpoint directory (config.json +
            # *.safetensors [+ tokenizer.json]) — the "image" analog
            from ..models.llama import config_from_hf
            return ModelInstance(model, config_from_hf(path), self.device,
                                 self.engine_cfg, weights_path=path)
        raise ModelNotFound(f"cannot load {model!r}")

    def _get_instance(self, model: str) -> ModelInstance:
        with self._lock:
            inst = self._instances.get(model)
            if inst is None:
                if self.tp_size > 1 and self.tp_rank == 0:
                    par.broadcast_obj(("instance", model))
                inst = self._make_instance(model)
                inst._model_idx = len(self._instance_order)
                self._instance_order.append(model)
                self._instances[model] = inst
                inst.start()
            return inst

    # ---------- EngineBackend ----------

    def attach(self, agent) -> None:
        model = agent.model
        from .echo import ECHO_MODELS
        if model in ECHO_MODELS:
            # echo agents need no weights; emulate with tiny state
            with self._lock:
                self._agent_model[agent.id] = "echo"
                self._ckpts.pop(agent.id, None)
            self.store.hset(f"agent:{agent.id}:metrics", "engine", "echo")
            self._echo_attached = getattr(self, "_echo_attached", {})
            self._echo_attached[agent.id] = {"paused": False, "requests": 0,
                                             "tokens": 0}
            return
        inst = self._get_instance(model)
        ckpt = self._ckpts.pop(agent.id, None)
        if ckpt is None:
            ckpt = self._load_disk_ckpt(agent.id)
        inst.bind(agent, seq_id=agent.id, ckpt=ckpt)  # workers restore their shard
        with self._lock:
            self._agent_model[agent.id] = model

    def detach(self, agent_id: str, offload_kv: bool = True) -> bool:
        with self._lock:
            model = self._agent_model.pop(agent_id, None)
        if model is None:
            return False
        if model == "echo":
            getattr(self, "_echo_attached", {}).pop(agent_id, None)
            return bool(offload_kv)
        inst = self._instances.get(model)
        if inst is None:
            return False
        ckpt = inst.unbind(agent_id, offload=offload_kv)
        if ckpt is not None:
            self._ckpts[agent_id] = ckpt
            self._save_disk_ckpt(agent_id, ckpt)
            return True
        return False

    def export_kv(self, agent_id: str) -> Optional[KVCheckpoint]:
        """Snapshot an agent's conversation KV WITHOUT detaching it (backup
        path): live agents offload copy-on-read under the step mutex;
        stopped agents return their cached/disk checkpoint."""
        model = self._agent_model.get(agent_id)
        if model in (None, "echo"):
            return None
        inst = self._instances.get(model)
        if inst is None:
            return None
        b = inst.binding(agent_id)
        if b is None:
            return self._ckpts.get(agent_id) or self._load_disk_ckpt(agent_id)
        if inst.tp_size > 1:
            # a live TP snapshot would capture only rank 0's shard (the
            # workers' shards live in their own processes); stop/resume
            # checkpoints per-rank correctly — skip live KV here
            return None
        locked = True
        if not inst.sync_mode:
            locked = inst._acquire_step_mutex_unless_stuck()
            if not locked:
                return None  # wedged engine: no live snapshot possible
        try:
            if inst.async_decode:
                inst.drain_async()
            return inst.kvm.offload(b.seq_id, free=False)
        finally:
            if locked and not inst.sync_mode:
                inst._step_mutex.release()

    def import_kv(self, agent_id: str, ckpt: KVCheckpoint) -> None:
        """Stage a checkpoint so the agent's NEXT start restores it (backup
        restore path — restored agents begin stopped)."""
        self._ckpts[agent_id] = ckpt
        self._save_disk_ckpt(agent_id, ckpt)

    def purge_agent(self, agent_id: str) -> None:
        """Drop every engine-side trace of a REMOVED agent: in-memory and
        on-disk KV checkpoints (detach keeps them on purpose — a crashed
        or stopped agent must be resumable; a removed one must not leak
        checkpoint files into state_root/kv_ckpt)."""
        self._ckpts.pop(agent_id, None)
        p = self._ckpt_path(agent_id)
        if os.path.exists(p):
            try:
                os.remove(p)
            except OSError:
                pass

    def pause(self, agent_id: str) -> None:
        b = self._binding(agent_id)
        if b is not None:
            b.paused = True
        eb = getattr(self, "_echo_attached", {}).get(agent_id)
        if eb is not None:
            eb["paused"] = True

    def unpause(self, agent_id: str) -> None:
        b = self._binding(agent_id)
        if b is not None:
            b.paused = False
            inst = self._instances.get(self._agent_model.get(agent_id, ""))
            if inst:
                with inst._lock:
                    inst._pump_agent(b)
        eb = getattr(self, "_echo_attached", {}).get(agent_id)
        if eb is not None:
            eb["paused"] = False

    def reset_conversation(self, agent_id: str) -> None:
        """/clear hook: schedule a KV reset so the next chat starts from an
        empty sequence (clearing only the history list would leave the next
        turn's system prompt prefilled on top of the stale multi-turn KV —
        the agent would keep attending to the cleared conversation)."""
        b = self._binding(agent_id)
        if b is not None:
            b.pending_reset = True

    def _binding(self, agent_id: str) -> Optional[AgentBinding]:
        model = self._agent_model.get(agent_id)
        if model in (None, "echo"):
            return None
        inst = self._instances.get(model)
        return inst.binding(agent_id) if inst else None

    def is_attached(self, agent_id: str) -> bool:
        return agent_id in self._agent_model

    def attached_ids(self) -> List[str]:
        with self._lock:
            return list(self._agent_model)

    def engine_status(self, agent_id: str) -> str:
        model = self._agent_model.get(agent_id)
        if model is None:
            return "missing"
        if model == "echo":
            eb = getattr(self, "_echo_attached", {}).get(agent_id)
            return "paused" if (eb and eb["paused"]) else "running"
        b = self._binding(agent_id)
        if b is None:
            return "missing"
        return "paused" if b.paused else "running"

    def health_probe(self, agent_id: str) -> bool:
        model = self._agent_model.get(agent_id)
        if model is None:
            return False
        if model == "echo":
            return True
        inst = self._instances.get(model)
        return inst is not None and inst.alive()

    # ---------- chat ----------

    def _submit_chat(self, agent_id: str, message: str, kwargs: Dict[str, Any],
                     stream: bool = False):
        """Shared head of chat/chat_stream: validate, build + enqueue the
        GenRequest. Returns (inst, binding, req, timeout_s)."""
        from ..wal import EngineUnavailable

        model = self._agent_model.get(agent_id)
        inst = self._instances.get(model)
        b = inst.binding(agent_id) if inst else None
        if inst is None or b is None or not inst.alive():
            raise EngineUnavailable(f"engine for {agent_id} is down")
        if b.paused:
            raise EngineUnavailable(f"agent {agent_id} is paused")
        agent = b.agent
        sampling = dict(agent.sampling or {})
        sampling.update(kwargs)
        trace_id = str(sampling.pop("trace_id", ""))
        prompt = self._build_prompt(agent, message)
        req = GenRequest(
            agent_id=agent_id,
            prompt_tokens=inst.tokenizer.encode(prompt),
            max_new=int(sampling.get("max_tokens", DEFAULT_MAX_NEW)),
            temperature=float(sampling.get("temperature", 0.0)),
            top_p=float(sampling.get("top_p", 1.0)),
            seed=int(sampling.get("seed", 0)),
            trace_id=trace_id,
        )
        if stream:
            req.stream_q = queue.Queue()
        with inst._lock:
            # the binding may have been unbound between our lookup and
            # taking the lock — enqueueing then would orphan the request
            # in a dead queue until the caller's timeout (unbind drains
            # under this same lock, so this check closes the race)
            if inst._bindings.get(agent_id) is not b:
                raise EngineUnavailable(f"agent {agent_id} detached")
            b.queue.put(req)
            inst._pump_agent(b)
        return inst, b, req, float(sampling.get("timeout_s", 120.0))

    def _chat_payload(self, agent_id: str, model: str, inst, b,
                      req: GenRequest, message: str) -> Dict[str, Any]:
        """Shared tail: history + metrics writes, response envelope."""
        text = inst.tokenizer.decode(req.generated)
        hist_key = f"agent:{agent_id}:conversations"
        self.store.rpush(hist_key, {"user": message, "assistant": text,
                                    "ts": time.time()})
        n = self.store.llen(hist_key)
        if n > 50:
            self.store.ltrim(hist_key, n - 50, -1)
        self.store.hset(f"agent:{agent_id}:metrics", "total_requests", b.requests)
        return {
            "response": text,
            "model": model,
            "trace_id": req.trace_id or None,
            "tokens": len(req.generated),
            "ttft_s": (req.first_token_t - req.enq_t) if req.first_token_t else None,
            "e2e_s": (req.fin_t - req.enq_t) if req.fin_t else None,
        }

    def chat(self, agent_id: str, message: str, **kwargs: Any) -> Dict[str, Any]:
        from ..wal import EngineUnavailable

        model = self._agent_model.get(agent_id)
        if model is None:
            raise EngineUnavailable(f"agent {agent_id} is not attached")
        if model == "echo":
            return self._echo_chat(agent_id, message)
        inst, b, req, timeout = self._submit_chat(agent_id, message, kwargs)
        if not req.done.wait(timeout):
            raise EngineUnavailable(f"generation timed out after {timeout}s")
        if req.error:
            raise EngineUnavailable(req.error)
        return self._chat_payload(agent_id, model, inst, b, req, message)

    def chat_stream(self, agent_id: str, message: str, **kwargs: Any):
        """Streaming chat: yields {"token", "text"} events as the scheduler
        produces tokens, then one final {"done": True, **payload} event —
        the SSE data frames of the /chat?stream path. Token->text deltas
        re-decode the full id list each step so multi-byte UTF-8 sequences
        only surface once complete."""
        from ..wal import EngineUnavailable

        model = self._agent_model.get(agent_id)
        if model is None:
            raise EngineUnavailable(f"agent {agent_id} is not attached")
        if model == "echo":
            payload = self._echo_chat(agent_id, message)
            for word in payload["response"].split(" "):
                yield {"token": None, "text": word + " "}
            yield {"done": True, **payload}
            return
        inst, b, req, timeout = self._submit_chat(agent_id, message, kwargs,
                                                  stream=True)
        deadline = time.time() + timeout
        dec = inst.tokenizer.stream_decoder()
        while True:
            try:
                tok = req.stream_q.get(timeout=0.1)
            except queue.Empty:
                # fallback for completions that bypass _finish_or_run
                # (detach/unbind drain paths set done+error directly)
                if req.done.is_set() and req.stream_q.empty():
                    break
                if time.time() > deadline:
                    raise EngineUnavailable(
                        f"generation timed out after {timeout}s")
                continue
            if tok is None:
                break
            yield {"token": int(tok), "text": dec.feed(int(tok))}
        if req.error:
            raise EngineUnavailable(req.error)
        tail = dec.flush()  # a dangling partial byte sequence
        if tail:
            yield {"token": None, "text": tail}
        yield {"done": True,
               **self._chat_payload(agent_id, model, inst, b, req, message)}

    def _build_prompt(self, agent, message: str) -> str:
        """This turn's incremental prompt; earlier turns are already in the
        agent's KV sequence. Redis-analog history provides the replayable
        text context (examples/gpt-agent/app.py:89-92 last-3 pattern)."""
        parts = []
        if agent.system_prompt and self.store.llen(
                f"agent:{agent.id}:conversations") == 0:
            parts.append(f"[system] {agent.system_prompt}\n")
        parts.append(f"[user] {message}\n[assistant] ")
        return "".join(parts)

    def _echo_chat(self, agent_id: str, message: str) -> Dict[str, Any]:
        eb = getattr(self, "_echo_attached", {}).get(agent_id)
        from ..wal import EngineUnavailable
        if eb is None:
            raise EngineUnavailable(f"agent {agent_id} not attached")
        if eb["paused"]:
            raise EngineUnavailable(f"agent {agent_id} is paused")
        hist_key = f"agent:{agent_id}:conversations"
        ctx = self.store.lrange(hist_key, -3, -1)
        reply = f"echo({len(ctx)}): {message}"
        self.store.rpush(hist_key, {"user": message, "assistant": reply,
                                    "ts": time.time()})
        eb["requests"] += 1
        eb["tokens"] += len(reply.split())
        return {"response": reply, "model": "echo", "tokens": eb["tokens"]}

    # ---------- checkpoint persistence (server-restart recovery) ----------

    def _ckpt_path(self, agent_id: str) -> str:
        d = os.path.join(self.state_root, "kv_ckpt")
        os.makedirs(d, exist_ok=True)
        suffix = f".rank{self.tp_rank}" if self.tp_size > 1 else ""
        return os.path.join(d, f"{agent_id}{suffix}.pt")

    def _save_disk_ckpt(self, agent_id: str, ckpt: KVCheckpoint) -> None:
        torch.save({"length": ckpt.length, "n_pages": ckpt.n_pages,
                    "data": ckpt.data.cpu()}, self._ckpt_path(agent_id))

    def _load_disk_ckpt(self, agent_id: str) -> Optional[KVCheckpoint]:
        p = self._ckpt_path(agent_id)
        if not os.path.exists(p):
            return None
        try:
            d = torch.load(p, map_location="cpu", weights_only=True)
            return KVCheckpoint(length=d["length"], n_pages=d["n_pages"],
                                data=d["data"])
        except Exception:
            return None

    # ---------- stats ----------

    def stats(self) -> Dict[str, Any]:
        agents: Dict[str, Any] = {}
        for aid, st in getattr(self, "_echo_attached", {}).items():
            agents[aid] = {"requests": st["requests"], "tokens": st["tokens"],
                           "paused": st["paused"]}
        models = {}
        with self._lock:  # _get_instance can insert concurrently
            instances = list(self._instances.items())
        for name, inst in instances:
            with inst._lock:
                for aid, b in inst._bindings.items():
                    agents[aid] = {
                        "requests": b.requests, "tokens": b.tokens,
                        "paused": b.paused,
                        "kv_bytes": inst.kvm.seq_bytes(b.seq_id),
                        "kv_pages": len(inst.kvm._seqs.get(b.seq_id).pages)
                        if inst.kvm.has_seq(b.seq_id) else 0,
                        "batch_occupancy": (inst.occupancy_acc / inst.steps)
                        if inst.steps else 0.0,
                    }
            models[name] = {
                "steps": inst.steps,
                "decode_tokens": inst.decode_tokens,
                "prefill_tokens": inst.prefill_tokens,
                # execution-mode diagnostics: use_graph flips False on a
                # capture failure; buckets list which batch sizes captured
                "use_graph": inst.use_graph,
                "async_decode": inst.async_decode,
                "graph_buckets": sorted(inst._graphs.keys()),
                "step_ms_ema": round(getattr(inst, "_step_ms_ema", 0.0), 3),
                "phase_ms": {k: round(v, 3)
                             for k, v in inst._phase_ms.items()},
                "kv_pages_used": inst.kvm.used_pages,
                "kv_pages_free": inst.kvm.free_pages,
                "kv_pages_shared": len(inst.kvm._refs),
                "shared_prefixes": len(inst._prefixes),
                "kv_dtype": str(inst.kvm.dtype).replace("torch.", ""),
                "running": len(inst.running),
                "stuck": inst.stuck(),
                "last_step_age_s": round(time.time() - inst.last_step_t, 3),
            }
        hbm = {}
        if self.device.startswith("cuda") and torch.cuda.is_available():
            free, total = torch.cuda.mem_get_info()
            hbm = {"hbm_total_bytes": total, "hbm_free_bytes": free,
                   "hbm_torch_allocated": torch.cuda.memory_allocated()}
        out = {"engine": "llm", "device": self.device, "agents": agents,
               "models": models}
        out.update(hbm)
        return out

    def shutdown(self):
        if self.tp_size > 1 and self.tp_rank == 0 and par.is_tp():
            par.broadcast_obj(("shutdown",))
        for inst in self._instances.values():
            inst.stop()

    # ---------- TP worker loop (ranks > 0) ----------

    def run_worker(self):
        """SPMD worker: executes the step plans rank 0 broadcasts.
        Page/slot allocation is deterministic, so this rank's KV pool
        mirrors rank 0's without any per-step tensor metadata exchange.

        Decode (the per-token hot path) arrives as ONE int vector —
        [OP_DECODE, model_idx, B, slots, tokens] — and the worker enqueues
        the step with NO host synchronize: kernels and the in-graph RCCL
        all-reduces are stream-ordered, so the worker runs ahead of the
        GPU and plan latency overlaps compute (round 1 synced per
        command, stacking host latency on every token)."""
        assert self.tp_size > 1 and self.tp_rank > 0
        while True:
            cmd = par.recv_cmd()
            if isinstance(cmd, list):  # int-vector fast path
                op = cmd[0]
                if op == OP_DECODE:
                    inst = self._instances[self._instance_order[cmd[1]]]
                    B = cmd[2]
                    kvm = inst.kvm
                    plan = [(kvm.seq_of_slot(s), t)
                            for s, t in zip(cmd[3:3 + B], cmd[3 + B:3 + 2 * B])]
                    inst._decode_exec(plan)
                elif op == OP_DECODE_ASYNC:
                    inst = self._instances[self._instance_order[cmd[1]]]
                    inst._decode_async_worker(cmd)
                elif op == OP_ROLLBACK:
                    inst = self._instances[self._instance_order[cmd[1]]]
                    n = cmd[2]
                    inst.kvm.rollback_many(
                        [inst.kvm.seq_of_slot(s) for s in cmd[3:3 + n]])
                elif op == OP_BARRIER:
                    # timing fence (bench.py --tp): drain this rank's
                    # stream, then rendezvous so rank 0's clock bounds
                    # every rank
                    if self.device.startswith("cuda"):
                        torch.cuda.synchronize()
                    par.barrier()
                continue
            op = cmd[0]
            if op == "shutdown":
                break
            if op == "instance":
                with self._lock:
                    if cmd[1] not in self._instances:
                        inst = self._make_instance(cmd[1])
                        inst._model_idx = len(self._instance_order)
                        self._instance_order.append(cmd[1])
                        self._instances[cmd[1]] = inst
                continue
            inst = self._instances[cmd[1]]
            if op == "prefill":
                inst._prefill_exec(cmd[2])
            elif op == "decode":
                inst._decode_exec(cmd[2])
            elif op == "bind":
                seq_id, has_ckpt = cmd[2], cmd[3]
                ckpt = self._ckpts.pop(seq_id, None) if has_ckpt else None
                if has_ckpt and ckpt is None:
                    ckpt = self._load_disk_ckpt(seq_id)
                inst._bind_seq(seq_id, ckpt)
            elif op == "unbind":
                seq_id, offload = cmd[2], cmd[3]
                if offload:
                    ckpt = inst.kvm.offload(seq_id, free=True)
                    if ckpt is not None:
                        self._ckpts[seq_id] = ckpt
                        self._save_disk_ckpt(seq_id, ckpt)
                else:
                    inst.kvm.free_seq(seq_id)
