"""Asyncio wrapper around LLMEngine for the HTTP server, including the TP
driver/worker protocol.

TP>1 runs SPMD: every rank owns an identical scheduler+runner; rank 0
broadcasts {adds, aborts} before each step so all schedulers take identical
decisions, and the model's collectives (all-reduce / all-gather) keep ranks
in lockstep. Sampling noise comes from each rank's identically-seeded
generator, so sampled tokens agree without an extra broadcast.
"""

from __future__ import annotations

import asyncio
import concurrent.futures
import time
from dataclasses import dataclass, field

from ..config import EngineConfig
from ..engine import LLMEngine, SamplingParams, StepOutput
from ..parallel.comm import (
    get_tp_world_size,
    tp_broadcast_object,
    tp_broadcast_tensor,
)
from .metrics import EngineMetrics


@dataclass
class RequestAdd:
    request_id: str
    token_ids: list[int]
    sampling: dict  # SamplingParams fields (picklable for broadcast)
    hold_pages: bool = False  # disaggregated prefill: keep pages for extract


def apply_msg(engine: LLMEngine, msg: dict) -> list[str]:
    """Apply adds/aborts; returns request ids rejected at admission (e.g.
    prompt too long) so the driver can fail their streams instead of
    letting an exception kill the engine loop."""
    rejected: list[str] = []
    for add in msg.get("adds", ()):
        try:
            engine.add_request(
                add.token_ids, SamplingParams(**add.sampling), add.request_id,
                hold_pages=add.hold_pages,
            )
        except ValueError:
            rejected.append(add.request_id)
    for rid in msg.get("aborts", ()):
        engine.abort_request(rid)
    return rejected


def worker_loop(cfg: EngineConfig) -> None:
    """Ranks > 0: follow the driver's broadcasts forever."""
    engine = LLMEngine(cfg)
    while True:
        msg = tp_broadcast_object(None, src=0)
        if msg is None or msg.get("stop"):
            break
        if "extract" in msg:
            # Disaggregated prefill: every rank joins the head all-gather
            # inside extract_kv; only rank 0 ships the result over the wire.
            engine.extract_prefilled(msg["extract"])
            continue
        if "inject" in msg:
            m = msg["inject"]
            kv = tp_broadcast_tensor(None, src=0)
            engine.add_prefilled(
                m["token_ids"], m["first_token"], kv,
                SamplingParams(**m["sampling"]), m["request_id"],
            )
            continue
        apply_msg(engine, msg)
        engine.step()


@dataclass
class _Stream:
    queue: asyncio.Queue = field(default_factory=asyncio.Queue)
    prev_token_time: float | None = None
    arrival_time: float = field(default_factory=time.time)


class AsyncEngine:
    """Drives LLMEngine from an asyncio loop (rank 0)."""

    def __init__(self, cfg: EngineConfig, model_name: str = "model",
                 disagg_mode: str | None = None):
        self.cfg = cfg
        self.engine = LLMEngine(cfg)
        self.metrics = EngineMetrics(model_name)
        self.disagg_mode = disagg_mode  # None | "prefill" | "decode"
        self.http_transport = None  # httpx transport override (tests)
        self.streams: dict[str, _Stream] = {}
        self.pending_adds: list[RequestAdd] = []
        self.pending_aborts: list[str] = []
        self._wakeup: asyncio.Event | None = None
        self._task: asyncio.Task | None = None
        self.failed: BaseException | None = None  # fatal engine-loop error
        # Collectives + forward run on ONE dedicated thread.
        self._executor = concurrent.futures.ThreadPoolExecutor(max_workers=1)
        self._stopped = False

    @property
    def model_cfg(self):
        return self.engine.model_cfg

    async def start(self):
        self._wakeup = asyncio.Event()
        self._task = asyncio.get_running_loop().create_task(self._run())

    async def stop(self):
        self._stopped = True
        if self._wakeup:
            self._wakeup.set()
        if self._task:
            await self._task
        if get_tp_world_size() > 1:
            await asyncio.get_running_loop().run_in_executor(
                self._executor, tp_broadcast_object, {"stop": True}, 0
            )
        self._executor.shutdown(wait=False)

    # ---- request API ----
    def submit(self, request_id: str, token_ids: list[int],
               sampling: SamplingParams, hold_pages: bool = False) -> _Stream:
        st = _Stream()
        if self.failed is not None:  # engine loop is gone — end immediately
            st.queue.put_nowait(None)
            return st
        self.streams[request_id] = st
        self.pending_adds.append(
            RequestAdd(request_id, token_ids, sampling.__dict__.copy(), hold_pages)
        )
        self.metrics.prompt_tokens.inc(len(token_ids))
        if self._wakeup:
            self._wakeup.set()
        return st

    # ---- prefill/decode disaggregation (reference delegates this to
    # SGLang's --disaggregation-mode; here it is first-party over a TCP KV
    # page transfer — SURVEY.md §2.2) ----
    async def disagg_prefill(self, request_id: str, token_ids: list[int],
                             sampling: SamplingParams):
        """Prefill-instance side: run the prompt, sample the first token,
        return (first_token, finish_reason, kv pages tensor). TP>1: workers
        are told to join the extract (head all-gather) in lockstep."""
        import dataclasses

        sp = dataclasses.replace(sampling, max_tokens=1)
        st = self.submit(request_id, token_ids, sp, hold_pages=True)
        out = await st.queue.get()
        assert out is not None and out.finished
        await st.queue.get()  # sentinel
        loop = asyncio.get_running_loop()

        def _extract():
            if get_tp_world_size() > 1:
                tp_broadcast_object({"extract": request_id}, src=0)
            _, kv = self.engine.extract_prefilled(request_id)
            return kv

        kv = await loop.run_in_executor(self._executor, _extract)
        return out.new_token_id, out.finish_reason, kv

    async def disagg_inject(self, request_id: str, token_ids: list[int],
                            first_token: int, kv,
                            sampling: SamplingParams) -> _Stream:
        """Decode-instance side: admit a remotely prefilled sequence and
        return its output stream (first token already queued). TP>1: the
        full-head KV is broadcast; each rank injects its head slice."""
        st = _Stream()
        self.streams[request_id] = st
        loop = asyncio.get_running_loop()

        def _inject():
            if get_tp_world_size() > 1:
                tp_broadcast_object({"inject": {
                    "request_id": request_id,
                    "token_ids": token_ids,
                    "first_token": first_token,
                    "sampling": sampling.__dict__.copy(),
                }}, src=0)
                tp_broadcast_tensor(kv, src=0)
            return self.engine.add_prefilled(
                token_ids, first_token, kv, sampling, request_id
            )

        seq = await loop.run_in_executor(self._executor, _inject)
        self.metrics.prompt_tokens.inc(len(token_ids))
        self.metrics.generation_tokens.inc()
        now = time.time()
        self.metrics.ttft.observe(now - seq.arrival_time)
        st.prev_token_time = now
        st.queue.put_nowait(StepOutput(
            request_id=request_id, seq_id=seq.seq_id, new_token_id=first_token,
            finished=seq.is_finished, finish_reason=seq.finish_reason,
            num_prompt_tokens=seq.num_prompt_tokens, num_output_tokens=1,
        ))
        if seq.is_finished:
            self.metrics.record_finished(
                seq.finish_reason, seq.num_prompt_tokens, 1)
            st.queue.put_nowait(None)
            self.streams.pop(request_id, None)
        if self._wakeup:
            self._wakeup.set()
        return st

    def abort(self, request_id: str) -> None:
        self.pending_aborts.append(request_id)
        st = self.streams.pop(request_id, None)
        if st:
            st.queue.put_nowait(None)
        if self._wakeup:
            self._wakeup.set()

    # ---- engine loop ----
    def _sync_iteration(self, msg: dict):
        if get_tp_world_size() > 1:
            tp_broadcast_object(msg, src=0)
        rejected = apply_msg(self.engine, msg)
        return self.engine.step(), rejected

    async def _run(self):
        loop = asyncio.get_running_loop()
        while not self._stopped:
            if not (self.pending_adds or self.pending_aborts or self.engine.has_work()):
                self._wakeup.clear()
                await self._wakeup.wait()
                continue
            msg = {"adds": self.pending_adds, "aborts": self.pending_aborts}
            self.pending_adds, self.pending_aborts = [], []
            try:
                outputs, rejected = await loop.run_in_executor(
                    self._executor, self._sync_iteration, msg
                )
            except Exception as exc:  # fatal: fail open streams, go unready
                import traceback

                self.failed = exc
                traceback.print_exc()
                for st in self.streams.values():
                    st.queue.put_nowait(None)
                self.streams.clear()
                return
            for rid in rejected:
                st = self.streams.pop(rid, None)
                if st:
                    st.queue.put_nowait(None)  # stream ends with no tokens
            now = time.time()
            for out in outputs:
                st = self.streams.get(out.request_id)
                if st is None:
                    continue
                if st.prev_token_time is None:
                    # TTFT measured against request arrival (the stream's
                    # own clock: a sequence that finishes on its first
                    # token is already out of scheduler.running)
                    self.metrics.ttft.observe(now - st.arrival_time)
                else:
                    self.metrics.tpot.observe(now - st.prev_token_time)
                st.prev_token_time = now
                self.metrics.generation_tokens.inc()
                st.queue.put_nowait(out)
                if out.finished:
                    self.metrics.record_finished(
                        out.finish_reason, out.num_prompt_tokens,
                        out.num_output_tokens)
                    st.queue.put_nowait(None)  # sentinel
                    self.streams.pop(out.request_id, None)
            sched = self.engine.scheduler
            self.metrics.num_requests_running.set(sched.num_running)
            self.metrics.num_requests_waiting.set(sched.num_waiting)
            alloc = sched.allocator
            self.metrics.gpu_cache_usage_perc.set(
                1.0 - alloc.num_free / max(alloc.num_blocks, 1)
            )
            hits, queries = self.engine.prefix_cache_stats
            if queries:  # counters are monotone: set via inc of the delta
                self.metrics.prefix_cache_queries.inc(
                    queries - self.metrics.prefix_cache_queries._value.get()
                )
                self.metrics.prefix_cache_hits.inc(
                    hits - self.metrics.prefix_cache_hits._value.get()
                )
            if self.engine.spec_enabled:
                self.metrics.spec_draft_tokens.inc(
                    self.engine.spec_drafted_tokens
                    - self.metrics.spec_draft_tokens._value.get()
                )
                self.metrics.spec_accepted_tokens.inc(
                    self.engine.spec_accepted_tokens
                    - self.metrics.spec_accepted_tokens._value.get()
                )

    async def generate_stream(self, request_id: str, token_ids: list[int],
                              sampling: SamplingParams,
                              prefill_addr: str | None = None):
        """Async iterator of StepOutputs for one request. With a
        prefill_addr (disaggregated decode instance), the prompt is prefilled
        remotely and its KV pages pulled before decoding locally."""
        if prefill_addr and self.disagg_mode == "decode":
            from .disagg import remote_prefill

            first, reason, kv = await remote_prefill(
                prefill_addr, request_id, token_ids, sampling,
                transport=self.http_transport,
            )
            st = await self.disagg_inject(
                request_id, token_ids, first, kv, sampling
            )
        else:
            st = self.submit(request_id, token_ids, sampling)
        while True:
            out = await st.queue.get()
            if out is None:
                return
            yield out
