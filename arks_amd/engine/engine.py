"""LLMEngine: the synchronous continuous-batching core.

add_request() -> step() loop -> StepOutputs. The OpenAI server wraps this in
an asyncio loop (arks_amd.server); bench.py drives it directly.
"""

from __future__ import annotations

import time

from ..config import EngineConfig
from .model_runner import ModelRunner
from .scheduler import Scheduler
from .sequence import SamplingParams, Sequence, SeqStatus, StepOutput


class LLMEngine:
    def __init__(self, cfg: EngineConfig):
        self.cfg = cfg
        self.model_cfg = cfg.model_config()
        # Sliding-window attention: each layer masks keys outside its
        # window (per-layer Qwen2 mixing honored via
        # ModelConfig.layer_window). KV page dropping + the prefix-cache
        # gate apply only when EVERY layer slides (uniform_window) — a
        # single full-attention layer keeps the whole history live, and a
        # cached prefix page must never be dropped+recycled while its
        # digest is still matchable.
        self.window = self.model_cfg.uniform_window()
        if self.window and cfg.enable_prefix_caching:
            cfg.enable_prefix_caching = False
        self.runner = ModelRunner(cfg, self.model_cfg)
        t0 = time.time()
        self.runner.load_weights()
        self.load_seconds = time.time() - t0
        allocator = self.runner.init_kv_cache()
        self.scheduler = Scheduler(
            allocator,
            max_num_seqs=cfg.max_num_seqs,
            max_num_batched_tokens=cfg.max_num_batched_tokens,
            max_model_len=cfg.max_model_len,
            mixed_batching=cfg.enable_mixed_batching,
        )
        # serving metrics
        self.total_prompt_tokens = 0
        self.total_output_tokens = 0
        # speculative decoding (prompt-lookup or draft-model) stats
        self.spec_enabled = cfg.speculative in ("ngram", "draft")
        self.spec_drafted_tokens = 0
        self.spec_accepted_tokens = 0
        self.draft = None
        if cfg.speculative == "draft":
            from .draft import DraftRunner

            self.draft = DraftRunner(cfg, self.model_cfg, self.runner.num_blocks,
                                     self.runner.device, self.runner.dtype)

    @property
    def prefix_cache_stats(self) -> tuple[int, int]:
        """(hit_tokens, query_tokens) of the prefix cache (0, 0 if disabled)."""
        alloc = self.scheduler.allocator
        return getattr(alloc, "hit_tokens", 0), getattr(alloc, "query_tokens", 0)

    # ---- request API ----
    def add_request(
        self,
        prompt_token_ids: list[int],
        sampling: SamplingParams | None = None,
        request_id: str | None = None,
        arrival_time: float | None = None,
        hold_pages: bool = False,
    ) -> Sequence:
        seq = Sequence(prompt_token_ids, sampling, request_id, arrival_time)
        seq.hold_pages = hold_pages
        self.scheduler.add(seq)
        self.total_prompt_tokens += seq.num_prompt_tokens
        return seq

    def abort_request(self, request_id: str) -> bool:
        return self.scheduler.abort(request_id)

    # ---- prefill/decode disaggregation ----
    def extract_prefilled(self, request_id: str):
        """After a hold_pages request finished (1 token), pull its KV pages
        for transfer and release them. Returns (prompt_len, kv tensor)."""
        seq = self.scheduler.take_held(request_id)
        if seq is None:
            raise KeyError(f"no held sequence {request_id!r}")
        # KV covers the prompt positions only (the generated token's KV is
        # written by the decode instance's first step).
        nb = (seq.num_prompt_tokens + self.cfg.block_size - 1) // self.cfg.block_size
        kv = self.runner.extract_kv(seq.block_table[:nb])
        self.scheduler.allocator.free(seq.block_table)
        seq.block_table = []
        return seq.num_prompt_tokens, kv

    def add_prefilled(
        self,
        prompt_token_ids: list[int],
        first_token: int,
        kv,
        sampling: SamplingParams | None = None,
        request_id: str | None = None,
        arrival_time: float | None = None,
    ) -> Sequence:
        """Decode-instance entry: admit a sequence whose prompt KV was
        computed remotely. Injects the pages and enters RUNNING directly."""
        from .kv_cache import BlockAllocator

        seq = Sequence(prompt_token_ids, sampling, request_id, arrival_time)
        need = BlockAllocator.blocks_needed(seq.num_prompt_tokens, self.cfg.block_size)
        if not self.scheduler.allocator.can_allocate(need):
            raise RuntimeError("KV cache exhausted on decode instance")
        seq.block_table = self.scheduler.allocator.allocate(need)
        self.runner.inject_kv(seq.block_table, kv)
        self.scheduler.allocator.register_prefix(
            prompt_token_ids, seq.block_table, 0
        )
        seq.num_cached_tokens = seq.num_prompt_tokens
        seq.append_token(first_token)
        self.total_prompt_tokens += seq.num_prompt_tokens
        self.total_output_tokens += 1
        if seq.check_finished(self.model_cfg.eos_token_id):
            self.scheduler._release(seq)
        else:
            seq.status = SeqStatus.RUNNING
            self.scheduler.running.append(seq)
        return seq

    def has_work(self) -> bool:
        return self.scheduler.has_work()

    # ---- the step loop ----
    def step(self) -> list[StepOutput]:
        if self.window:
            self._drop_window_pages()
        sb = self.scheduler.schedule()
        if sb is None:
            return []
        if self.spec_enabled and not sb.is_prefill:
            return self._spec_step(sb)
        new_tokens = self.runner.execute(sb)
        logprobs = self.runner.take_logprobs()
        outputs: list[StepOutput] = []
        eos = self.model_cfg.eos_token_id
        for i, (seq, tok) in enumerate(zip(sb.seqs, new_tokens)):
            if seq.status is not SeqStatus.RUNNING:  # aborted mid-step
                continue
            seq.append_token(tok)
            self.total_output_tokens += 1
            finished = seq.check_finished(eos)
            if not finished and seq.num_tokens >= self.cfg.max_model_len:
                # context-window cap: never grow past max_model_len (the
                # graph block-table buffers are sized to it)
                seq.finish_reason = "length"
                seq.status = SeqStatus.FINISHED
                seq.finish_time = time.time()
                finished = True
            lp, top = logprobs.get(i, (None, None))
            outputs.append(
                StepOutput(
                    request_id=seq.request_id,
                    seq_id=seq.seq_id,
                    new_token_id=tok,
                    finished=finished,
                    finish_reason=seq.finish_reason,
                    num_prompt_tokens=seq.num_prompt_tokens,
                    num_output_tokens=seq.num_output_tokens,
                    logprob=lp,
                    top_logprobs=top,
                    prompt_logprobs=(
                        list(seq.prompt_logprob_values)
                        if seq.sampling.prompt_logprobs
                        and seq.num_output_tokens == 1 else None
                    ),
                )
            )
        self.scheduler.free_finished()
        return outputs

    def _drop_window_pages(self) -> None:
        """Free KV pages wholly below every running sequence's attention
        window (block-table entries become -1; the kernels never read them).
        Keeps long SWA generations at O(window) KV instead of O(length)."""
        bs = self.cfg.block_size
        for seq in self.scheduler.running:
            if seq.num_tokens <= self.window:
                continue
            # pages [0, lim) hold only tokens < num_tokens - window
            lim = (seq.num_tokens - self.window) // bs
            start = getattr(seq, "window_dropped", 0)
            if lim <= start:
                continue
            drop = [b for b in seq.block_table[start:lim] if b >= 0]
            if drop:
                self.scheduler.allocator.free(drop)
            for p in range(start, lim):
                seq.block_table[p] = -1
            seq.window_dropped = lim

    def _spec_step(self, sb) -> list[StepOutput]:
        """Decode step with prompt-lookup speculation: propose drafts from
        each sequence's history, extend its KV pages to cover them, verify
        in one forward, emit 1..k+1 tokens per seq (greedy-exact)."""
        from .kv_cache import BlockAllocator
        from .spec import eligible, eligible_sampled, propose_ngram_cached

        alloc = self.scheduler.allocator
        bs = self.cfg.block_size
        k = self.cfg.num_speculative_tokens
        drafts: list[list[int]] = []
        if self.draft is not None:
            # draft-model proposal: pages must cover the drafted positions
            # BEFORE propose() (the draft writes its mirrored KV there), so
            # eligibility includes room for the full k and the allocation
            # happens up front
            cand: list = []
            for seq in sb.seqs:
                ok = (seq.num_tokens + k <= self.cfg.max_model_len
                      and (eligible(seq) or eligible_sampled(seq)))
                if ok:
                    need = (BlockAllocator.blocks_needed(seq.num_tokens + k, bs)
                            - len(seq.block_table))
                    if need > 0:
                        if alloc.can_allocate(need):
                            seq.block_table.extend(alloc.allocate(need))
                        else:
                            ok = False
                cand.append(ok)
            picked = [s for s, ok in zip(sb.seqs, cand) if ok]
            proposed = self.draft.propose(picked, k) if picked else []
            it = iter(proposed)
            drafts = [next(it) if ok else [] for ok in cand]
        else:
            for seq in sb.seqs:
                d: list[int] = []
                room = self.cfg.max_model_len - seq.num_tokens
                if room > 0 and (eligible(seq) or eligible_sampled(seq)):
                    d = propose_ngram_cached(seq, min(k, room))
                if d:
                    need = (
                        BlockAllocator.blocks_needed(seq.num_tokens + len(d), bs)
                        - len(seq.block_table)
                    )
                    if need > 0:
                        if alloc.can_allocate(need):
                            seq.block_table.extend(alloc.allocate(need))
                        else:
                            d = []  # no KV room — plain decode this step
                drafts.append(d)
        if not any(drafts):
            # nothing to verify: take the normal (hipGraph) decode path
            emitted = [[t] for t in self.runner.execute(sb)]
        else:
            emitted = self.runner.execute_spec(sb, drafts)
        logprobs = self.runner.take_logprobs()
        outputs: list[StepOutput] = []
        eos = self.model_cfg.eos_token_id
        for i, (seq, toks) in enumerate(zip(sb.seqs, emitted)):
            if seq.status is not SeqStatus.RUNNING:  # aborted mid-step
                continue
            self.spec_drafted_tokens += len(drafts[i])
            if drafts[i]:
                self.spec_accepted_tokens += len(toks) - 1
                if self.draft is not None:
                    # accepted drafted positions already hold valid draft
                    # KV; the correction/bonus token catches up next round
                    seq._draft_len += len(toks) - 1
            for tok in toks:
                seq.append_token(tok)
                self.total_output_tokens += 1
                finished = seq.check_finished(eos)
                if not finished and seq.num_tokens >= self.cfg.max_model_len:
                    seq.finish_reason = "length"
                    seq.status = SeqStatus.FINISHED
                    seq.finish_time = time.time()
                    finished = True
                lp, top = logprobs.get(i, (None, None))
                outputs.append(
                    StepOutput(
                        request_id=seq.request_id,
                        seq_id=seq.seq_id,
                        new_token_id=tok,
                        finished=finished,
                        finish_reason=seq.finish_reason,
                        num_prompt_tokens=seq.num_prompt_tokens,
                        num_output_tokens=seq.num_output_tokens,
                        logprob=lp,
                        top_logprobs=top,
                    )
                )
                if finished:
                    break
        self.scheduler.free_finished()
        return outputs

    # ---- convenience: run to completion (tests / offline use) ----
    def generate(
        self, prompts: list[list[int]], sampling: SamplingParams | None = None
    ) -> list[list[int]]:
        seqs = [self.add_request(p, sampling) for p in prompts]
        while self.has_work():
            self.step()
        return [s.output_token_ids for s in seqs]
