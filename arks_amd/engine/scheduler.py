"""Continuous-batching scheduler (the subset of vLLM's engineering the
runtime slot needs — SURVEY.md §7 'hard parts' #2).

Policy (v0, vLLM-v0-like):
- Prefill-priority: if waiting requests fit the token budget and KV blocks,
  schedule them as one varlen prefill batch.
- Otherwise run one decode step over all RUNNING sequences, allocating one
  new block per sequence when it crosses a page boundary.
- On KV exhaustion, preempt the most-recently-admitted sequence
  (free its pages, recompute later) until the rest fits.
"""

from __future__ import annotations

from collections import deque
from dataclasses import dataclass, field

from .kv_cache import BlockAllocator
from .sequence import Sequence, SeqStatus


@dataclass
class ScheduledBatch:
    seqs: list[Sequence]
    is_prefill: bool
    # number of new tokens to run for each seq (prefill: whole prompt;
    # decode: 1)
    num_new_tokens: list[int] = field(default_factory=list)

    @property
    def total_tokens(self) -> int:
        return sum(self.num_new_tokens)


class Scheduler:
    def __init__(
        self,
        allocator: BlockAllocator,
        max_num_seqs: int = 256,
        max_num_batched_tokens: int = 8192,
        max_model_len: int = 8192,
        mixed_batching: bool = True,
    ):
        self.allocator = allocator
        self.max_num_seqs = max_num_seqs
        self.max_num_batched_tokens = max_num_batched_tokens
        self.max_model_len = max_model_len
        self.mixed_batching = mixed_batching
        self.waiting: deque[Sequence] = deque()
        self.running: list[Sequence] = []
        # finished seqs with hold_pages: pages stay allocated until the
        # disaggregation layer extracts their KV (release via take_held)
        self.held: dict[str, Sequence] = {}
        self.num_preemptions = 0

    # --- API ---
    def add(self, seq: Sequence) -> None:
        if seq.num_prompt_tokens > self.max_model_len:
            raise ValueError(
                f"prompt length {seq.num_prompt_tokens} exceeds max_model_len "
                f"{self.max_model_len}"
            )
        if (
            BlockAllocator.blocks_needed(seq.num_prompt_tokens, self.allocator.block_size)
            > self.allocator.num_blocks
        ):
            raise ValueError("prompt does not fit in the KV cache at all")
        seq.status = SeqStatus.WAITING
        self.waiting.append(seq)

    def abort(self, request_id: str) -> bool:
        for i, s in enumerate(self.waiting):
            if s.request_id == request_id:
                s.status = SeqStatus.ABORTED
                self._release(s)  # chunked prefills hold pages while WAITING
                del self.waiting[i]
                return True
        for s in self.running:
            if s.request_id == request_id:
                s.status = SeqStatus.ABORTED
                self._release(s)
                self.running.remove(s)
                return True
        return False

    def has_work(self) -> bool:
        return bool(self.waiting) or bool(self.running)

    @property
    def num_waiting(self) -> int:
        return len(self.waiting)

    @property
    def num_running(self) -> int:
        return len(self.running)

    # --- scheduling ---
    def schedule(self) -> ScheduledBatch | None:
        """One step's batch. With work waiting AND sequences decoding, build
        a MIXED batch: every running sequence contributes its next token and
        the remaining token budget admits prefill chunks — all through the
        paged extend path, so a long prefill never stalls in-flight decodes
        (decode-only steps still take the hipGraph fast path)."""
        if self.waiting and self.running and self.mixed_batching:
            decode_part = self._schedule_decode()
            if decode_part is not None:
                budget_left = self.max_num_batched_tokens - len(decode_part.seqs)
                chunk_part = (
                    self._schedule_prefill(budget=max(budget_left, 0))
                    if budget_left > 0 else None
                )
                if chunk_part is None:
                    return decode_part
                for seq in decode_part.seqs:
                    seq.num_cached_tokens = seq.num_tokens - 1
                return ScheduledBatch(
                    seqs=decode_part.seqs + chunk_part.seqs,
                    is_prefill=True,  # runs the paged extend path
                    num_new_tokens=[1] * len(decode_part.seqs)
                    + chunk_part.num_new_tokens,
                )
        batch = self._schedule_prefill()
        if batch is not None:
            return batch
        return self._schedule_decode()

    def _schedule_prefill(self, budget: int | None = None) -> ScheduledBatch | None:
        seqs: list[Sequence] = []
        ntoks: list[int] = []
        if budget is None:
            budget = self.max_num_batched_tokens
        while self.waiting and budget > 0:
            seq = self.waiting[0]
            # num_tokens (not num_prompt_tokens): a preempted sequence is
            # recomputed over prompt + already-generated tokens.
            n = seq.num_tokens
            toks = seq.all_token_ids
            if len(self.running) + len(seqs) >= self.max_num_seqs:
                break
            if not seq.block_table:
                # Fresh admission: reuse cached full blocks of the prompt
                # (refs taken here must be released on every non-admit
                # path). Cap at n-1 so at least one token produces logits.
                # prompt_logprobs needs logits at EVERY prompt position, so
                # those requests skip prefix reuse and recompute in full.
                if seq.sampling.prompt_logprobs:
                    reused, ncached = [], 0
                else:
                    reused, ncached = self.allocator.match_prefix(toks, n - 1)
                seq.num_cached_tokens = ncached
            else:
                # Continuation of a chunked prefill: pages for the covered
                # range are already held.
                reused, ncached = [], seq.num_cached_tokens
            n_new = n - ncached
            # Chunked prefill: run at most `budget` new tokens this step;
            # the remainder stays WAITING (the extend kernel attends the
            # chunk over the full paged prefix).
            n_run = min(n_new, budget)
            partial = n_run < n_new
            covered = ncached + n_run
            need = (
                BlockAllocator.blocks_needed(covered, self.allocator.block_size)
                - len(reused) - len(seq.block_table)
            )
            if not self.allocator.can_allocate(max(need, 0)):
                self.allocator.free(reused)
                break
            if reused:
                seq.block_table = reused + seq.block_table
            if need > 0:
                seq.block_table.extend(self.allocator.allocate(need))
            self.allocator.register_prefix(
                toks[:covered], seq.block_table, len(reused)
            )
            if not partial:
                self.waiting.popleft()
                seq.status = SeqStatus.RUNNING
            seqs.append(seq)
            ntoks.append(n_run)
            budget -= n_run
            if partial:
                break  # budget exhausted on this seq
        if not seqs:
            return None
        # partial (chunked) seqs stay in `waiting`; only completed prefills
        # enter the decode set
        self.running.extend(s for s in seqs if s.status is SeqStatus.RUNNING)
        return ScheduledBatch(seqs=seqs, is_prefill=True, num_new_tokens=ntoks)

    def _schedule_decode(self) -> ScheduledBatch | None:
        if not self.running:
            return None
        # Ensure each running sequence has a slot for its next token;
        # preempt from the back (most recent) on exhaustion.
        scheduled: list[Sequence] = []
        i = 0
        while i < len(self.running):
            seq = self.running[i]
            need = BlockAllocator.blocks_needed(
                seq.num_tokens, self.allocator.block_size
            ) - len(seq.block_table)
            if need > 0 and not self.allocator.can_allocate(need):
                # Victims must be unvisited (not already in this batch).
                if not self._preempt_last(exclude=seq, protected=scheduled):
                    # nothing left to preempt but this seq itself
                    self._preempt(seq)
                    self.running.remove(seq)
                    continue
                continue  # retry same seq
            if need > 0:
                seq.block_table.extend(self.allocator.allocate(need))
            scheduled.append(seq)
            i += 1
        if not scheduled:
            return None
        return ScheduledBatch(
            seqs=scheduled, is_prefill=False, num_new_tokens=[1] * len(scheduled)
        )

    def _preempt_last(self, exclude: Sequence, protected: list[Sequence]) -> bool:
        for seq in reversed(self.running):
            if seq is exclude or seq in protected:
                continue
            self._preempt(seq)
            self.running.remove(seq)
            return True
        return False

    def _preempt(self, seq: Sequence) -> None:
        self.num_preemptions += 1
        self._release(seq)
        seq.status = SeqStatus.WAITING
        # Recompute-style preemption: prompt grows to include generated tokens
        # so the whole context is prefil­led again on readmission.
        seq.num_cached_tokens = 0
        self.waiting.appendleft(seq)

    def _release(self, seq: Sequence) -> None:
        if seq.block_table:
            self.allocator.free(seq.block_table)
            seq.block_table = []
        seq._draft_len = 0  # draft-model KV mirror follows the pages

    def free_finished(self) -> None:
        for seq in self.running:
            if seq.is_finished:
                if seq.hold_pages:
                    self.held[seq.request_id] = seq
                else:
                    self._release(seq)
        self.running = [s for s in self.running if not s.is_finished]

    def take_held(self, request_id: str) -> Sequence | None:
        """Remove and return a held finished sequence (pages still ref'd;
        caller must allocator.free(seq.block_table) when done)."""
        return self.held.pop(request_id, None)
