"""Draft-model speculative decoding (the vLLM `--speculative-model <path>`
capability of the reference's runtime slot; the ngram/prompt-lookup
variant lives in spec.py — both feed the same verify forward in
model_runner.execute_spec, so acceptance stays greedy-/distribution-exact).

Design: the draft model keeps its own paged KV pool with the SAME block
ids as the target (the pool is allocated with the target allocator's
num_blocks, and the draft simply indexes pages by the sequence's existing
block_table). No second allocator, no extra tables: when the engine
extends or frees a sequence's pages, the draft's pages follow.

Per-sequence sync: `seq._draft_len` = how many committed tokens the draft
KV covers. A propose() round first catches up (one extend forward over
tokens[draft_len:]), then greedily drafts k tokens with k-1 decode
forwards, writing draft KV for the drafted positions into the mirrored
pages (positions beyond the committed length are scratch: a later
catch-up overwrites them, and attention never reads past the declared kv
length). After the target verifies, the engine advances _draft_len by
the accepted count — accepted drafted positions already hold valid draft
KV, the correction/bonus token is caught up next round. Preemption frees
the pages and resets _draft_len via the scheduler's release hook.
"""

from __future__ import annotations

import torch

from .forward_batch import ForwardBatch
from .sequence import Sequence


class DraftRunner:
    def __init__(self, engine_cfg, target_cfg, num_blocks: int,
                 device: torch.device, dtype=torch.bfloat16):
        from ..config import ModelConfig, PRESET_CONFIGS
        from ..models import create_model
        from ..parallel.comm import get_tp_world_size

        spec = engine_cfg.draft_model or ""
        if not spec:
            raise ValueError("speculative='draft' requires draft_model")
        if spec.startswith("preset:"):
            dcfg = PRESET_CONFIGS[spec.split(":", 1)[1]]
            path = None
        else:
            dcfg = ModelConfig.from_pretrained(spec)
            path = spec
        if dcfg.vocab_size != target_cfg.vocab_size:
            raise ValueError(
                f"draft vocab {dcfg.vocab_size} != target "
                f"{target_cfg.vocab_size} (tokenizers must match)")
        if dcfg.uniform_window() or target_cfg.uniform_window():
            raise ValueError(
                "draft-model speculation with sliding-window page dropping "
                "is not supported")
        tp = get_tp_world_size()
        # TP: the engine runs SPMD (every rank executes _spec_step), so the
        # sharded draft forward's collectives stay in lockstep and greedy
        # proposals are rank-identical (tests/test_tp_gloo.py TP2 case)
        if dcfg.num_key_value_heads % max(tp, 1) != 0:
            raise ValueError("draft kv heads not divisible by TP degree")
        self.cfg = dcfg
        self.device = device
        self.block_size = engine_cfg.block_size
        self.model = create_model(dcfg, dtype=dtype)
        if path:
            from ..loader.safetensors_loader import load_model_weights

            load_model_weights(self.model, path, device)
        else:
            self.model.to(device)
            self.model.random_init(engine_cfg.seed + 1)
        self.model.to(device)
        self.model.eval()
        if hasattr(self.model, "extend_rope_table"):
            self.model.extend_rope_table(engine_cfg.max_model_len)
        # mirrored KV pool: same block ids as the target allocator
        nkv = dcfg.num_key_value_heads // max(tp, 1)
        self.kv_caches = [
            (
                torch.zeros(num_blocks, nkv, self.block_size, dcfg.head_dim,
                            dtype=dtype, device=device),
                torch.zeros(num_blocks, nkv, self.block_size, dcfg.head_dim,
                            dtype=dtype, device=device),
            )
            for _ in range(dcfg.num_hidden_layers)
        ]

    # ---------------- proposal ----------------
    @torch.inference_mode()
    def propose(self, seqs: list[Sequence], k: int) -> list[list[int]]:
        """Greedy k-token drafts for `seqs` (each must have pages covering
        num_tokens + k - 1 positions already). Returns one draft list per
        seq, all of length k."""
        if not seqs or k <= 0:
            return [[] for _ in seqs]
        dev = self.device
        bs = self.block_size

        # --- phase 1: catch-up extend over committed-but-unseen tokens ---
        input_ids: list[int] = []
        positions: list[int] = []
        slots: list[int] = []
        cu = [0]
        q_lens: list[int] = []
        kv_lens: list[int] = []
        for seq in seqs:
            dl = getattr(seq, "_draft_len", 0)
            toks = seq.all_token_ids[dl:seq.num_tokens]
            assert toks, "draft catch-up span empty"
            input_ids += toks
            for pos in range(dl, seq.num_tokens):
                positions.append(pos)
                slots.append(seq.block_table[pos // bs] * bs + pos % bs)
            cu.append(cu[-1] + len(toks))
            q_lens.append(len(toks))
            kv_lens.append(seq.num_tokens)
        max_blocks = max(len(s.block_table) for s in seqs)
        bt = torch.zeros((len(seqs), max_blocks), dtype=torch.int32)
        for i, seq in enumerate(seqs):
            bt[i, : len(seq.block_table)] = torch.tensor(
                seq.block_table, dtype=torch.int32)
        bt = bt.to(dev)
        fb = ForwardBatch(
            is_prefill=True,
            input_ids=torch.tensor(input_ids, dtype=torch.int64, device=dev),
            positions=torch.tensor(positions, dtype=torch.int64, device=dev),
            slot_mapping=torch.tensor(slots, dtype=torch.int64, device=dev),
            cu_seqlens=torch.tensor(cu, dtype=torch.int32, device=dev),
            seq_lens_list=q_lens,
            block_tables=bt,
            seq_lens=torch.tensor(kv_lens, dtype=torch.int32, device=dev),
            logits_indices=torch.tensor(
                [c - 1 for c in cu[1:]], dtype=torch.int64, device=dev),
        )
        logits = self.model(fb, self.kv_caches)
        cur = logits.argmax(dim=-1).tolist()
        drafts = [[t] for t in cur]

        # --- phase 2: k-1 greedy decode forwards over the drafted tail ---
        for j in range(1, k):
            ids = [d[-1] for d in drafts]
            pos = [s.num_tokens + j - 1 for s in seqs]
            slot = [s.block_table[p // bs] * bs + p % bs
                    for s, p in zip(seqs, pos)]
            fb = ForwardBatch(
                is_prefill=False,
                input_ids=torch.tensor(ids, dtype=torch.int64, device=dev),
                positions=torch.tensor(pos, dtype=torch.int64, device=dev),
                slot_mapping=torch.tensor(slot, dtype=torch.int64, device=dev),
                block_tables=bt,
                seq_lens=torch.tensor([p + 1 for p in pos],
                                      dtype=torch.int32, device=dev),
            )
            logits = self.model(fb, self.kv_caches)
            nxt = logits.argmax(dim=-1).tolist()
            for d, t in zip(drafts, nxt):
                d.append(t)

        # draft KV now covers committed history; drafted positions are
        # scratch until the engine advances _draft_len by the accepted count
        for seq in seqs:
            seq._draft_len = seq.num_tokens
        return drafts
