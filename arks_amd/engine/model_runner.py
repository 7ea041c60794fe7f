"""Model runner: owns the model, the KV cache tensors, batch preparation and
sampling. Sized for MI355X: the KV pool is carved out of the 288 GB HBM3E
after weights, per gpu_memory_utilization."""

from __future__ import annotations

import torch

from .. import ops
from ..config import EngineConfig, ModelConfig
from ..models import create_model
from ..parallel.comm import get_tp_world_size
from .forward_batch import ForwardBatch
from .kv_cache import BlockAllocator, kv_cache_block_bytes
from .scheduler import ScheduledBatch
from .sequence import Sequence


class ModelRunner:
    def __init__(self, engine_cfg: EngineConfig, model_cfg: ModelConfig | None = None):
        self.cfg = engine_cfg
        self.model_cfg = model_cfg or engine_cfg.model_config()
        self.device = torch.device(
            engine_cfg.device if torch.cuda.is_available() or engine_cfg.device == "cpu"
            else "cpu"
        )
        self.dtype = torch.bfloat16
        torch.manual_seed(engine_cfg.seed)
        self.model = create_model(self.model_cfg, dtype=self.dtype)
        self._weights_loaded = False
        self.kv_caches: list[tuple[torch.Tensor, torch.Tensor]] = []
        self.num_blocks = 0
        self._rng = None  # device RNG for sampling noise

    # ---------------- initialization ----------------
    def load_weights(self) -> None:
        if self.cfg.model_path:
            from ..loader.safetensors_loader import load_model_weights

            load_model_weights(self.model, self.cfg.model_path, self.device)
        else:
            # move first so random-init generates directly on the GPU
            self.model.to(self.device)
            self.model.random_init(self.cfg.seed)
        self.model.to(self.device)
        self._weights_loaded = True

    def profile_num_blocks(self) -> int:
        """Size the KV pool from free HBM after weights (or the override)."""
        if self.cfg.kv_cache_blocks is not None:
            return self.cfg.kv_cache_blocks
        mc = self.model_cfg
        tp = get_tp_world_size()
        block_bytes = kv_cache_block_bytes(
            mc.num_hidden_layers,
            max(mc.num_key_value_heads // tp, 1),
            mc.head_dim,
            self.cfg.block_size,
        )
        if self.device.type == "cuda":
            free, total = torch.cuda.mem_get_info(self.device)
            budget = int(total * self.cfg.gpu_memory_utilization) - (total - free)
            budget = max(budget, 2 * block_bytes)
        else:
            budget = 64 * 1024 * 1024  # CPU tests: small pool
        return max(budget // block_bytes, 16)

    def init_kv_cache(self) -> BlockAllocator:
        assert self._weights_loaded, "load_weights() first"
        mc = self.model_cfg
        tp = get_tp_world_size()
        self.num_blocks = self.profile_num_blocks()
        nkv = max(mc.num_key_value_heads // tp, 1)
        shape = (self.num_blocks, nkv, self.cfg.block_size, mc.head_dim)
        self.kv_caches = [
            (
                torch.zeros(shape, dtype=self.dtype, device=self.device),
                torch.zeros(shape, dtype=self.dtype, device=self.device),
            )
            for _ in range(mc.num_hidden_layers)
        ]
        return BlockAllocator(self.num_blocks, self.cfg.block_size)

    # ---------------- batch prep ----------------
    def prepare_batch(self, sb: ScheduledBatch) -> ForwardBatch:
        bs = self.cfg.block_size
        if sb.is_prefill:
            input_ids: list[int] = []
            positions: list[int] = []
            slot_mapping: list[int] = []
            cu = [0]
            seq_lens_list: list[int] = []
            logits_idx: list[int] = []
            for seq in sb.seqs:
                toks = seq.all_token_ids
                n = len(toks)
                input_ids.extend(toks)
                positions.extend(range(n))
                for pos in range(n):
                    b = seq.block_table[pos // bs]
                    slot_mapping.append(b * bs + pos % bs)
                cu.append(cu[-1] + n)
                seq_lens_list.append(n)
                logits_idx.append(cu[-1] - 1)
                seq.num_cached_tokens = n
            dev = self.device
            return ForwardBatch(
                is_prefill=True,
                input_ids=torch.tensor(input_ids, dtype=torch.int64, device=dev),
                positions=torch.tensor(positions, dtype=torch.int64, device=dev),
                slot_mapping=torch.tensor(slot_mapping, dtype=torch.int64, device=dev),
                cu_seqlens=torch.tensor(cu, dtype=torch.int32, device=dev),
                seq_lens_list=seq_lens_list,
                logits_indices=torch.tensor(logits_idx, dtype=torch.int64, device=dev),
            )
        # decode
        input_ids = [s.last_token() for s in sb.seqs]
        positions = [s.num_tokens - 1 for s in sb.seqs]
        slot_mapping = []
        seq_lens = []
        max_blocks = max(len(s.block_table) for s in sb.seqs)
        bt = torch.zeros((len(sb.seqs), max_blocks), dtype=torch.int32)
        for i, seq in enumerate(sb.seqs):
            pos = seq.num_tokens - 1  # the new token's position
            slot_mapping.append(seq.block_table[pos // bs] * bs + pos % bs)
            seq_lens.append(seq.num_tokens)
            bt[i, : len(seq.block_table)] = torch.tensor(
                seq.block_table, dtype=torch.int32
            )
            seq.num_cached_tokens = seq.num_tokens
        dev = self.device
        return ForwardBatch(
            is_prefill=False,
            input_ids=torch.tensor(input_ids, dtype=torch.int64, device=dev),
            positions=torch.tensor(positions, dtype=torch.int64, device=dev),
            slot_mapping=torch.tensor(slot_mapping, dtype=torch.int64, device=dev),
            block_tables=bt.to(dev),
            seq_lens=torch.tensor(seq_lens, dtype=torch.int32, device=dev),
        )

    # ---------------- execution ----------------
    @torch.no_grad()
    def execute(self, sb: ScheduledBatch) -> list[int]:
        """Run one forward + sampling; returns one new token id per seq."""
        fb = self.prepare_batch(sb)
        logits = self.model(fb, self.kv_caches)  # [num_seqs, vocab]
        return self.sample(logits, sb.seqs)

    def sample(self, logits: torch.Tensor, seqs: list[Sequence]) -> list[int]:
        temps = [s.sampling.temperature for s in seqs]
        if all(t == 0.0 for t in temps):
            ids = ops.greedy_sample(logits.contiguous())
        else:
            t = torch.tensor(temps, dtype=torch.float32, device=logits.device)
            u = torch.rand(
                logits.shape, dtype=torch.float32, device=logits.device
            )
            ids = ops.sample_tokens(logits.contiguous(), t, u)
        return ids.tolist()
