"""Model runner: owns the model, the KV cache tensors, batch preparation,
hipGraph decode capture and sampling. Sized for MI355X: the KV pool is
carved out of the 288 GB HBM3E after weights, per gpu_memory_utilization.

Decode steps replay hipGraphs captured per padded batch size (the decode
inner loop is launch-bound at small batch — profiles/r01): persistent input
buffers, padded rows neutralized via slot_mapping=-1 / seq_len=1.
GEMM algorithm selection uses the shipped TunableOp table
(arks_amd/data/tunableop_gfx950.csv) read-only; set ARKS_TUNABLEOP_TUNE=1
to re-tune on new shapes.
"""

from __future__ import annotations

import os

import torch

from .. import ops
from ..config import EngineConfig, ModelConfig
from ..parallel.comm import get_tp_rank, get_tp_world_size, tp_all_gather
from .forward_batch import ForwardBatch
from .kv_cache import BlockAllocator, PrefixCachingAllocator, kv_cache_block_bytes
from .scheduler import ScheduledBatch
from .sequence import Sequence


class _DecodeGraph:
    def __init__(self, graph, logits, batch_size: int):
        self.graph = graph
        self.logits = logits
        self.batch_size = batch_size


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
        self._setup_tunableop()
        from ..models import create_model  # lazy: breaks the import cycle

        self.model = create_model(self.model_cfg, dtype=self.dtype)
        self._weights_loaded = False
        self.kv_caches: list[tuple[torch.Tensor, torch.Tensor]] = []
        self.num_blocks = 0
        self._graphs: dict[int, "_DecodeGraph"] = {}
        self._graph_pool = None
        self._graph_bufs: dict | None = None
        self.use_graphs = (
            self.device.type == "cuda" and not engine_cfg.enforce_eager
        )

    # capture sizes: padded decode batch sizes with their own graphs
    GRAPH_SIZES = [1, 2, 4, 8, 16, 24, 32, 48, 64, 96, 128, 192, 256]

    def _setup_tunableop(self) -> None:
        if self.device.type != "cuda":
            return
        try:
            import torch.cuda.tunable as tunable

            tunable.enable(True)
            if os.environ.get("ARKS_TUNABLEOP_TUNE") == "1":
                tunable.tuning_enable(True)
            else:
                tunable.tuning_enable(False)
                path = os.path.join(
                    os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                    "data", "tunableop_gfx950.csv",
                )
                if os.path.exists(path):
                    tunable.read_file(path)
        except Exception:
            pass  # older torch: run with default GEMM algos

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
        if self.cfg.quantization == "fp8":
            self.model.quantize_fp8()
        elif self.cfg.quantization is not None:
            raise ValueError(f"unknown quantization {self.cfg.quantization!r}")
        # serving past max_position_embeddings: the rope table must cover
        # every reachable position (OOB table reads fault on GPU)
        if hasattr(self.model, "extend_rope_table"):
            self.model.extend_rope_table(self.cfg.max_model_len)
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
            dtype_bytes=1 if self.cfg.kv_cache_dtype == "fp8" else 2,
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
        kv_dtype = (
            torch.float8_e4m3fn if self.cfg.kv_cache_dtype == "fp8"
            else self.dtype
        )
        self.kv_caches = [
            (
                torch.zeros(shape, dtype=kv_dtype, device=self.device),
                torch.zeros(shape, dtype=kv_dtype, device=self.device),
            )
            for _ in range(mc.num_hidden_layers)
        ]
        if self.use_graphs:
            self._init_graph_buffers()
        if self.cfg.enable_prefix_caching:
            return PrefixCachingAllocator(self.num_blocks, self.cfg.block_size)
        return BlockAllocator(self.num_blocks, self.cfg.block_size)

    # ---------------- disaggregated prefill (KV page transfer) ----------------
    def extract_kv(self, block_table: list[int]) -> torch.Tensor:
        """Gather a sequence's KV pages to one host tensor
        [layers, 2(k/v), nblocks, num_kv_heads, block_size, head_dim] for
        transfer to a decode instance (the reference's PD
        split delegates this to SGLang's disaggregation-mode, reference
        arksdisaggregatedapplication_controller.go:1672-1724)."""
        # sliding-window-dropped pages are -1: clamp to a valid page (the
        # receiver never reads out-of-window content)
        bt = torch.tensor(block_table, dtype=torch.long,
                          device=self.device).clamp(min=0)
        layers = [torch.stack((kc[bt], vc[bt]), 0) for kc, vc in self.kv_caches]
        kv = torch.stack(layers, 0)
        # TP>1: each rank holds a contiguous nkv/tp head slice — all-gather
        # to the full head set so the wire format is TP-degree independent
        # (prefill TP degree need not match the decode instance's).
        kv = tp_all_gather(kv, dim=3)
        return kv.cpu()

    def inject_kv(self, block_table: list[int], kv: torch.Tensor) -> None:
        """Scatter a transferred KV tensor (extract_kv layout, full head
        set) into this runner's pages — each TP rank takes its head slice."""
        nkv_local = self.kv_caches[0][0].shape[1]
        world, rank = get_tp_world_size(), get_tp_rank()
        assert kv.shape[0] == len(self.kv_caches) and kv.shape[2] == len(block_table)
        assert kv.shape[3] == nkv_local * world, (
            f"KV head count {kv.shape[3]} != {nkv_local}*tp{world}"
        )
        if world > 1:
            kv = kv[:, :, :, rank * nkv_local:(rank + 1) * nkv_local]
        bt = torch.tensor(block_table, dtype=torch.long, device=self.device)
        kv = kv.to(device=self.device, dtype=self.kv_caches[0][0].dtype,
                   non_blocking=True)
        for li, (kc, vc) in enumerate(self.kv_caches):
            kc[bt] = kv[li, 0]
            vc[bt] = kv[li, 1]

    # ---------------- batch prep ----------------
    def prepare_batch(self, sb: ScheduledBatch) -> ForwardBatch:
        bs = self.cfg.block_size
        if sb.is_prefill:
            # Cached prefixes (prefix cache hits) are skipped: only the new
            # suffix runs as q rows; attention then reads the full paged KV
            # through the extend kernel. On GPU, EVERY prefill takes the
            # extend path (the 8-wave ladder kernel + kv-split live there;
            # the contiguous-varlen kernel is slower and pages are written
            # either way); the CPU reference keeps the contiguous path
            # exercised for fresh prefills.
            use_extend = (self.device.type == "cuda"
                          or any(s.num_cached_tokens > 0 for s in sb.seqs))
            input_ids: list = []  # np.int64 arrays, one per seq
            positions: list = []
            slot_mapping: list = []
            cu = [0]
            q_lens: list[int] = []
            kv_lens: list[int] = []
            logits_idx: list[int] = []
            lp_meta: list[tuple[int, int, int, int]] = []
            self._lp_prefill = None
            import numpy as np

            for i, seq in enumerate(sb.seqs):
                toks = seq.all_token_ids
                c = seq.num_cached_tokens
                # chunked prefill: run exactly the scheduled token count
                # (== everything remaining unless the budget split it)
                n = c + sb.num_new_tokens[i]
                input_ids.append(np.asarray(toks[c:n], dtype=np.int64))
                pos = np.arange(c, n, dtype=np.int64)
                positions.append(pos)
                # vectorized slot mapping (a per-token Python loop here cost
                # ~2.8 ms of GPU idle per 8k-token chunk)
                btn = np.asarray(seq.block_table, dtype=np.int64)
                slot_mapping.append(btn[pos // bs] * bs + pos % bs)
                cu.append(cu[-1] + (n - c))
                q_lens.append(n - c)
                kv_lens.append(n)
                if seq.sampling.prompt_logprobs and c < seq.num_prompt_tokens:
                    # echo/prompt-scoring: lm_head on EVERY row of this seq
                    # (sampling still reads the last row — see execute())
                    off = len(logits_idx)
                    logits_idx.extend(range(cu[-2], cu[-1]))
                    lp_meta.append((i, off, n - c, c))
                else:
                    logits_idx.append(cu[-1] - 1)
                seq.num_cached_tokens = n
            if lp_meta:
                # per-seq index of its SAMPLING row within the gathered
                # logits (lp seqs contributed a whole span, others one row)
                sample_pos = []
                lp_by_seq = {m[0]: m for m in lp_meta}
                g = 0
                for i in range(len(sb.seqs)):
                    m = lp_by_seq.get(i)
                    if m is not None:
                        sample_pos.append(g + m[2] - 1)
                        g += m[2]
                    else:
                        sample_pos.append(g)
                        g += 1
                self._lp_prefill = (lp_meta, sample_pos)
            dev = self.device
            block_tables = None
            seq_lens = None
            ext_tiles = None
            if use_extend:
                max_blocks = max(len(s.block_table) for s in sb.seqs)
                bt = torch.zeros((len(sb.seqs), max_blocks), dtype=torch.int32)
                for i, seq in enumerate(sb.seqs):
                    bt[i, : len(seq.block_table)] = torch.tensor(
                        seq.block_table, dtype=torch.int32
                    )
                block_tables = bt.to(dev)
                seq_lens = torch.tensor(kv_lens, dtype=torch.int32, device=dev)
                ext_tiles = ops.build_extend_tiles(
                    q_lens, kv_lens,
                    self.cfg.kv_cache_dtype == "fp8", dev,
                    num_q_heads=max(
                        self.model_cfg.num_attention_heads
                        // get_tp_world_size(), 1),
                )
            return ForwardBatch(
                is_prefill=True,
                input_ids=torch.from_numpy(np.concatenate(input_ids)).to(dev),
                positions=torch.from_numpy(np.concatenate(positions)).to(dev),
                slot_mapping=torch.from_numpy(np.concatenate(slot_mapping)).to(dev),
                cu_seqlens=torch.tensor(cu, dtype=torch.int32, device=dev),
                seq_lens_list=q_lens,
                block_tables=block_tables,
                seq_lens=seq_lens,
                ext_tiles=ext_tiles,
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

    # ---------------- hipGraph decode ----------------
    def _init_graph_buffers(self) -> None:
        # round max_num_seqs up to the nearest capture size (capped at 256;
        # MoE decode uses the dense bmm path, capture-safe through the full
        # 256-size graph ladder)
        cap = self.GRAPH_SIZES[-1]
        if self.model_cfg.num_local_experts > 0:
            from ..models.llama_family import MoEMLP

            cap = MoEMLP.DENSE_TOKENS
        bmax = min(cap, max(self.cfg.max_num_seqs, 1))
        self._bmax = next(s for s in self.GRAPH_SIZES if s >= bmax)
        mb = (self.cfg.max_model_len + self.cfg.block_size - 1) // self.cfg.block_size
        dev = self.device
        self._graph_bufs = {
            "input_ids": torch.zeros(self._bmax, dtype=torch.int64, device=dev),
            "positions": torch.zeros(self._bmax, dtype=torch.int64, device=dev),
            "slot_mapping": torch.full((self._bmax,), -1, dtype=torch.int64, device=dev),
            "block_tables": torch.zeros(self._bmax, mb, dtype=torch.int32, device=dev),
            "seq_lens": torch.ones(self._bmax, dtype=torch.int32, device=dev),
            # pinned host staging
            "h_input_ids": torch.zeros(self._bmax, dtype=torch.int64, pin_memory=True),
            "h_positions": torch.zeros(self._bmax, dtype=torch.int64, pin_memory=True),
            "h_slot_mapping": torch.full((self._bmax,), -1, dtype=torch.int64, pin_memory=True),
            "h_block_tables": torch.zeros(self._bmax, mb, dtype=torch.int32, pin_memory=True),
            "h_seq_lens": torch.ones(self._bmax, dtype=torch.int32, pin_memory=True),
        }
        for name in ("input_ids", "positions", "slot_mapping", "block_tables", "seq_lens"):
            self._graph_bufs["np_" + name] = self._graph_bufs["h_" + name].numpy()

    def _capture(self, bs: int) -> "_DecodeGraph":
        b = self._graph_bufs
        fb = ForwardBatch(
            is_prefill=False,
            input_ids=b["input_ids"][:bs],
            positions=b["positions"][:bs],
            slot_mapping=b["slot_mapping"][:bs],
            block_tables=b["block_tables"][:bs],
            seq_lens=b["seq_lens"][:bs],
        )
        # warmup (materializes workspaces outside the graph)
        self.model(fb, self.kv_caches)
        torch.cuda.synchronize(self.device)
        g = torch.cuda.CUDAGraph()
        ctx = (
            torch.cuda.graph(g, pool=self._graph_pool)
            if self._graph_pool is not None
            else torch.cuda.graph(g)
        )
        with ctx:
            logits = self.model(fb, self.kv_caches)
        if self._graph_pool is None:
            self._graph_pool = g.pool()
        return _DecodeGraph(graph=g, logits=logits, batch_size=bs)

    def _graph_decode(self, sb: ScheduledBatch) -> torch.Tensor:
        B = len(sb.seqs)
        bs = next(s for s in self.GRAPH_SIZES if s >= B)
        if bs not in self._graphs:
            self._graphs[bs] = self._capture(bs)
        b = self._graph_bufs
        bsz = self.cfg.block_size
        # numpy views of the pinned staging buffers: scalar stores are ~50x
        # cheaper than torch indexed assignment on the decode hot path
        np_ids = b["np_input_ids"]
        np_pos = b["np_positions"]
        np_slot = b["np_slot_mapping"]
        np_len = b["np_seq_lens"]
        np_bt = b["np_block_tables"]
        for i, seq in enumerate(sb.seqs):
            pos = seq.num_tokens - 1
            np_ids[i] = seq.output_token_ids[-1] if seq.output_token_ids else seq.prompt_token_ids[-1]
            np_pos[i] = pos
            np_slot[i] = seq.block_table[pos // bsz] * bsz + pos % bsz
            np_len[i] = seq.num_tokens
            bt = seq.block_table
            np_bt[i, : len(bt)] = bt
            seq.num_cached_tokens = seq.num_tokens
        if B < bs:  # neutralize padding rows
            np_ids[B:bs] = 0
            np_pos[B:bs] = 0
            np_slot[B:bs] = -1
            np_len[B:bs] = 1
            np_bt[B:bs, 0] = 0
        for name in ("input_ids", "positions", "slot_mapping", "seq_lens"):
            b[name][:bs].copy_(b["h_" + name][:bs], non_blocking=True)
        b["block_tables"][:bs].copy_(b["h_block_tables"][:bs], non_blocking=True)
        gr = self._graphs[bs]
        gr.graph.replay()
        return gr.logits[:B]

    # ---------------- execution ----------------
    @torch.no_grad()
    def execute(self, sb: ScheduledBatch) -> list[int]:
        """Run one forward + sampling; returns one new token id per seq."""
        if (
            not sb.is_prefill
            and self.use_graphs
            and self._graph_bufs is not None
            and len(sb.seqs) <= self._bmax
        ):
            logits = self._graph_decode(sb)
            return self.sample(logits, sb.seqs)
        fb = self.prepare_batch(sb)
        logits = self.model(fb, self.kv_caches)  # [num_seqs, vocab]
        if sb.is_prefill and self._lp_prefill is not None:
            logits = self._finish_prompt_logprobs(logits, sb)
        return self.sample(logits, sb.seqs)

    _lp_prefill: tuple | None = None

    def _finish_prompt_logprobs(self, logits: torch.Tensor,
                                sb: ScheduledBatch) -> torch.Tensor:
        """Score prompt tokens for echo/prompt_logprobs requests: the row at
        position p predicts token p+1, so targets 1..P-1 accumulate across
        prefill chunks. Returns the [num_seqs, vocab] sampling logits."""
        lp_meta, sample_pos = self._lp_prefill
        self._lp_prefill = None
        for i, off, qn, c in lp_meta:
            seq = sb.seqs[i]
            P = seq.num_prompt_tokens
            toks = seq.all_token_ids
            lsm = torch.log_softmax(logits[off:off + qn].float(), dim=-1)
            vals = seq.prompt_logprob_values
            for j in range(qn):
                q = c + j + 1  # target position scored by this row
                if q > P - 1:
                    break
                if q - 1 < len(vals):
                    continue  # already scored (preemption recompute)
                vals.append(float(lsm[j, toks[q]]))
        idx = torch.tensor(sample_pos, dtype=torch.long, device=logits.device)
        return logits[idx]

    def execute_spec(self, sb: ScheduledBatch,
                     drafts: list[list[int]]) -> list[list[int]]:
        """Speculative decode step: verify each sequence's draft tokens in
        ONE extend-attention forward (row j scores position n-1+j given the
        draft prefix) and return the emitted tokens per seq — the longest
        agreeing draft prefix plus the model's own token at the first
        disagreement (or the bonus token after a full match). Greedy-exact;
        non-eligible seqs ride along with an empty draft and go through the
        full sampler on their single row."""
        from .spec import eligible

        bs = self.cfg.block_size
        dev = self.device
        input_ids: list[int] = []
        positions: list[int] = []
        slot_mapping: list[int] = []
        cu = [0]
        q_lens: list[int] = []
        kv_lens: list[int] = []
        for seq, d in zip(sb.seqs, drafts):
            toks = [seq.last_token()] + d
            start = seq.num_tokens - 1
            input_ids += toks
            positions += range(start, start + len(toks))
            for pos in range(start, start + len(toks)):
                slot_mapping.append(seq.block_table[pos // bs] * bs + pos % bs)
            cu.append(cu[-1] + len(toks))
            q_lens.append(len(toks))
            kv_lens.append(seq.num_tokens + len(d))
            seq.num_cached_tokens = seq.num_tokens
        max_blocks = max(len(s.block_table) for s in sb.seqs)
        bt = torch.zeros((len(sb.seqs), max_blocks), dtype=torch.int32)
        for i, seq in enumerate(sb.seqs):
            bt[i, : len(seq.block_table)] = torch.tensor(
                seq.block_table, dtype=torch.int32
            )
        fb = ForwardBatch(
            is_prefill=True,
            input_ids=torch.tensor(input_ids, dtype=torch.int64, device=dev),
            positions=torch.tensor(positions, dtype=torch.int64, device=dev),
            slot_mapping=torch.tensor(slot_mapping, dtype=torch.int64, device=dev),
            cu_seqlens=torch.tensor(cu, dtype=torch.int32, device=dev),
            seq_lens_list=q_lens,
            block_tables=bt.to(dev),
            seq_lens=torch.tensor(kv_lens, dtype=torch.int32, device=dev),
            # logits at EVERY row, not just the last per seq
            logits_indices=torch.arange(cu[-1], dtype=torch.int64, device=dev),
        )
        logits = self.model(fb, self.kv_caches)  # [rows, vocab]
        greedy = ops.greedy_sample(logits.contiguous()).tolist()
        emitted: list[list[int] | None] = [None] * len(sb.seqs)
        # sampled-eligible seqs with drafts: exact rejection sampling
        from .spec import eligible_sampled, reject_sample_token

        rej = [i for i, (s, d) in enumerate(zip(sb.seqs, drafts))
               if d and not eligible(s) and eligible_sampled(s)]
        for i in rej:
            seq, d = sb.seqs[i], drafts[i]
            qn = cu[i + 1] - cu[i]
            rows = logits[cu[i]:cu[i + 1]].float()
            sp = seq.sampling
            if sp.top_p < 1.0 or sp.top_k > 0 or sp.min_p > 0.0:
                rows = self._apply_top_p_top_k(rows, [seq] * qn)
            probs = torch.softmax(rows / sp.temperature, dim=-1)
            u = torch.rand(2, qn, device=logits.device).tolist()
            out: list[int] = []
            for j, dt in enumerate(d):
                ok, tok = reject_sample_token(probs[j], dt, u[0][j], u[1][j])
                out.append(tok)
                if not ok:
                    break
            else:
                # full acceptance: bonus token by inverse CDF of the last row
                cdfb = torch.cumsum(probs[qn - 1], 0)
                out.append(int(torch.searchsorted(
                    cdfb, torch.tensor(u[1][qn - 1], dtype=cdfb.dtype,
                                       device=cdfb.device)
                ).clamp(max=cdfb.numel() - 1)))
            emitted[i] = out
        other = [i for i, s in enumerate(sb.seqs)
                 if not eligible(s) and emitted[i] is None]
        if other:
            rows = torch.tensor([cu[i + 1] - 1 for i in other],
                                dtype=torch.long, device=logits.device)
            sub = self.sample(logits[rows], [sb.seqs[i] for i in other])
            lp = self._last_logprobs
            if lp:  # remap sampler sub-batch indices to batch positions
                self._last_logprobs = {other[j]: v for j, v in lp.items()}
            for j, i in enumerate(other):
                emitted[i] = [sub[j]]
        for i, (seq, d) in enumerate(zip(sb.seqs, drafts)):
            if emitted[i] is not None:
                continue
            g = greedy[cu[i]:cu[i + 1]]
            out: list[int] = []
            for j, dt in enumerate(d):
                out.append(g[j])
                if g[j] != dt:
                    break
            else:
                out.append(g[len(d)])  # bonus token after a full match
            emitted[i] = out
        return emitted  # type: ignore[return-value]

    def sample(self, logits: torch.Tensor, seqs: list[Sequence]) -> list[int]:
        if any(s.sampling.logit_bias or s.sampling.min_tokens for s in seqs):
            logits = self._apply_bias_min_tokens(logits, seqs)
        if any(s.sampling.has_penalties for s in seqs):
            logits = self._apply_penalties(logits, seqs)
        temps = [s.sampling.temperature for s in seqs]
        if all(t == 0.0 for t in temps):
            ids = ops.greedy_sample(logits.contiguous())
        else:
            logits = self._apply_top_p_top_k(logits, seqs)
            t = torch.tensor(temps, dtype=torch.float32, device=logits.device)
            u = torch.rand(
                logits.shape, dtype=torch.float32, device=logits.device
            )
            # per-request seeded rows draw from their own generator
            for i, s in enumerate(seqs):
                if s.sampling.seed is not None:
                    gen = getattr(s, "_rng", None)
                    if gen is None:
                        gen = torch.Generator(device=logits.device)
                        gen.manual_seed(s.sampling.seed)
                        s._rng = gen
                    u[i] = torch.rand(
                        logits.shape[1], dtype=torch.float32,
                        device=logits.device, generator=gen,
                    )
            ids = ops.sample_tokens(logits.contiguous(), t, u)
        out = ids.tolist()
        self._last_logprobs = self._gather_logprobs(logits, seqs, out)
        return out

    _last_logprobs: dict | None = None

    def take_logprobs(self) -> dict:
        """Per-seq-index logprob info from the last sample() call."""
        lp, self._last_logprobs = self._last_logprobs, None
        return lp or {}

    @staticmethod
    def _gather_logprobs(logits, seqs, chosen: list[int]) -> dict | None:
        rows = [i for i, s in enumerate(seqs) if s.sampling.logprobs is not None]
        if not rows:
            return None
        out: dict[int, tuple[float, dict[int, float] | None]] = {}
        idx = torch.tensor(rows, dtype=torch.long, device=logits.device)
        lps = torch.log_softmax(logits[idx].float(), dim=-1)
        for j, i in enumerate(rows):
            n = seqs[i].sampling.logprobs
            chosen_lp = float(lps[j, chosen[i]])
            top = None
            if n:
                v, t = lps[j].topk(n)
                top = {int(ti): float(vi) for vi, ti in zip(v, t)}
            out[i] = (chosen_lp, top)
        return out

    def _apply_bias_min_tokens(self, logits: torch.Tensor,
                               seqs: list[Sequence]) -> torch.Tensor:
        """OpenAI logit_bias, and min_tokens (mask eos/stop token logits to
        -inf until the sequence has emitted min_tokens — vLLM semantics)."""
        logits = logits.clone()
        eos = self.model_cfg.eos_token_id
        for i, seq in enumerate(seqs):
            sp = seq.sampling
            if sp.logit_bias:
                for t, b in sp.logit_bias.items():
                    logits[i, t] += b
            if sp.min_tokens and seq.num_output_tokens < sp.min_tokens:
                logits[i, eos] = float("-inf")
                for t in sp.stop_token_ids:
                    logits[i, t] = float("-inf")
        return logits

    @staticmethod
    def _apply_penalties(logits: torch.Tensor, seqs: list[Sequence]) -> torch.Tensor:
        """OpenAI presence/frequency penalties (on generated tokens) and
        vLLM-style repetition penalty (prompt + generated)."""
        logits = logits.clone()
        dev = logits.device
        for i, seq in enumerate(seqs):
            sp = seq.sampling
            if not sp.has_penalties:
                continue
            row = logits[i].float()
            if sp.presence_penalty or sp.frequency_penalty:
                if seq.output_token_ids:
                    ids, counts = torch.unique(
                        torch.tensor(seq.output_token_ids, device=dev),
                        return_counts=True,
                    )
                    row[ids] -= sp.presence_penalty
                    row[ids] -= sp.frequency_penalty * counts.float()
            if sp.repetition_penalty != 1.0:
                seen = torch.unique(torch.tensor(seq.all_token_ids, device=dev))
                vals = row[seen]
                row[seen] = torch.where(
                    vals > 0, vals / sp.repetition_penalty,
                    vals * sp.repetition_penalty,
                )
            logits[i] = row.to(logits.dtype)
        return logits

    @staticmethod
    def _apply_top_p_top_k(logits: torch.Tensor, seqs: list[Sequence]) -> torch.Tensor:
        """Mask logits outside each row's top-p nucleus / top-k set (rows
        with temperature 0 or no constraint pass through)."""
        rows = [
            i for i, s in enumerate(seqs)
            if s.sampling.temperature > 0.0
            and (s.sampling.top_p < 1.0 or s.sampling.top_k > 0
                 or s.sampling.min_p > 0.0)
        ]
        if not rows:
            return logits
        logits = logits.clone()
        idx = torch.tensor(rows, dtype=torch.long, device=logits.device)
        sub = logits[idx].float()
        sorted_logits, sorted_idx = sub.sort(dim=-1, descending=True)
        keep = torch.ones_like(sorted_logits, dtype=torch.bool)
        # Temperature applies BEFORE the top-p/min-p nucleus (vLLM
        # semantics): the kept set is computed from softmax(logits/T).
        # Sort order is T-invariant, so only the probs need scaling; the
        # masked logits stay unscaled (sampling divides by T later).
        temps = torch.tensor(
            [seqs[i].sampling.temperature for i in rows],
            dtype=torch.float32, device=logits.device,
        ).unsqueeze(1)
        probs = torch.softmax(sorted_logits / temps, dim=-1)
        cum = probs.cumsum(dim=-1)
        for j, i in enumerate(rows):
            sp = seqs[i].sampling
            if sp.top_p < 1.0:
                # keep tokens while cumulative prob (exclusive) < top_p
                keep[j] &= (cum[j] - probs[j]) < sp.top_p
            if sp.top_k > 0:
                keep[j, sp.top_k:] = False
            if sp.min_p > 0.0:
                # vLLM min_p: drop tokens with prob < min_p * max prob
                keep[j] &= probs[j] >= sp.min_p * probs[j, 0]
            keep[j, 0] = True  # always keep the best token
        sub = sub.masked_fill(~keep, float("-inf"))
        # scatter back to original column order
        restored = torch.full_like(sub, float("-inf"))
        restored.scatter_(1, sorted_idx, sub)
        logits[idx] = restored.to(logits.dtype)
        return logits
