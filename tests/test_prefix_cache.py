"""Prefix caching: allocator content-addressing, scheduler reuse, engine
equivalence (cached vs uncached outputs identical), extend-attention ref.

The capability mirrors SGLang's radix cache, which the reference delegates to
external runtime images (reference arksapplication_controller.go:956-969);
here it is first-party (arks_amd/engine/kv_cache.py).
"""

import torch

from arks_amd.config import EngineConfig
from arks_amd.engine import LLMEngine, SamplingParams
from arks_amd.engine.kv_cache import BlockAllocator, PrefixCachingAllocator
from arks_amd.engine.scheduler import Scheduler
from arks_amd.engine.sequence import Sequence
from arks_amd.ops import ref


def test_allocator_match_and_refcount():
    a = PrefixCachingAllocator(8, block_size=4)
    toks = list(range(12))
    # no hits on an empty cache
    blocks, n = a.match_prefix(toks, len(toks) - 1)
    assert blocks == [] and n == 0
    got = a.allocate(3)
    a.register_prefix(toks, got, 0)
    # a second identical prompt matches full blocks only, capped below n-1
    blocks, n = a.match_prefix(toks, len(toks) - 1)
    assert n == 8 and blocks == got[:2]  # 12 tokens -> cap 11 -> 2 blocks
    a.free(blocks)  # rollback path
    # free the original seq: its registered blocks park in LRU, still match
    a.free(got)
    assert a.num_free == 8
    blocks, n = a.match_prefix(toks, 12)
    assert n == 12 and blocks == got
    a.free(blocks)


def test_allocator_eviction_drops_content():
    a = PrefixCachingAllocator(4, block_size=4)
    t1 = [1] * 8
    b1 = a.allocate(2)
    a.register_prefix(t1, b1, 0)
    a.free(b1)  # 2 LRU + 2 virgin
    # allocating 4 blocks must evict both cached blocks
    b2 = a.allocate(4)
    assert sorted(b2) == sorted(range(4))
    blocks, n = a.match_prefix(t1, 8)
    assert n == 0
    a.free(b2)


def test_allocator_no_override_register():
    a = PrefixCachingAllocator(8, block_size=4)
    toks = [7] * 4
    b1 = a.allocate(1)
    a.register_prefix(toks, b1, 0)
    b2 = a.allocate(1)
    a.register_prefix(toks, b2, 0)  # duplicate content: first wins
    blocks, n = a.match_prefix(toks, 99)
    assert blocks == b1
    a.free(blocks)
    a.free(b1)
    a.free(b2)
    # b2 was never registered: it recycles as virgin, b1 stays matchable
    blocks, n = a.match_prefix(toks, 99)
    assert blocks == b1 and n == 4


def test_scheduler_prefix_reuse_counts_new_tokens_only():
    a = PrefixCachingAllocator(64, block_size=4)
    s = Scheduler(a, max_num_seqs=8, max_num_batched_tokens=64, max_model_len=256)
    prompt = list(range(40, 60))  # 20 tokens = 5 blocks
    s1 = Sequence(prompt)
    s.add(s1)
    b1 = s.schedule()
    assert b1.is_prefill and b1.num_new_tokens == [20]
    # same prompt again: 4 full blocks (16 tokens) cached. s1 is decoding,
    # so this schedules as a MIXED batch: s1's next token + s2's new suffix.
    s2 = Sequence(prompt)
    s.add(s2)
    b2 = s.schedule()
    assert b2.is_prefill and b2.num_new_tokens == [1, 4]
    assert b2.seqs == [s1, s2]
    assert s2.num_cached_tokens == 16
    assert s2.block_table[:4] == s1.block_table[:4]  # shared pages
    assert s2.block_table[4] != s1.block_table[4]


def _engine(enable_cache: bool) -> LLMEngine:
    return LLMEngine(EngineConfig(
        preset="tiny", device="cpu", kv_cache_blocks=128, max_model_len=512,
        enable_prefix_caching=enable_cache, seed=7,
    ))


def test_engine_cached_outputs_match_uncached():
    prompts = [[5, 9, 2, 8] * 8, [5, 9, 2, 8] * 8, [3, 1, 4] * 6]
    sp = SamplingParams(max_tokens=6, ignore_eos=True)
    base = _engine(False).generate(prompts, sp)
    eng = _engine(True)
    # prime the cache, then repeat the same prompts
    first = eng.generate(prompts, sp)
    hits0, _ = eng.prefix_cache_stats
    second = eng.generate(prompts, sp)
    hits1, queries = eng.prefix_cache_stats
    assert base == first == second
    assert hits1 > hits0, "second round must hit the prefix cache"


def test_engine_cache_hit_after_preemption():
    eng = LLMEngine(EngineConfig(
        preset="tiny", device="cpu", kv_cache_blocks=12, max_model_len=512,
        enable_prefix_caching=True, seed=3,
    ))
    # two sequences that cannot decode together for long in 12 blocks
    prompts = [[1, 2, 3, 4] * 8, [9, 8, 7] * 10]
    out = eng.generate(prompts, SamplingParams(max_tokens=40, ignore_eos=True))
    assert all(len(o) == 40 for o in out)


def test_extend_ref_matches_prefill_ref_when_uncached():
    torch.manual_seed(0)
    Hq, Hkv, D, bs = 4, 2, 64, 16
    lens = [19, 33]
    T = sum(lens)
    q = torch.randn(T, Hq, D, dtype=torch.bfloat16)
    k = torch.randn(T, Hkv, D, dtype=torch.bfloat16)
    v = torch.randn(T, Hkv, D, dtype=torch.bfloat16)
    cu = torch.tensor([0, lens[0], T], dtype=torch.int32)
    ref_out = ref.attention_prefill_varlen(q, k, v, cu, scale=D ** -0.5)
    # pack the same K/V into pages
    nb = [(n + bs - 1) // bs for n in lens]
    k_cache = torch.zeros(sum(nb) + 1, Hkv, bs, D, dtype=torch.bfloat16)
    v_cache = torch.zeros_like(k_cache)
    bt = torch.zeros(2, max(nb), dtype=torch.int32)
    blk = 1
    for i, n in enumerate(lens):
        s = int(cu[i])
        for j in range(nb[i]):
            tok = min(bs, n - j * bs)
            k_cache[blk, :, :tok] = k[s + j * bs: s + j * bs + tok].transpose(0, 1)
            v_cache[blk, :, :tok] = v[s + j * bs: s + j * bs + tok].transpose(0, 1)
            bt[i, j] = blk
            blk += 1
    kv_lens = torch.tensor(lens, dtype=torch.int32)
    ext = ref.attention_extend_paged(q, k_cache, v_cache, bt, kv_lens, cu, D ** -0.5)
    assert torch.allclose(ref_out.float(), ext.float(), atol=2e-2, rtol=2e-2)


def test_extend_ref_cached_prefix_matches_full_prefill_suffix():
    torch.manual_seed(1)
    Hq, Hkv, D, bs = 4, 2, 64, 16
    L, cached = 45, 32
    k = torch.randn(L, Hkv, D, dtype=torch.bfloat16)
    v = torch.randn(L, Hkv, D, dtype=torch.bfloat16)
    qfull = torch.randn(L, Hq, D, dtype=torch.bfloat16)
    cu_full = torch.tensor([0, L], dtype=torch.int32)
    full = ref.attention_prefill_varlen(qfull, k, v, cu_full, scale=D ** -0.5)
    nb = (L + bs - 1) // bs
    k_cache = torch.zeros(nb, Hkv, bs, D, dtype=torch.bfloat16)
    v_cache = torch.zeros_like(k_cache)
    for j in range(nb):
        tok = min(bs, L - j * bs)
        k_cache[j, :, :tok] = k[j * bs: j * bs + tok].transpose(0, 1)
        v_cache[j, :, :tok] = v[j * bs: j * bs + tok].transpose(0, 1)
    bt = torch.arange(nb, dtype=torch.int32).unsqueeze(0)
    qn = qfull[cached:]
    cu_q = torch.tensor([0, L - cached], dtype=torch.int32)
    ext = ref.attention_extend_paged(
        qn, k_cache, v_cache, bt, torch.tensor([L], dtype=torch.int32), cu_q,
        D ** -0.5,
    )
    assert torch.allclose(full[cached:].float(), ext.float(), atol=2e-2, rtol=2e-2)


def test_chunked_prefill_matches_unchunked():
    """A prompt longer than max_num_batched_tokens prefills over several
    steps (extend path) and must produce identical greedy output."""
    prompt = [(7 * i + 3) % 250 for i in range(100)]
    sp = SamplingParams(max_tokens=6, ignore_eos=True)

    def run(budget, cache=True):
        eng = LLMEngine(EngineConfig(
            preset="tiny", device="cpu", kv_cache_blocks=128,
            max_model_len=256, max_num_batched_tokens=budget,
            enable_prefix_caching=cache, seed=13,
        ))
        out = eng.generate([prompt], sp)
        return out[0], eng

    full, _ = run(8192)
    chunked, eng = run(32)
    assert chunked == full
    # chunked + prefix-cache hit on a repeat
    again = eng.generate([prompt], sp)[0]
    assert again == full
    hits, _ = eng.prefix_cache_stats
    assert hits > 0
    # chunked with caching off
    nocache, _ = run(32, cache=False)
    assert nocache == full


def test_chunked_prefill_interleaves_multiple_seqs():
    prompts = [[1, 2, 3] * 20, [9, 8] * 25, [4] * 7]
    sp = SamplingParams(max_tokens=5, ignore_eos=True)

    def run(budget):
        eng = LLMEngine(EngineConfig(
            preset="tiny", device="cpu", kv_cache_blocks=128,
            max_model_len=256, max_num_batched_tokens=budget, seed=5,
        ))
        return eng.generate(prompts, sp)

    assert run(24) == run(8192)
