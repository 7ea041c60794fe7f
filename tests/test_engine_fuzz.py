"""Property-based stress: random add/step/abort sequences against the CPU
engine; after every step the prefix-caching allocator's books must balance
(every reference accounted for by a live block-table slot, free sets
disjoint and complete) and the engine must terminate."""

import pytest
from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

from arks_amd.config import EngineConfig
from arks_amd.engine import LLMEngine, SamplingParams
from arks_amd.engine.kv_cache import PrefixCachingAllocator


def audit(engine: LLMEngine) -> None:
    alloc = engine.scheduler.allocator
    assert isinstance(alloc, PrefixCachingAllocator)
    holders: dict[int, int] = {}
    sched = engine.scheduler
    live = list(sched.running) + list(sched.waiting) + list(sched.held.values())
    for seq in live:
        for b in seq.block_table:
            holders[b] = holders.get(b, 0) + 1
    for b in range(alloc.num_blocks):
        assert alloc._ref[b] == holders.get(b, 0), (
            f"block {b}: ref {alloc._ref[b]} vs holders {holders.get(b, 0)}"
        )
    virgin = set(alloc._virgin)
    lru = set(alloc._lru.keys())
    held = {b for b, r in enumerate(alloc._ref) if r > 0}
    assert not virgin & lru and not virgin & held and not lru & held
    assert len(virgin) + len(lru) + len(held) == alloc.num_blocks
    # every cached digest points at a block that actually carries it
    for d, b in alloc._cached.items():
        assert alloc._digest[b] == d


@settings(max_examples=15, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(
    actions=st.lists(
        st.one_of(
            st.tuples(st.just("add"), st.integers(1, 60), st.integers(1, 8),
                      st.integers(0, 3)),
            st.tuples(st.just("step"), st.just(0), st.just(0), st.just(0)),
            st.tuples(st.just("abort"), st.integers(0, 30), st.just(0),
                      st.just(0)),
        ),
        min_size=4, max_size=30,
    ),
)
@pytest.mark.parametrize("speculative", [None, "ngram", "draft"])
def test_engine_invariants_under_random_ops(actions, speculative):
    eng = LLMEngine(EngineConfig(
        preset="tiny", device="cpu", kv_cache_blocks=24, max_model_len=96,
        max_num_batched_tokens=48, max_num_seqs=8, seed=5,
        speculative=speculative,
        draft_model="preset:tiny" if speculative == "draft" else None,
    ))
    rid = 0
    for kind, a, b, c in actions:
        if kind == "add":
            # prompt built from a small token alphabet so prefix hits occur
            prompt = [(a * 7 + i * (c + 1)) % 90 for i in range(min(a, 90))]
            eng.add_request(prompt, SamplingParams(max_tokens=b, ignore_eos=True),
                            request_id=f"f{rid}")
            rid += 1
        elif kind == "step":
            eng.step()
        else:
            eng.abort_request(f"f{a}")
        audit(eng)
    # drain to completion; books still balance, all pages return
    guard = 0
    while eng.has_work():
        eng.step()
        audit(eng)
        guard += 1
        assert guard < 2000, "engine failed to terminate"
    alloc = eng.scheduler.allocator
    assert alloc.num_free == alloc.num_blocks


@settings(max_examples=12, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(
    actions=st.lists(
        st.one_of(
            st.tuples(st.just("add"), st.integers(8, 120), st.integers(1, 24)),
            st.just(("step",)),
            st.integers(0, 6).map(lambda i: ("abort", i)),
        ),
        min_size=4, max_size=24,
    ),
)
def test_sliding_window_dropping_never_changes_outputs(actions):
    """Fuzz: under random add/step/abort sequences on an SWA model, KV
    page dropping must not change any emitted token (dropped pages are
    outside every query's window by construction) and the allocator's
    free count must never go negative."""
    import dataclasses

    import arks_amd.config as C

    swa = dataclasses.replace(C.PRESET_CONFIGS["tiny"], sliding_window=32)
    C.PRESET_CONFIGS["tiny-fuzz-swa"] = swa
    try:
        def run(drop: bool):
            e = LLMEngine(EngineConfig(
                preset="tiny-fuzz-swa", device="cpu", kv_cache_blocks=256,
                max_model_len=256, seed=7,
            ))
            if not drop:
                e._drop_window_pages = lambda: None
            outs = {}
            rid = 0
            live = []
            for act in actions:
                if act[0] == "add":
                    _, ln, maxt = act
                    prompt = [(i * 17 + rid) % 90 for i in range(ln)]
                    s = e.add_request(prompt, SamplingParams(
                        max_tokens=maxt, ignore_eos=True),
                        request_id=f"r{rid}")
                    live.append(s)
                    rid += 1
                elif act[0] == "step":
                    e.step()
                else:
                    i = act[1]
                    if i < len(live):
                        e.abort_request(live[i].request_id)
            while e.has_work():
                e.step()
            assert e.scheduler.allocator.num_free >= 0
            for s in live:
                outs[s.request_id] = list(s.output_token_ids)
            return outs

        assert run(True) == run(False)
    finally:
        C.PRESET_CONFIGS.pop("tiny-fuzz-swa", None)
