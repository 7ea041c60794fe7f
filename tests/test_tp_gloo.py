"""Tensor-parallel correctness over gloo (world_size=2, CPU):
the TP=2 SPMD engine must produce exactly the TP=1 greedy tokens."""

import multiprocessing as mp
import os
import socket

import pytest
import torch

PROMPTS = [[1, 5, 9, 20, 31, 7], [3, 3, 7, 90]]
MAX_TOKENS = 6


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _tp1_reference(preset="tiny", quantization=None):
    from arks_amd.config import EngineConfig
    from arks_amd.engine import LLMEngine, SamplingParams

    torch.manual_seed(0)
    e = LLMEngine(
        EngineConfig(preset=preset, device="cpu", kv_cache_blocks=128,
                     max_model_len=512, quantization=quantization)
    )
    return e.generate(PROMPTS, SamplingParams(max_tokens=MAX_TOKENS, ignore_eos=True))


def _tp_worker(rank: int, world: int, port: int, q, preset: str = "tiny",
               quantization=None, speculative=None, draft_model=None):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    try:
        from arks_amd.config import EngineConfig
        from arks_amd.engine import LLMEngine, SamplingParams
        from arks_amd.parallel import comm

        comm.init_tp(backend="gloo")
        torch.manual_seed(0)
        e = LLMEngine(
            EngineConfig(
                preset=preset, device="cpu", kv_cache_blocks=128,
                max_model_len=512, quantization=quantization,
                speculative=speculative, draft_model=draft_model,
            )
        )
        out = e.generate(PROMPTS, SamplingParams(max_tokens=MAX_TOKENS, ignore_eos=True))
        if rank == 0:
            q.put(("ok", out))
        comm.destroy_tp()
    except Exception as e:  # pragma: no cover
        import traceback

        q.put(("err", f"{e}\n{traceback.format_exc()}"))


@pytest.mark.timeout(180)
def test_tp2_matches_tp1_gloo():
    ref = _tp1_reference()

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = [ctx.Process(target=_tp_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    status, payload = q.get(timeout=150)
    for p in procs:
        p.join(timeout=60)
    assert status == "ok", payload
    assert payload == ref, f"TP=2 output {payload} != TP=1 {ref}"


@pytest.mark.timeout(180)
def test_tp2_sharded_weights_random_init_consistent():
    """random_init must give identical full tensors regardless of TP degree:
    verified indirectly by the generation equality above; here check the
    sharding math: column+row shard shapes on the tiny config."""
    from arks_amd.config import PRESET_CONFIGS
    from arks_amd.models import create_model

    cfg = PRESET_CONFIGS["tiny"]
    m = create_model(cfg)
    assert m.layers[0].self_attn.qkv_proj.weight.shape == (
        (cfg.num_attention_heads + 2 * cfg.num_key_value_heads) * cfg.head_dim,
        cfg.hidden_size,
    )
    assert m.layers[0].mlp.gate_up_proj.weight.shape == (
        2 * cfg.intermediate_size,
        cfg.hidden_size,
    )


@pytest.mark.timeout(240)
def test_tp2_moe_expert_parallel_matches_tp1():
    """Sparse-MoE expert parallelism: TP=2 shards whole experts per rank and
    all-reduces partial outputs; greedy tokens must equal TP=1."""
    ref = _tp1_reference("tiny-moe")

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = [
        ctx.Process(target=_tp_worker, args=(r, 2, port, q, "tiny-moe"))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    status, payload = q.get(timeout=200)
    for p in procs:
        p.join(timeout=60)
    assert status == "ok", payload
    assert payload == ref, f"TP=2 MoE output {payload} != TP=1 {ref}"


def _disagg_tp_worker(rank: int, world: int, port: int, q):
    """TP=2 disaggregated-prefill round trip at the engine level: prefill
    with held pages, extract (head all-gather) -> full-head KV, re-inject as
    a new request (each rank slices its heads), decode, and compare the
    continuation with a plain generate of the same prompt."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    try:
        from arks_amd.config import EngineConfig, PRESET_CONFIGS
        from arks_amd.engine import LLMEngine, SamplingParams
        from arks_amd.parallel import comm

        comm.init_tp(backend="gloo")
        torch.manual_seed(0)
        e = LLMEngine(EngineConfig(preset="tiny", device="cpu",
                                   kv_cache_blocks=128, max_model_len=512))
        prompt = [1, 5, 9, 20, 31, 7, 2, 8]
        e.add_request(prompt, SamplingParams(max_tokens=1, ignore_eos=True),
                      request_id="d0", hold_pages=True)
        outs = []
        while e.has_work():
            outs += e.step()
        first = next(o.new_token_id for o in outs if o.request_id == "d0")
        _, kv = e.extract_prefilled("d0")  # collective: both ranks in lockstep
        nkv_full = PRESET_CONFIGS["tiny"].num_key_value_heads
        assert kv.shape[3] == nkv_full, (kv.shape, nkv_full)
        e.add_prefilled(prompt, first, kv,
                        SamplingParams(max_tokens=5, ignore_eos=True), "d1")
        toks = [first]
        while e.has_work():
            for o in e.step():
                if o.request_id == "d1":
                    toks.append(o.new_token_id)
        ref = e.generate([prompt], SamplingParams(max_tokens=6, ignore_eos=True))[0]
        if rank == 0:
            q.put(("ok", (toks, ref)))
        comm.destroy_tp()
    except Exception as exc:  # pragma: no cover
        import traceback

        q.put(("err", f"{exc}\n{traceback.format_exc()}"))


@pytest.mark.timeout(240)
def test_tp2_disagg_kv_round_trip():
    """TP>1 disaggregation: extract gathers the full head set, inject
    re-shards it, and the decoded continuation matches a plain generate."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = [
        ctx.Process(target=_disagg_tp_worker, args=(r, 2, port, q))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    status, payload = q.get(timeout=200)
    for p in procs:
        p.join(timeout=60)
    assert status == "ok", payload
    toks, ref = payload
    # injected run: first token + 4 decode steps (max_tokens=5 counts the
    # remotely sampled first token); plain run generates 6.
    assert len(toks) == 5 and toks == ref[:5], (
        f"injected continuation {toks} != plain generate {ref}"
    )


@pytest.mark.timeout(240)
def test_tp2_fp8_matches_tp1_fp8():
    """fp8 W8A8 with TP sharding: per-shard weight scales + bf16 all-reduce
    must reproduce the TP=1 fp8 greedy tokens."""
    ref = _tp1_reference("tiny", quantization="fp8")

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = [
        ctx.Process(target=_tp_worker, args=(r, 2, port, q, "tiny", "fp8"))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    status, payload = q.get(timeout=200)
    for p in procs:
        p.join(timeout=60)
    assert status == "ok", payload
    assert payload == ref, f"TP=2 fp8 output {payload} != TP=1 {ref}"


def _async_disagg_worker(rank: int, world: int, port: int, q):
    """TP=2 disaggregation through the REAL serving stack: rank 0 drives an
    AsyncEngine (broadcast protocol), rank 1 runs worker_loop — covering the
    extract/inject worker messages end to end."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    try:
        import asyncio

        from arks_amd.config import EngineConfig
        from arks_amd.engine import SamplingParams
        from arks_amd.parallel import comm
        from arks_amd.server.async_engine import AsyncEngine, worker_loop

        comm.init_tp(backend="gloo")
        torch.manual_seed(0)
        cfg = EngineConfig(preset="tiny", device="cpu", kv_cache_blocks=128,
                           max_model_len=512)
        if rank != 0:
            worker_loop(cfg)  # returns on the driver's stop broadcast
            comm.destroy_tp()
            return

        async def go():
            eng = AsyncEngine(cfg, model_name="tiny")
            await eng.start()
            prompt = [4, 9, 2, 7, 7, 1]
            sp = SamplingParams(max_tokens=5, ignore_eos=True)
            # plain generate through the broadcast protocol
            plain = []
            async for out in eng.generate_stream("g0", list(prompt), sp):
                plain.append(out.new_token_id)
            # disaggregated: prefill+extract, then inject+decode
            first, reason, kv = await eng.disagg_prefill("p0", list(prompt), sp)
            # the inject stream queues the first token itself
            st = await eng.disagg_inject("d0", list(prompt), first, kv, sp)
            toks = []
            while True:
                out = await st.queue.get()
                if out is None:
                    break
                toks.append(out.new_token_id)
            await eng.stop()
            return plain, toks

        plain, toks = asyncio.new_event_loop().run_until_complete(go())
        q.put(("ok", (plain, toks)))
        comm.destroy_tp()
    except Exception as exc:  # pragma: no cover
        import traceback

        q.put(("err", f"{exc}\n{traceback.format_exc()}"))


@pytest.mark.timeout(240)
def test_tp2_async_engine_disagg_worker_protocol():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = [
        ctx.Process(target=_async_disagg_worker, args=(r, 2, port, q))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    status, payload = q.get(timeout=200)
    for p in procs:
        p.join(timeout=60)
    assert status == "ok", payload
    plain, toks = payload
    assert len(plain) == 5
    assert toks == plain, f"disagg {toks} != plain {plain}"


@pytest.mark.timeout(240)
def test_tp2_speculative_matches_tp1_plain():
    """Speculative decoding under TP=2 SPMD: every rank proposes/accepts
    identically (drafts are a pure function of the shared token history),
    so the output equals the plain TP=1 engine on the CPU fp32 path."""
    ref = _tp1_reference()

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = [
        ctx.Process(target=_tp_worker, args=(r, 2, port, q, "tiny", None, "ngram"))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    status, payload = q.get(timeout=200)
    for p in procs:
        p.join(timeout=60)
    assert status == "ok", payload
    assert payload == ref, f"TP=2 spec output {payload} != TP=1 {ref}"


@pytest.mark.timeout(180)
def test_tp2_draft_speculation_matches_tp1():
    """Draft-model speculation under TP: the sharded draft forward runs
    SPMD on every rank (its all-reduces keep lockstep), proposals are
    rank-identical, outputs match the plain TP=1 engine."""
    ref = _tp1_reference()

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = [ctx.Process(target=_tp_worker,
                         args=(r, 2, port, q, "tiny", None, "draft",
                               "preset:tiny"))
             for r in range(2)]
    for p in procs:
        p.start()
    status, payload = q.get(timeout=150)
    for p in procs:
        p.join(timeout=60)
    assert status == "ok", payload
    assert payload == ref, f"TP=2 draft-spec {payload} != TP=1 {ref}"
