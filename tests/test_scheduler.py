"""CPU tests for the continuous-batching scheduler and block allocator."""

import pytest

from arks_amd.engine.kv_cache import BlockAllocator
from arks_amd.engine.scheduler import Scheduler
from arks_amd.engine.sequence import SamplingParams, Sequence, SeqStatus


def mk_sched(blocks=64, **kw):
    return Scheduler(BlockAllocator(blocks, 16), **kw)


def test_allocator_roundtrip():
    a = BlockAllocator(8, 16)
    b1 = a.allocate(3)
    assert a.num_free == 5
    a.free(b1)
    assert a.num_free == 8
    with pytest.raises(RuntimeError):
        a.allocate(9)


def test_prefill_batches_waiting():
    s = mk_sched(max_num_batched_tokens=100)
    s.add(Sequence([1] * 40))
    s.add(Sequence([1] * 40))
    s.add(Sequence([1] * 40))
    b = s.schedule()
    # chunked prefill: the 3rd seq rides along with the remaining 20-token
    # budget and stays WAITING
    assert b.is_prefill and len(b.seqs) == 3
    assert b.num_new_tokens == [40, 40, 20]
    assert [seq.status for seq in b.seqs] == [
        SeqStatus.RUNNING, SeqStatus.RUNNING, SeqStatus.WAITING]
    assert s.num_running == 2 and s.num_waiting == 1
    b.seqs[2].num_cached_tokens = 20  # (the runner does this after the step)
    b2 = s.schedule()
    # mixed batch: the two decoding seqs ride along with the chunk remainder
    assert b2.is_prefill and len(b2.seqs) == 3
    assert b2.num_new_tokens == [1, 1, 20]
    assert s.num_running == 3 and s.num_waiting == 0


def test_decode_after_prefill_grows_blocks():
    s = mk_sched()
    seq = Sequence([1] * 16)  # exactly one full block
    s.add(seq)
    s.schedule()
    assert len(seq.block_table) == 1
    seq.append_token(5)  # engine appends after the prefill step
    b = s.schedule()
    assert not b.is_prefill
    assert len(seq.block_table) == 2  # token 17 needs a second page


def test_preemption_on_kv_exhaustion():
    s = mk_sched(blocks=4)
    s1 = Sequence([1] * 30)  # 2 blocks
    s2 = Sequence([2] * 30)  # 2 blocks
    s.add(s1)
    s.add(s2)
    b = s.schedule()
    assert len(b.seqs) == 2
    # grow both past the boundary -> each needs a 3rd block, none free
    for seq in (s1, s2):
        for t in range(3):
            seq.append_token(0)
    b = s.schedule()
    assert s.num_preemptions == 1
    assert len(b.seqs) == 1 and b.seqs[0] is s1
    assert s2.status is SeqStatus.WAITING and s2.block_table == []
    # preempted seq recomputes prompt+generated on readmission
    assert s.waiting[0].num_tokens == 33


def test_abort():
    s = mk_sched()
    seq = Sequence([1] * 8)
    s.add(seq)
    assert s.abort(seq.request_id)
    assert not s.has_work()
    seq2 = Sequence([1] * 8)
    s.add(seq2)
    s.schedule()
    assert s.abort(seq2.request_id)
    assert s.allocator.num_free == s.allocator.num_blocks


def test_finished_frees_blocks():
    s = mk_sched()
    seq = Sequence([1] * 8, SamplingParams(max_tokens=1))
    s.add(seq)
    s.schedule()
    seq.append_token(3)
    assert seq.check_finished(eos_token_id=99)  # max_tokens reached
    s.free_finished()
    assert s.allocator.num_free == s.allocator.num_blocks
    assert not s.has_work()


def test_add_rejects_oversize():
    s = mk_sched(blocks=2, max_model_len=10_000)
    with pytest.raises(ValueError):
        s.add(Sequence([1] * 100))
