"""Property-based scheduler/allocator invariants (hypothesis).

Random interleavings of add/step/abort over a tiny engine must never leak
KV blocks, double-allocate, or corrupt chunked-prefill bookkeeping — the
failure modes of continuous batching under churn (SURVEY §7 risk list:
"scheduler correctness under churn").
"""

from __future__ import annotations

import pytest
from hypothesis import given, settings
from hypothesis import strategies as st

from llmq_amd.engine.config import EngineConfig
from llmq_amd.engine.engine import LLMEngine
from llmq_amd.engine.sampling_params import SamplingParams

pytestmark = pytest.mark.integration


def make_engine() -> LLMEngine:
    return LLMEngine(EngineConfig(
        model="tiny-llama", device="cpu", load_weights=False,
        max_num_seqs=4, max_model_len=128, num_kv_blocks=48,
        max_prefill_tokens=40, kv_block_size=16,
    ))


def check_invariants(eng: LLMEngine) -> None:
    sched = eng.scheduler
    alloc = eng.allocator
    live = list(sched.running) + list(sched.prefilling) + list(sched.waiting)
    held = []
    for seq in live:
        held.extend(seq.block_table)
    # no double allocation
    assert len(held) == len(set(held)), "block allocated twice"
    # conservation: free + held == total
    assert alloc.num_free + len(held) == alloc.num_blocks, (
        f"leak: free={alloc.num_free} held={len(held)} total={alloc.num_blocks}"
    )
    # seat cap: running + mid-prefill sequences can never exceed
    # max_num_seqs (admission must count STRANDED prefilling seqs too —
    # otherwise oversized decode batches fall off the hipGraph path)
    assert len(sched.running) + len(sched.prefilling) <= sched.max_num_seqs, (
        f"seats over cap: running={len(sched.running)} "
        f"prefilling={len(sched.prefilling)} cap={sched.max_num_seqs}"
    )
    # chunk bookkeeping
    for seq in sched.prefilling:
        assert 0 < seq.prefilled < seq.num_tokens
        assert seq not in sched.running
    for seq in sched.waiting:
        assert not seq.block_table
    # held blocks cover every prefilled token
    bs = sched.block_size
    for seq in list(sched.running) + list(sched.prefilling):
        assert len(seq.block_table) * bs >= seq.prefilled


def test_sequence_uid_unique_across_request_id_reuse():
    """request_ids may be reused across sequence lifetimes; the uid used to
    key per-sequence caches (hipGraph block-table staging rows) must not be.
    """
    from llmq_amd.engine.scheduler import Sequence

    params = SamplingParams(temperature=0.0, max_tokens=4)
    a = Sequence("batch-0", [1, 2, 3], params)
    b = Sequence("batch-0", [4, 5, 6], params)
    assert a.request_id == b.request_id
    assert a.uid != b.uid
    # the staging key distinguishes same-id same-length different-life rows
    assert (a.uid, a.num_preemptions, 3) != (b.uid, b.num_preemptions, 3)


@settings(max_examples=20, deadline=None)
@given(st.lists(
    st.one_of(
        st.tuples(st.just("add"), st.integers(1, 90)),    # prompt length
        st.tuples(st.just("step"), st.just(0)),
        st.tuples(st.just("abort"), st.integers(0, 30)),  # request index
    ),
    min_size=5, max_size=40,
))
def test_no_block_leaks_under_churn(ops):
    eng = make_engine()
    params = SamplingParams(temperature=0.0, max_tokens=4, ignore_eos=True)
    n = 0
    for op, arg in ops:
        if op == "add":
            eng.add_request(f"r{n}", prompt_token_ids=list(range(arg)), params=params)
            n += 1
        elif op == "step":
            eng.step()
        else:
            eng.abort_request(f"r{arg % max(n, 1)}")
        check_invariants(eng)
    # drain to completion: everything must finish and free its blocks
    guard = 0
    while eng.has_unfinished() and guard < 500:
        eng.step()
        check_invariants(eng)
        guard += 1
    assert guard < 500, "engine failed to drain"
    assert eng.allocator.num_free == eng.allocator.num_blocks
