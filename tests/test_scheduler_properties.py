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


def test_oversized_prompt_rejected_loudly():
    """A prompt that could never be seated (needs more blocks than the whole
    pool) must be rejected at add time, not left to livelock the admission
    loop (head-of-line blocking with empty steps forever)."""
    eng = LLMEngine(EngineConfig(
        model="tiny-llama", device="cpu", load_weights=False,
        max_num_seqs=2, max_model_len=128, num_kv_blocks=4,  # 64 slots
        max_prefill_tokens=40, kv_block_size=16,
    ))
    params = SamplingParams(temperature=0.0, max_tokens=4, ignore_eos=True)
    with pytest.raises(ValueError, match="never"):
        eng.add_request("big", prompt_token_ids=list(range(100)), params=params)
    # a fitting request still works end-to-end
    eng.add_request("ok", prompt_token_ids=list(range(30)), params=params)
    guard = 0
    while eng.has_unfinished() and guard < 50:
        eng.step()
        guard += 1
    assert guard < 50


def test_preemption_recompute_matches_unpreempted():
    """Preemption-by-recompute must be output-invisible: a pool too small
    for the batch forces evict+re-prefill churn, and the greedy tokens must
    equal a run with a roomy pool (fp32 CPU: bitwise-deterministic)."""
    params = SamplingParams(temperature=0.0, max_tokens=24, ignore_eos=True)
    prompts = [list(range(1 + 7 * i, 40 + 5 * i)) for i in range(4)]

    def run(num_blocks):
        eng = LLMEngine(EngineConfig(
            model="tiny-llama", device="cpu", load_weights=False,
            max_num_seqs=4, max_model_len=128, num_kv_blocks=num_blocks,
            max_prefill_tokens=64, kv_block_size=16,
        ))
        for i, p in enumerate(prompts):
            eng.add_request(f"r{i}", prompt_token_ids=p, params=params)
        seqs = list(eng._seqs.values())  # keep refs past finish-time pop
        toks = {f"r{i}": [] for i in range(4)}
        guard = 0
        while eng.has_unfinished() and guard < 400:
            for out in eng.step():
                toks[out.request_id].extend(out.new_token_ids)
            guard += 1
        assert guard < 400, "engine failed to drain"
        return toks, max(s.num_preemptions for s in seqs)

    roomy, p0 = run(64)
    tight, p1 = run(13)  # 4 seqs want ~19 blocks -> forced evict+recompute
    assert p0 == 0, "roomy run unexpectedly preempted"
    assert p1 > 0, "tight run never preempted - test is vacuous, shrink the pool"
    for rid in roomy:
        assert len(roomy[rid]) == 24
        assert roomy[rid] == tight[rid], f"{rid}: preemption changed tokens"
