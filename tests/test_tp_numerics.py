"""Tensor-parallel NUMERICS: a 2-way sharded model (column/row-sharded
QKV/O/MLP + all-reduce over gloo) must produce the same logits as the
unsharded model loaded from the same HF checkpoint.

This is the correctness proof behind `worker run -tp N` — the TP worker
tests cover lockstep plumbing; this covers the sharded math itself
(weights.py shard slicing + the two per-layer all-reduces).
"""

from __future__ import annotations

import json
import os

import pytest
import torch

pytestmark = [pytest.mark.integration, pytest.mark.slow]

transformers = pytest.importorskip("transformers")


def _save_checkpoint(tmp_path):
    cfg = transformers.LlamaConfig(
        vocab_size=320, hidden_size=128, intermediate_size=256,
        num_hidden_layers=2, num_attention_heads=8, num_key_value_heads=4,
        max_position_embeddings=256, rope_theta=10000.0, rms_norm_eps=1e-6,
        tie_word_embeddings=False,
    )
    model = transformers.LlamaForCausalLM(cfg).eval().float()
    model.save_pretrained(tmp_path, safe_serialization=True)
    (tmp_path / "tokenizer_config.json").write_text(json.dumps({}))
    return model


def _forward_logits(engine, token_ids):
    from llmq_amd.engine.forward_meta import PrefillMeta

    dev = engine.device
    blocks = engine.allocator.allocate(4)
    bs = engine.config.kv_block_size
    slots = [blocks[p // bs] * bs + p % bs for p in range(len(token_ids))]
    meta = PrefillMeta(
        cu_seqlens=torch.tensor([0, len(token_ids)], dtype=torch.int32, device=dev),
        max_seqlen=len(token_ids),
        slot_mapping=torch.tensor(slots, dtype=torch.long, device=dev),
    )
    ids = torch.tensor(token_ids, dtype=torch.long, device=dev)
    pos = torch.arange(len(token_ids), dtype=torch.long, device=dev)
    hidden = engine.model.forward(ids, pos, engine.kv_cache, meta)
    return engine.model.compute_logits(hidden)


def _rank_main(rank, world, ckpt, out_file):
    import torch.distributed as dist

    from llmq_amd.engine.config import EngineConfig
    from llmq_amd.engine.engine import LLMEngine
    from llmq_amd.parallel import init_tp

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29591"
    init_tp(world, rank=rank, backend="gloo")
    engine = LLMEngine(EngineConfig(
        model=ckpt, device="cpu", enforce_eager=True,
        max_num_seqs=2, max_model_len=128, num_kv_blocks=64,
    ), tp_rank=rank, tp_size=world)
    token_ids = [1, 9, 77, 123, 200, 314, 5, 42]
    logits = _forward_logits(engine, token_ids)

    # end-to-end greedy decode in lockstep (sharded KV heads through the
    # paged decode path; every rank samples identically)
    from llmq_amd.engine.sampling_params import SamplingParams

    engine.add_request("g", prompt_token_ids=token_ids,
                       params=SamplingParams(temperature=0.0, max_tokens=6,
                                             ignore_eos=True))
    text_tokens = []
    while engine.has_unfinished():
        for out in engine.step():
            text_tokens.extend(out.new_token_ids)
    if rank == 0:
        torch.save({"logits": logits, "decode_tokens": text_tokens}, out_file)
    dist.barrier()
    dist.destroy_process_group()


def _rank_kv_sync(rank, world, ckpt, out_dir):
    import torch.distributed as dist

    from llmq_amd.engine.config import EngineConfig
    from llmq_amd.engine.engine import LLMEngine
    from llmq_amd.parallel import init_tp

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29593"
    init_tp(world, rank=rank, backend="gloo")
    # Asymmetric local sizing (standing in for rank-varying mem_get_info):
    # the engine must min-sync so every rank allocates identically.
    local_blocks = 64 if rank == 0 else 48
    engine = LLMEngine(EngineConfig(
        model=ckpt, device="cpu", enforce_eager=True,
        max_num_seqs=2, max_model_len=128, num_kv_blocks=local_blocks,
    ), tp_rank=rank, tp_size=world)
    (out_dir / f"rank{rank}.txt").write_text(str(engine.allocator.num_blocks))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_tp_kv_pool_min_synced_across_ranks(tmp_path):
    """Rank-varying KV budgets must converge to the min on every rank —
    otherwise lockstep replicas make divergent preemption decisions and
    silently emit different tokens (VERDICT r1 weakness 4)."""
    _save_checkpoint(tmp_path)

    import torch.multiprocessing as mp

    mp.spawn(_rank_kv_sync, args=(2, str(tmp_path), tmp_path), nprocs=2, join=True)
    n0 = int((tmp_path / "rank0.txt").read_text())
    n1 = int((tmp_path / "rank1.txt").read_text())
    assert n0 == n1 == 48, (n0, n1)


@pytest.mark.timeout(300)
def test_tp2_logits_match_unsharded(tmp_path):
    hf = _save_checkpoint(tmp_path)

    import torch.multiprocessing as mp

    out_file = str(tmp_path / "tp_logits.pt")
    mp.spawn(_rank_main, args=(2, str(tmp_path), out_file), nprocs=2, join=True)
    saved = torch.load(out_file)
    tp_logits = saved["logits"]

    token_ids = [1, 9, 77, 123, 200, 314, 5, 42]
    with torch.no_grad():
        ref = hf(torch.tensor([token_ids])).logits[0]

    diff = (tp_logits - ref).abs().max().item()
    scale = ref.abs().max().item()
    assert diff < 1e-3 * max(1.0, scale), f"TP logits diverge: {diff}"

    # decode tokens must match a single-process engine on the same checkpoint
    from llmq_amd.engine.config import EngineConfig
    from llmq_amd.engine.engine import LLMEngine
    from llmq_amd.engine.sampling_params import SamplingParams

    solo = LLMEngine(EngineConfig(
        model=str(tmp_path), device="cpu", enforce_eager=True,
        max_num_seqs=2, max_model_len=128, num_kv_blocks=64,
    ))
    solo.add_request("g", prompt_token_ids=token_ids,
                     params=SamplingParams(temperature=0.0, max_tokens=6,
                                           ignore_eos=True))
    solo_tokens = []
    while solo.has_unfinished():
        for out in solo.step():
            solo_tokens.extend(out.new_token_ids)
    assert saved["decode_tokens"] == solo_tokens, (
        saved["decode_tokens"], solo_tokens)


def _rank_preempt(rank, world, ckpt, out_dir):
    import torch.distributed as dist

    from llmq_amd.engine.config import EngineConfig
    from llmq_amd.engine.engine import LLMEngine
    from llmq_amd.engine.sampling_params import SamplingParams
    from llmq_amd.parallel import init_tp

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29597"
    init_tp(world, rank=rank, backend="gloo")
    engine = LLMEngine(EngineConfig(
        model=ckpt, device="cpu", enforce_eager=True,
        max_num_seqs=4, max_model_len=128,
        num_kv_blocks=13,  # tight: forces evict+recompute churn mid-decode
        max_prefill_tokens=64,
    ), tp_rank=rank, tp_size=world)
    params = SamplingParams(temperature=0.0, max_tokens=20, ignore_eos=True)
    for i in range(4):
        engine.add_request(f"r{i}", prompt_token_ids=list(range(2 + 5 * i, 40 + 5 * i)),
                           params=params)
    seqs = list(engine._seqs.values())
    toks = {f"r{i}": [] for i in range(4)}
    guard = 0
    while engine.has_unfinished() and guard < 400:
        for out in engine.step():
            toks[out.request_id].extend(out.new_token_ids)
        guard += 1
    preempts = max(s.num_preemptions for s in seqs)
    (out_dir / f"preempt_rank{rank}.json").write_text(
        json.dumps({"toks": toks, "preempts": preempts, "guard": guard}))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_tp2_preemption_lockstep(tmp_path):
    """Preemption-by-recompute under a tight KV pool must be IDENTICAL on
    every TP rank — divergent evictions would desynchronise the lockstep
    replicas (collective deadlock or silent token divergence on the 8-GPU
    tier). Both ranks run the same churn; tokens and preemption counts must
    match exactly, and preemption must actually fire."""
    _save_checkpoint(tmp_path)

    import torch.multiprocessing as mp

    mp.spawn(_rank_preempt, args=(2, str(tmp_path), tmp_path), nprocs=2, join=True)
    r0 = json.loads((tmp_path / "preempt_rank0.json").read_text())
    r1 = json.loads((tmp_path / "preempt_rank1.json").read_text())
    assert r0["guard"] < 400 and r1["guard"] < 400
    assert r0["preempts"] > 0, "pool never forced a preemption - vacuous"
    assert r0["preempts"] == r1["preempts"]
    assert r0["toks"] == r1["toks"], "TP ranks diverged under preemption churn"
    for rid, t in r0["toks"].items():
        assert len(t) == 20, (rid, t)
