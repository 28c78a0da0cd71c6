"""Guard-band (canary) pass over the HIP kernels.

No compute-sanitizer ships in this image, so out-of-bounds WRITES are
checked directly: every input/output tensor is carved out of an
over-allocated buffer with canary halos on both sides, and the halos are
verified bitwise after each op. Run under ``AMD_SERIALIZE_KERNEL=3`` (the
CI recipe — tests/sanitize_gpu.sh) so any kernel fault aborts at the
offending launch instead of surfacing later. Canaries do not catch OOB
READS; those surface as faults under serialized launches or as numerics
failures in test_gpu_kernels.
"""

from __future__ import annotations

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from llmq_amd import ops
else:
    pytest.skip("no GPU", allow_module_level=True)

DEV = torch.device("cuda:0")
PAD = 4096  # canary elements on each side
CANARY = 0x5A


class Guarded:
    """Allocate tensors inside canary halos; verify() checks every halo."""

    def __init__(self):
        self._bufs = []

    def tensor(self, *shape, dtype=torch.bfloat16, fill="randn"):
        n = 1
        for s in shape:
            n *= s
        esz = torch.tensor([], dtype=dtype).element_size()
        raw = torch.full(((n + 2 * PAD) * esz,), CANARY, dtype=torch.uint8, device=DEV)
        mid = raw[PAD * esz: (PAD + n) * esz].view(dtype).view(*shape)
        if fill == "randn":
            mid.copy_(torch.randn(shape, device=DEV).to(dtype))
        elif fill == "zeros":
            mid.zero_()
        self._bufs.append((raw, esz, n))
        return mid

    def from_values(self, vals, dtype):
        t = self.tensor(*vals.shape, dtype=dtype, fill="zeros")
        t.copy_(vals.to(dtype))
        return t

    def verify(self):
        torch.cuda.synchronize()
        for i, (raw, esz, n) in enumerate(self._bufs):
            lo = raw[: PAD * esz]
            hi = raw[(PAD + n) * esz:]
            assert bool((lo == CANARY).all()), f"buffer {i}: LOW canary clobbered"
            assert bool((hi == CANARY).all()), f"buffer {i}: HIGH canary clobbered"


@pytest.fixture()
def g():
    gg = Guarded()
    yield gg
    gg.verify()


def test_decode_attention_guarded(g, monkeypatch):
    B, G, KVH, D, BS = 7, 2, 8, 256, 16
    H = G * KVH
    max_blocks = 70
    NB = B * max_blocks + 1
    kc = g.tensor(NB, KVH, BS, D)
    vc = g.tensor(NB, KVH, BS, D)
    q = g.tensor(B, H, D)
    bt = g.from_values(
        torch.arange(1, 1 + B * max_blocks).reshape(B, max_blocks), torch.int32)
    ctx = g.from_values(torch.tensor([1, 63, 64, 65, 129, 1000, 1025]), torch.int32)
    for pipe in ("0", "64", "128"):
        monkeypatch.setenv("LLMQ_DECODE_PIPE", pipe)
        out = ops.paged_decode_attention(q, kc, vc, bt, ctx, D ** -0.5, 0.0, 0)
        assert out.isfinite().all()
    # split-KV path (scratch + merge kernels)
    out = ops.paged_decode_attention(q[:2], kc, vc, bt[:2], ctx[:2], D ** -0.5, 0.0, 0)
    assert out.isfinite().all()


def test_prefill_attention_guarded(g, monkeypatch):
    B, H, KVH, D, L = 3, 16, 8, 256, 300
    T = B * L
    q = g.tensor(T, H, D)
    k = g.tensor(T, KVH, D)
    v = g.tensor(T, KVH, D)
    cu = g.from_values(torch.arange(0, T + 1, L), torch.int32)
    for pipe in ("0", "1"):
        monkeypatch.setenv("LLMQ_PREFILL_PIPE", pipe)
        out = ops.varlen_prefill_attention(q, k, v, cu, L, D ** -0.5)
        assert out.isfinite().all()


def test_elementwise_and_cache_guarded(g):
    rows, hidden = 33, 3584
    x = g.tensor(rows, hidden)
    w = g.tensor(hidden)
    res = g.tensor(rows, hidden)
    ops.rmsnorm(x, w, 1e-6, 0.0)
    ops.fused_add_rmsnorm(x, res, w, 1e-6)
    act = g.tensor(rows, 2 * hidden)
    ops.silu_and_mul(act)
    ops.gelu_tanh_and_mul(act)
    # rope + cache write
    T, HQ, HK, D2, BS, NB2 = 29, 16, 8, 256, 16, 40
    from llmq_amd.ops import torch_ref

    cs = torch_ref.build_rope_cache(128, D2, 10000.0, DEV)
    qkv_q = g.tensor(T, HQ, D2)
    qkv_k = g.tensor(T, HK, D2)
    vval = g.tensor(T, HK, D2)
    kc = g.tensor(NB2, HK, BS, D2, fill="zeros")
    vc = g.tensor(NB2, HK, BS, D2, fill="zeros")
    pos = g.from_values(torch.arange(T), torch.int64)
    slots = g.from_values(torch.randperm(NB2 * BS)[:T], torch.int64)
    ops.rope_and_cache(qkv_q, qkv_k, vval, kc, vc, pos, cs, slots)


def test_sampler_guarded(g):
    B, V = 64, 50257
    logits = g.tensor(B, V, dtype=torch.float32)
    temps = g.tensor(B, dtype=torch.float32, fill="zeros")
    temps.fill_(0.8)
    tps = g.tensor(B, dtype=torch.float32, fill="zeros")
    tps.fill_(0.9)
    tks = g.from_values(torch.full((B,), 50), torch.int64)
    out = g.from_values(torch.zeros(B), torch.int64)
    keys = g.from_values(torch.zeros(B), torch.int64)
    zs = g.from_values(torch.zeros(B), torch.int32)
    bound = ops.topk_topp_bound(logits, temps, tps, tks)
    ops.sample_gumbel_argmax(out, keys, logits, temps, zs, zs, 3, 1, bound)
    assert bool((out >= 0).all()) and bool((out < V).all())
