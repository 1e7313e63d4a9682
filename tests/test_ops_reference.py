"""Internal-consistency tests of the fp32 reference ops (the ground truth the
HIP kernels are verified against in tests/test_ops_gpu.py)."""
import math

import torch

from bee2bee_amd.ops import reference as R


def test_rmsnorm_formula():
    x = torch.randn(4, 8)
    w = torch.randn(8)
    y = R.rmsnorm(x, w, 1e-5)
    for i in range(4):
        denom = math.sqrt(float((x[i] ** 2).mean()) + 1e-5)
        expected = x[i] / denom * w
        assert torch.allclose(y[i], expected, atol=1e-5)


def test_fused_add_rmsnorm():
    x, r = torch.randn(4, 8), torch.randn(4, 8)
    w = torch.ones(8)
    y, new_r = R.fused_add_rmsnorm(x, r, w, 1e-6)
    assert torch.allclose(new_r, x + r, atol=1e-6)
    assert torch.allclose(y, R.rmsnorm(x + r, w, 1e-6), atol=1e-6)


def test_rope_preserves_norm_and_rotates():
    T, nh, hd = 5, 2, 16
    q = torch.randn(T, nh, hd)
    k = torch.randn(T, 1, hd)
    q0, k0 = q.clone(), k.clone()
    cos, sin = R.rope_tables(64, hd, 10000.0, "cpu")
    pos = torch.arange(T, dtype=torch.int32)
    R.rope_inplace(q, k, pos, cos, sin)
    # rotation preserves per-pair norms
    def pair_norms(t):
        return t[..., : hd // 2] ** 2 + t[..., hd // 2 :] ** 2

    assert torch.allclose(pair_norms(q), pair_norms(q0), atol=1e-4)
    # position 0 is identity
    assert torch.allclose(q[0], q0[0], atol=1e-6)
    assert not torch.allclose(q[1], q0[1])
    # relative property: dot(q_rot(m), k_rot(n)) depends only on m-n
    qq = torch.randn(hd)
    kk = torch.randn(hd)

    def rot(v, p):
        t = v.clone().view(1, 1, hd)
        R.rope_inplace(t, t.clone(), torch.tensor([p], dtype=torch.int32), cos, sin)
        return t.view(hd)

    d1 = torch.dot(rot(qq, 3), rot(kk, 1))
    d2 = torch.dot(rot(qq, 10), rot(kk, 8))
    assert torch.allclose(d1, d2, atol=1e-3)


def test_kv_store_and_decode_matches_prefill():
    """decode(last token) over the paged cache == prefill's last row."""
    torch.manual_seed(0)
    nq, nkv, hd, bs = 4, 2, 16, 4
    L = 10
    q = torch.randn(L, nq, hd)
    k = torch.randn(L, nkv, hd)
    v = torch.randn(L, nkv, hd)
    scale = hd**-0.5
    cu = torch.tensor([0, L], dtype=torch.int32)
    full = R.attn_prefill(q, k, v, cu, L, scale, causal=True)

    n_blocks = 8
    k_cache = torch.zeros(n_blocks, nkv, bs, hd)
    v_cache = torch.zeros(n_blocks, nkv, bs, hd)
    # store sequence into blocks [2, 0, 5]
    blocks = [2, 0, 5]
    slots = torch.tensor(
        [blocks[p // bs] * bs + p % bs for p in range(L)], dtype=torch.int32
    )
    R.kv_cache_store(k, v, k_cache, v_cache, slots)
    block_table = torch.tensor([blocks], dtype=torch.int32)
    seq_lens = torch.tensor([L], dtype=torch.int32)
    out = R.attn_decode(q[-1:].clone(), k_cache, v_cache, block_table, seq_lens, scale)
    assert torch.allclose(out[0], full[-1], atol=1e-4)


def test_chunked_prefill_matches_full():
    torch.manual_seed(1)
    nq, nkv, hd, bs = 4, 2, 16, 4
    L, chunk = 12, 5  # history 7, new chunk 5
    q = torch.randn(L, nq, hd)
    k = torch.randn(L, nkv, hd)
    v = torch.randn(L, nkv, hd)
    scale = hd**-0.5
    cu = torch.tensor([0, L], dtype=torch.int32)
    full = R.attn_prefill(q, k, v, cu, L, scale, causal=True)

    k_cache = torch.zeros(8, nkv, bs, hd)
    v_cache = torch.zeros(8, nkv, bs, hd)
    blocks = [1, 3, 0]
    slots = torch.tensor(
        [blocks[p // bs] * bs + p % bs for p in range(L)], dtype=torch.int32
    )
    R.kv_cache_store(k, v, k_cache, v_cache, slots)
    out = R.attn_decode_with_history(
        q[-chunk:].clone(),
        k_cache,
        v_cache,
        torch.tensor([blocks], dtype=torch.int32),
        torch.tensor([L], dtype=torch.int32),
        torch.tensor([chunk], dtype=torch.int32),
        scale,
    )
    assert torch.allclose(out, full[-chunk:], atol=1e-4)


def test_swiglu():
    x = torch.randn(3, 8)
    y = R.swiglu(x)
    g, u = x.chunk(2, -1)
    assert torch.allclose(y, torch.nn.functional.silu(g) * u, atol=1e-6)


def test_varlen_prefill_batches_independent():
    torch.manual_seed(2)
    nq, nkv, hd = 2, 1, 8
    l1, l2 = 4, 6
    q = torch.randn(l1 + l2, nq, hd)
    k = torch.randn(l1 + l2, nkv, hd)
    v = torch.randn(l1 + l2, nkv, hd)
    cu = torch.tensor([0, l1, l1 + l2], dtype=torch.int32)
    out = R.attn_prefill(q, k, v, cu, max(l1, l2), hd**-0.5)
    # each sequence standalone must match
    out1 = R.attn_prefill(
        q[:l1], k[:l1], v[:l1], torch.tensor([0, l1], dtype=torch.int32), l1, hd**-0.5
    )
    assert torch.allclose(out[:l1], out1, atol=1e-5)


def test_moe_topk_gate():
    logits = torch.tensor([[1.0, 5.0, 2.0, 4.0]])
    w, idx = R.moe_topk_gate(logits, 2)
    assert idx[0].tolist() == [1, 3]
    assert abs(float(w.sum()) - 1.0) < 1e-5
