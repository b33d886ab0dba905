"""HIP kernel numerics vs plain fp32 PyTorch references (runs on MI355X)."""

import pytest
import torch

from quoracle_amd.ops import reference

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dev():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    return torch.device("cuda:0")


def _ops():
    from quoracle_amd import ops
    # fail loudly: on a GPU box the extension must be present
    ops.ext()
    return ops


def test_rmsnorm_fused_matches_reference(dev):
    ops = _ops()
    torch.manual_seed(0)
    rows, n = 33, 4096
    x = torch.randn(rows, n, device=dev, dtype=torch.bfloat16)
    res = torch.randn(rows, n, device=dev, dtype=torch.bfloat16)
    w = torch.randn(n, device=dev, dtype=torch.bfloat16)
    y = torch.empty_like(x)
    res_copy = res.clone()
    ops.rmsnorm_fused(y, x, res_copy, w, 1e-5)
    ref_y, ref_res = reference.rmsnorm(x, w, residual=res, eps=1e-5)
    assert torch.allclose(y.float(), ref_y, atol=5e-2, rtol=5e-2)
    assert torch.allclose(res_copy.float(), ref_res, atol=2e-2, rtol=2e-2)


def test_rmsnorm_no_residual(dev):
    ops = _ops()
    torch.manual_seed(1)
    x = torch.randn(5, 1024, device=dev, dtype=torch.bfloat16)
    w = torch.ones(1024, device=dev, dtype=torch.bfloat16)
    y = torch.empty_like(x)
    ops.rmsnorm_fused(y, x, None, w, 1e-5)
    ref_y, _ = reference.rmsnorm(x, w, eps=1e-5)
    assert torch.allclose(y.float(), ref_y, atol=5e-2, rtol=5e-2)


def test_swiglu_matches_reference(dev):
    ops = _ops()
    torch.manual_seed(2)
    gu = torch.randn(17, 2 * 1408, device=dev, dtype=torch.bfloat16)
    out = torch.empty(17, 1408, device=dev, dtype=torch.bfloat16)
    ops.swiglu(out, gu)
    ref = reference.swiglu(gu)
    assert torch.allclose(out.float(), ref, atol=5e-2, rtol=5e-2)


def test_rope_matches_reference(dev):
    ops = _ops()
    torch.manual_seed(3)
    T, Hq, Hk, D = 9, 8, 2, 128
    q = torch.randn(T, Hq, D, device=dev, dtype=torch.bfloat16)
    k = torch.randn(T, Hk, D, device=dev, dtype=torch.bfloat16)
    pos = torch.randint(0, 5000, (T,), device=dev, dtype=torch.int32)
    ref_q = reference.rope(q, pos, theta=500000.0)
    ref_k = reference.rope(k, pos, theta=500000.0)
    ops.rope_inplace(q, k, pos, 500000.0)
    assert torch.allclose(q.float(), ref_q, atol=3e-2, rtol=3e-2)
    assert torch.allclose(k.float(), ref_k, atol=3e-2, rtol=3e-2)


def _build_paged_cache(dev, seqs, Hkv, D, BS=16):
    """seqs: list of [T, Hkv, D] (k, v) tuples -> cache tensors + tables."""
    total_blocks = sum((kv[0].shape[0] + BS - 1) // BS for kv in seqs) + 1
    kcache = torch.zeros(total_blocks, Hkv, BS, D, device=dev,
                         dtype=torch.bfloat16)
    vcache = torch.zeros_like(kcache)
    maxb = max((kv[0].shape[0] + BS - 1) // BS for kv in seqs)
    tables = torch.zeros(len(seqs), maxb, device=dev, dtype=torch.int32)
    next_block = 1  # block 0 unused to catch indexing bugs
    ctx = []
    for s, (k, v) in enumerate(seqs):
        T = k.shape[0]
        ctx.append(T)
        nb = (T + BS - 1) // BS
        for b in range(nb):
            tables[s, b] = next_block
            lo, hi = b * BS, min((b + 1) * BS, T)
            kcache[next_block, :, : hi - lo] = k[lo:hi].transpose(0, 1)
            vcache[next_block, :, : hi - lo] = v[lo:hi].transpose(0, 1)
            next_block += 1
    ctx_t = torch.tensor(ctx, device=dev, dtype=torch.int32)
    return kcache, vcache, tables, ctx_t


def test_paged_attn_decode_matches_reference(dev):
    ops = _ops()
    torch.manual_seed(4)
    B, Hq, Hkv, D, BS = 3, 8, 2, 128, 16
    lens = [37, 128, 200]
    scale = D ** -0.5
    seqs = [(torch.randn(t, Hkv, D, device=dev, dtype=torch.bfloat16),
             torch.randn(t, Hkv, D, device=dev, dtype=torch.bfloat16))
            for t in lens]
    kcache, vcache, tables, ctx = _build_paged_cache(dev, seqs, Hkv, D, BS)
    q = torch.randn(B, Hq, D, device=dev, dtype=torch.bfloat16)
    out = torch.empty_like(q)
    ops.paged_attn_decode(out, q, kcache, vcache, tables, ctx, scale)
    for s in range(B):
        ref = reference.attention(q[s:s + 1], seqs[s][0], seqs[s][1], scale)
        assert torch.allclose(out[s].float(), ref[0], atol=4e-2, rtol=4e-2), \
            f"seq {s} mismatch: max err " \
            f"{(out[s].float() - ref[0]).abs().max().item()}"


def test_paged_attn_prefill_matches_reference(dev):
    ops = _ops()
    torch.manual_seed(5)
    Hq, Hkv, D, BS = 8, 2, 128, 16
    scale = D ** -0.5
    # two sequences: one pure prefill (no cached prefix), one chunked suffix
    # over a cached prefix of 50 tokens
    specs = [(0, 40), (50, 23)]  # (cached_prefix, new_tokens)
    seqs = []
    for cached, new in specs:
        total = cached + new
        seqs.append((torch.randn(total, Hkv, D, device=dev, dtype=torch.bfloat16),
                     torch.randn(total, Hkv, D, device=dev, dtype=torch.bfloat16)))
    kcache, vcache, tables, _ = _build_paged_cache(dev, seqs, Hkv, D, BS)

    q_rows = []
    tile_q0, tile_qn, tile_seq, tile_pos0 = [], [], [], []
    row = 0
    QT = 16
    q_all = []
    for s, (cached, new) in enumerate(specs):
        q = torch.randn(new, Hq, D, device=dev, dtype=torch.bfloat16)
        q_all.append(q)
        for t0 in range(0, new, QT):
            qn = min(QT, new - t0)
            tile_q0.append(row + t0)
            tile_qn.append(qn)
            tile_seq.append(s)
            tile_pos0.append(cached + t0)
        row += new
    q = torch.cat(q_all, dim=0)
    out = torch.empty_like(q)
    to_t = lambda lst: torch.tensor(lst, device=dev, dtype=torch.int32)
    ops.paged_attn_prefill(out, q, kcache, vcache, tables, to_t(tile_q0),
                           to_t(tile_qn), to_t(tile_seq), to_t(tile_pos0),
                           scale)
    row = 0
    for s, (cached, new) in enumerate(specs):
        k, v = seqs[s]
        ref = reference.attention(q[row:row + new], k, v, scale,
                                  causal_offset=cached)
        got = out[row:row + new].float()
        assert torch.allclose(got, ref, atol=4e-2, rtol=4e-2), \
            f"seq {s}: max err {(got - ref).abs().max().item()}"
        row += new


def test_kv_append_roundtrip(dev):
    ops = _ops()
    torch.manual_seed(6)
    T, Hkv, D, BS = 21, 2, 128, 16
    nblocks = 4
    kcache = torch.zeros(nblocks, Hkv, BS, D, device=dev, dtype=torch.bfloat16)
    vcache = torch.zeros_like(kcache)
    k = torch.randn(T, Hkv, D, device=dev, dtype=torch.bfloat16)
    v = torch.randn(T, Hkv, D, device=dev, dtype=torch.bfloat16)
    # tokens 0..20 of a sequence whose blocks are [2, 0, 3]
    blocks = [2, 0, 3]
    slots = torch.tensor([blocks[t // BS] * BS + t % BS for t in range(T)],
                         device=dev, dtype=torch.int32)
    ops.kv_append(kcache, vcache, k, v, slots)
    for t in range(T):
        blk, off = blocks[t // BS], t % BS
        assert torch.equal(kcache[blk, :, off], k[t])
        assert torch.equal(vcache[blk, :, off], v[t])


def test_cosine_sim_matrix_matches_reference(dev):
    ops = _ops()
    torch.manual_seed(7)
    x = torch.randn(6, 384, device=dev, dtype=torch.float32)
    x[3] = 0.0  # zero vector -> 0 similarity
    out = torch.empty(6, 6, device=dev, dtype=torch.float32)
    ops.cosine_sim_matrix(out, x)
    ref = reference.cosine_sim_matrix(x)
    assert torch.allclose(out, ref, atol=1e-5, rtol=1e-5)


def test_gather_rows(dev):
    ops = _ops()
    src = torch.randn(100, 256, device=dev, dtype=torch.bfloat16)
    rows = torch.tensor([3, 99, 0, 42], device=dev, dtype=torch.int32)
    out = torch.empty(4, 256, device=dev, dtype=torch.bfloat16)
    ops.gather_rows(out, src, rows)
    assert torch.equal(out, src[rows.long()])


def test_paged_attn_decode_split_matches_reference(dev):
    """Flash-decoding split path (long contexts) vs the fp32 reference."""
    ops = _ops()
    torch.manual_seed(14)
    B, Hq, Hkv, D, BS = 2, 8, 2, 128, 16
    lens = [1500, 3000]
    NS = 16
    scale = D ** -0.5
    seqs = [(torch.randn(t, Hkv, D, device=dev, dtype=torch.bfloat16),
             torch.randn(t, Hkv, D, device=dev, dtype=torch.bfloat16))
            for t in lens]
    kcache, vcache, tables, ctx = _build_paged_cache(dev, seqs, Hkv, D, BS)
    q = torch.randn(B, Hq, D, device=dev, dtype=torch.bfloat16)
    out = torch.empty_like(q)
    part_m = torch.empty((B, Hq, NS), dtype=torch.float32, device=dev)
    part_l = torch.empty_like(part_m)
    part_acc = torch.empty((B, Hq, NS, D), dtype=torch.float32, device=dev)
    ops.ext().paged_attn_decode_split(out, q, kcache, vcache, tables, ctx,
                                      scale, part_m, part_l, part_acc)
    for s in range(B):
        ref = reference.attention(q[s:s + 1], seqs[s][0], seqs[s][1], scale)
        assert torch.allclose(out[s].float(), ref[0], atol=4e-2, rtol=4e-2), \
            f"seq {s}: max err {(out[s].float() - ref[0]).abs().max().item()}"
    # single-pass kernel agrees with the split path
    out2 = torch.empty_like(q)
    ops.paged_attn_decode(out2, q, kcache, vcache, tables, ctx, scale)
    assert torch.allclose(out.float(), out2.float(), atol=3e-2, rtol=3e-2)


def test_paged_attn_prefill_split_matches_reference(dev):
    """Context-split prefill (small chunk over long cached context)."""
    ops = _ops()
    torch.manual_seed(15)
    Hq, Hkv, D, BS = 8, 2, 128, 16
    scale = D ** -0.5
    cached, new = 2000, 23
    total = cached + new
    k = torch.randn(total, Hkv, D, device=dev, dtype=torch.bfloat16)
    v = torch.randn(total, Hkv, D, device=dev, dtype=torch.bfloat16)
    kcache, vcache, tables, ctx = _build_paged_cache(dev, [(k, v)], Hkv, D, BS)
    q = torch.randn(new, Hq, D, device=dev, dtype=torch.bfloat16)
    out = torch.empty_like(q)
    ntiles = (new + 15) // 16
    t0 = torch.arange(ntiles, dtype=torch.int32, device=dev) * 16
    qn = torch.clamp(torch.full_like(t0, new) - t0, max=16)
    tseq = torch.zeros_like(t0)
    tpos = t0 + cached
    NS = 8
    part_m = torch.empty((ntiles, Hq, NS, 16), dtype=torch.float32, device=dev)
    part_l = torch.empty_like(part_m)
    part_acc = torch.empty((ntiles, Hq, NS, 16, D), dtype=torch.float32,
                           device=dev)
    ops.ext().paged_attn_prefill_split(out, q, kcache, vcache, tables,
                                       t0, qn, tseq, tpos, scale,
                                       part_m, part_l, part_acc)
    ref = reference.attention(q, k, v, scale, causal_offset=cached)
    assert torch.allclose(out.float(), ref, atol=4e-2, rtol=4e-2), \
        f"max err {(out.float() - ref).abs().max().item()}"


def test_paged_attn_prefill_mfma_matches_reference(dev):
    """MFMA-tiled prefill (matrix cores) vs fp32 reference, incl. tails."""
    ops = _ops()
    torch.manual_seed(16)
    Hq, Hkv, D, BS = 32, 8, 128, 16
    scale = D ** -0.5
    cached, new = 777, 100          # non-aligned: exercises masks and tails
    total = cached + new
    k = torch.randn(total, Hkv, D, device=dev, dtype=torch.bfloat16)
    v = torch.randn(total, Hkv, D, device=dev, dtype=torch.bfloat16)
    kcache, vcache, tables, ctx = _build_paged_cache(dev, [(k, v)], Hkv, D, BS)
    q = torch.randn(new, Hq, D, device=dev, dtype=torch.bfloat16)
    out = torch.empty_like(q)
    ntiles = (new + 15) // 16
    t0 = torch.arange(ntiles, dtype=torch.int32, device=dev) * 16
    qn = torch.clamp(torch.full_like(t0, new) - t0, max=16)
    tseq = torch.zeros_like(t0)
    tpos = t0 + cached
    ops.ext().paged_attn_prefill_mfma(out, q, kcache, vcache, tables,
                                      t0, qn, tseq, tpos, scale)
    ref = reference.attention(q, k, v, scale, causal_offset=cached)
    err = (out.float() - ref).abs().max().item()
    assert torch.allclose(out.float(), ref, atol=4e-2, rtol=4e-2), \
        f"max err {err}"


def test_paged_attn_prefill_mfma_split_matches_reference(dev):
    """Context-split MFMA prefill (small chunk over long cached context)."""
    ops = _ops()
    torch.manual_seed(17)
    Hq, Hkv, D, BS = 32, 8, 128, 16
    scale = D ** -0.5
    cached, new = 2000, 23
    total = cached + new
    k = torch.randn(total, Hkv, D, device=dev, dtype=torch.bfloat16)
    v = torch.randn(total, Hkv, D, device=dev, dtype=torch.bfloat16)
    kcache, vcache, tables, ctx = _build_paged_cache(dev, [(k, v)], Hkv, D, BS)
    q = torch.randn(new, Hq, D, device=dev, dtype=torch.bfloat16)
    out = torch.empty_like(q)
    ntiles = (new + 15) // 16
    t0 = torch.arange(ntiles, dtype=torch.int32, device=dev) * 16
    qn = torch.clamp(torch.full_like(t0, new) - t0, max=16)
    tseq = torch.zeros_like(t0)
    tpos = t0 + cached
    NS = 8
    part_m = torch.empty((ntiles, Hq, NS, 16), dtype=torch.float32, device=dev)
    part_l = torch.empty_like(part_m)
    part_acc = torch.empty((ntiles, Hq, NS, 16, D), dtype=torch.float32,
                           device=dev)
    ops.ext().paged_attn_prefill_mfma_split(out, q, kcache, vcache, tables,
                                            t0, qn, tseq, tpos, scale,
                                            part_m, part_l, part_acc)
    ref = reference.attention(q, k, v, scale, causal_offset=cached)
    assert torch.allclose(out.float(), ref, atol=4e-2, rtol=4e-2), \
        f"max err {(out.float() - ref).abs().max().item()}"


def test_paged_attn_prefill_mfma32_matches_reference(dev):
    """8-wave 32-row-tile MFMA prefill vs fp32 reference (odd tails)."""
    ops = _ops()
    torch.manual_seed(18)
    Hq, Hkv, D, BS = 32, 8, 128, 16
    scale = D ** -0.5
    cached, new = 777, 100
    total = cached + new
    k = torch.randn(total, Hkv, D, device=dev, dtype=torch.bfloat16)
    v = torch.randn(total, Hkv, D, device=dev, dtype=torch.bfloat16)
    kcache, vcache, tables, ctx = _build_paged_cache(dev, [(k, v)], Hkv, D, BS)
    q = torch.randn(new, Hq, D, device=dev, dtype=torch.bfloat16)
    out = torch.empty_like(q)
    nt = (new + 31) // 32
    t0 = torch.arange(nt, dtype=torch.int32, device=dev) * 32
    qn = torch.clamp(torch.full_like(t0, new) - t0, max=32)
    tseq = torch.zeros_like(t0)
    tpos = t0 + cached
    ops.ext().paged_attn_prefill_mfma32(out, q, kcache, vcache, tables,
                                        t0, qn, tseq, tpos, scale)
    ref = reference.attention(q, k, v, scale, causal_offset=cached)
    assert torch.allclose(out.float(), ref, atol=4e-2, rtol=4e-2), \
        f"max err {(out.float() - ref).abs().max().item()}"


def test_paged_attn_decode_mfma_matches_reference(dev):
    """Matrix-core flash-decode (GQ heads padded into a 16-row MFMA tile)
    vs the fp32 reference, including a context that is not a multiple of
    the 64-key chunk or the split width."""
    ops = _ops()
    torch.manual_seed(21)
    B, Hq, Hkv, D, BS = 3, 32, 8, 128, 16
    lens = [1500, 3000, 137]
    NS = 12
    scale = D ** -0.5
    seqs = [(torch.randn(t, Hkv, D, device=dev, dtype=torch.bfloat16),
             torch.randn(t, Hkv, D, device=dev, dtype=torch.bfloat16))
            for t in lens]
    kcache, vcache, tables, ctx = _build_paged_cache(dev, seqs, Hkv, D, BS)
    q = torch.randn(B, Hq, D, device=dev, dtype=torch.bfloat16)
    out = torch.empty_like(q)
    part_m = torch.empty((B, Hq, NS), dtype=torch.float32, device=dev)
    part_l = torch.empty_like(part_m)
    part_acc = torch.empty((B, Hq, NS, D), dtype=torch.float32, device=dev)
    ops.ext().paged_attn_decode_mfma(out, q, kcache, vcache, tables, ctx,
                                     scale, part_m, part_l, part_acc)
    for s in range(B):
        ref = reference.attention(q[s:s + 1], seqs[s][0], seqs[s][1], scale)
        assert torch.allclose(out[s].float(), ref[0], atol=4e-2, rtol=4e-2), \
            f"seq {s}: max err {(out[s].float() - ref[0]).abs().max().item()}"


def test_paged_attn_decode_mfma_gq8(dev):
    """GQ=8 (llama-70b shape: 64 query heads over 8 KV heads)."""
    ops = _ops()
    torch.manual_seed(22)
    B, Hq, Hkv, D, BS = 2, 64, 8, 128, 16
    lens = [2048, 700]
    NS = 8
    scale = D ** -0.5
    seqs = [(torch.randn(t, Hkv, D, device=dev, dtype=torch.bfloat16),
             torch.randn(t, Hkv, D, device=dev, dtype=torch.bfloat16))
            for t in lens]
    kcache, vcache, tables, ctx = _build_paged_cache(dev, seqs, Hkv, D, BS)
    q = torch.randn(B, Hq, D, device=dev, dtype=torch.bfloat16)
    out = torch.empty_like(q)
    part_m = torch.empty((B, Hq, NS), dtype=torch.float32, device=dev)
    part_l = torch.empty_like(part_m)
    part_acc = torch.empty((B, Hq, NS, D), dtype=torch.float32, device=dev)
    ops.ext().paged_attn_decode_mfma(out, q, kcache, vcache, tables, ctx,
                                     scale, part_m, part_l, part_acc)
    for s in range(B):
        ref = reference.attention(q[s:s + 1], seqs[s][0], seqs[s][1], scale)
        assert torch.allclose(out[s].float(), ref[0], atol=4e-2, rtol=4e-2), \
            f"seq {s}: max err {(out[s].float() - ref[0]).abs().max().item()}"


def test_paged_attn_prefill_mfma64_matches_reference(dev):
    ops = _ops()
    torch.manual_seed(19)
    Hq, Hkv, D, BS = 32, 8, 128, 16
    scale = D ** -0.5
    cached, new = 777, 200          # several 64-row tiles + odd tail
    total = cached + new
    k = torch.randn(total, Hkv, D, device=dev, dtype=torch.bfloat16)
    v = torch.randn(total, Hkv, D, device=dev, dtype=torch.bfloat16)
    kcache, vcache, tables, ctx = _build_paged_cache(dev, [(k, v)], Hkv, D, BS)
    q = torch.randn(new, Hq, D, device=dev, dtype=torch.bfloat16)
    out = torch.empty_like(q)
    nt = (new + 63) // 64
    t0 = torch.arange(nt, dtype=torch.int32, device=dev) * 64
    qn = torch.clamp(torch.full_like(t0, new) - t0, max=64)
    tseq = torch.zeros_like(t0)
    tpos = t0 + cached
    ops.ext().paged_attn_prefill_mfma64(out, q, kcache, vcache, tables,
                                        t0, qn, tseq, tpos, scale)
    ref = reference.attention(q, k, v, scale, causal_offset=cached)
    assert torch.allclose(out.float(), ref, atol=4e-2, rtol=4e-2), \
        f"max err {(out.float() - ref).abs().max().item()}"


def test_paged_attn_prefill_t12_matches_reference(dev):
    """T12 128-row prefill (swapped QK^T, in-register softmax, permlane
    P exchange) vs the fp32 reference, incl. odd tails."""
    ops = _ops()
    torch.manual_seed(23)
    Hq, Hkv, D, BS = 32, 8, 128, 16
    scale = D ** -0.5
    cached, new = 777, 300          # 3 tiles of 128 with a 44-row tail
    total = cached + new
    k = torch.randn(total, Hkv, D, device=dev, dtype=torch.bfloat16)
    v = torch.randn(total, Hkv, D, device=dev, dtype=torch.bfloat16)
    kcache, vcache, tables, ctx = _build_paged_cache(dev, [(k, v)], Hkv, D, BS)
    q = torch.randn(new, Hq, D, device=dev, dtype=torch.bfloat16)
    out = torch.empty_like(q)
    nt = (new + 127) // 128
    t0 = torch.arange(nt, dtype=torch.int32, device=dev) * 128
    qn = torch.clamp(torch.full_like(t0, new) - t0, max=128)
    tseq = torch.zeros_like(t0)
    tpos = t0 + cached
    ops.ext().paged_attn_prefill_t12(out, q, kcache, vcache, tables,
                                     t0, qn, tseq, tpos, scale)
    ref = reference.attention(q, k, v, scale, causal_offset=cached)
    assert torch.allclose(out.float(), ref, atol=4e-2, rtol=4e-2), \
        f"max err {(out.float() - ref).abs().max().item()}"
