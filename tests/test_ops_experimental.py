"""EXPERIMENTAL kernels — excluded from the standard gpu suite (separate
marker); run explicitly with `-m gpu_experimental` on a GPU box to
validate before promoting into the dispatch policy."""

import pytest
import torch

from quoracle_amd.ops import reference
from test_ops_gpu import _build_paged_cache, _ops, dev  # noqa: F401

pytestmark = pytest.mark.gpu_experimental


def test_paged_attn_decode_split2_matches_reference(dev):
    ops = _ops()
    torch.manual_seed(20)
    B, Hq, Hkv, D, BS = 3, 8, 2, 128, 16
    lens = [1500, 3000, 137]
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
    ops.ext().paged_attn_decode_split2(out, q, kcache, vcache, tables, ctx,
                                       scale, part_m, part_l, part_acc)
    from quoracle_amd.ops import reference
    for s in range(B):
        ref = reference.attention(q[s:s + 1], seqs[s][0], seqs[s][1], scale)
        assert torch.allclose(out[s].float(), ref[0], atol=4e-2, rtol=4e-2), \
            f"seq {s}: max err {(out[s].float() - ref[0]).abs().max().item()}"


def test_paged_attn_prefill_t12_split_matches_reference(dev):
    """T12-split (wave-local 16-key chunks, in-register softmax, 16-key
    PV MFMA, per-wave partials) vs the fp32 reference."""
    ops = _ops()
    torch.manual_seed(27)
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
    NS8 = 6 * 8
    pm = torch.empty((ntiles, Hq, NS8, 16), dtype=torch.float32, device=dev)
    pl = torch.empty_like(pm)
    pa = torch.empty((ntiles, Hq, NS8, 16, D), dtype=torch.float32,
                     device=dev)
    ops.ext().paged_attn_prefill_t12_split(out, q, kcache, vcache, tables,
                                           t0, qn, tseq, tpos, scale,
                                           pm, pl, pa)
    ref = reference.attention(q, k, v, scale, causal_offset=cached)
    assert torch.allclose(out.float(), ref, atol=4e-2, rtol=4e-2), \
        f"max err {(out.float() - ref).abs().max().item()}"


def test_paged_attn_prefill_t12w_matches_reference(dev):
    """T12W (32x32x16 MFMA, 4-swap permlane A-fragments) vs the fp32
    reference, incl. an odd tail."""
    ops = _ops()
    torch.manual_seed(31)
    Hq, Hkv, D, BS = 32, 8, 128, 16
    scale = D ** -0.5
    cached, new = 777, 300
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
    ops.ext().paged_attn_prefill_t12w(out, q, kcache, vcache, tables,
                                      t0, qn, tseq, tpos, scale)
    ref = reference.attention(q, k, v, scale, causal_offset=cached)
    assert torch.allclose(out.float(), ref, atol=4e-2, rtol=4e-2), \
        f"max err {(out.float() - ref).abs().max().item()}"
