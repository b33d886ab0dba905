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
