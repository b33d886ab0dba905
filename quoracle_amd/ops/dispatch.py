"""Device dispatch for the compute ops.

GPU tensors go to the HIP extension and FAIL LOUDLY if it is missing — a
silent eager fallback can never masquerade as the native path on a GPU box.
CPU tensors go to the fp32 reference implementations so the whole engine
(model forward, KV pager, batcher) is testable without a GPU.
"""

from __future__ import annotations

from typing import Optional

import os

import torch

from . import ext
from . import reference as ref


def rmsnorm(y: torch.Tensor, x: torch.Tensor,
            residual: Optional[torch.Tensor], w: torch.Tensor,
            eps: float = 1e-5) -> torch.Tensor:
    """y = rmsnorm(x (+ residual)) * w; residual updated in place on GPU."""
    if x.is_cuda:
        ext().rmsnorm_fused(y, x, residual, w, eps)
        return y
    out, new_res = ref.rmsnorm(x, w, residual=residual, eps=eps)
    y.copy_(out.to(y.dtype))
    if residual is not None:
        residual.copy_(new_res.to(residual.dtype))
    return y


def swiglu(out: torch.Tensor, gate_up: torch.Tensor) -> torch.Tensor:
    if gate_up.is_cuda:
        ext().swiglu(out, gate_up)
        return out
    out.copy_(ref.swiglu(gate_up).to(out.dtype))
    return out


def rope_inplace(q: torch.Tensor, k: torch.Tensor, pos: torch.Tensor,
                 theta: float) -> None:
    if q.is_cuda:
        ext().rope_inplace(q, k, pos, theta)
        return
    q.copy_(ref.rope(q, pos, theta=theta).to(q.dtype))
    k.copy_(ref.rope(k, pos, theta=theta).to(k.dtype))


def kv_append(kcache, vcache, k, v, slots) -> None:
    if kcache.is_cuda:
        ext().kv_append(kcache, vcache, k, v, slots)
        return
    ref.kv_append(kcache, vcache, k, v, slots)


def paged_attn_decode(out, q, kcache, vcache, block_tables, ctx_lens,
                      scale: float, max_ctx: int = 0):
    """GQA decode attention.  On GPU, long contexts take the flash-decoding
    split path (context partitioned over NS workgroups + combine) so small
    agent batches still fill the 256 CUs; short contexts use the single-pass
    kernel (no workspace traffic)."""
    if q.is_cuda:
        B, Hq, D = q.shape
        if max_ctx >= 1024:
            ns = min(32, max(2, max_ctx // 256))
            part_m = torch.empty((B, Hq, ns), dtype=torch.float32,
                                 device=q.device)
            part_l = torch.empty_like(part_m)
            part_acc = torch.empty((B, Hq, ns, D), dtype=torch.float32,
                                   device=q.device)
            if D == 128 and not os.environ.get("QUORACLE_DECODE_VALU"):
                # matrix-core flash-decode (validated r2: 0.78 -> 4.3 TB/s at 7k ctx, 6.1 at 32k)
                ext().paged_attn_decode_mfma(out, q, kcache, vcache,
                                             block_tables, ctx_lens, scale,
                                             part_m, part_l, part_acc)
            else:
                ext().paged_attn_decode_split(out, q, kcache, vcache,
                                              block_tables, ctx_lens, scale,
                                              part_m, part_l, part_acc)
        else:
            ext().paged_attn_decode(out, q, kcache, vcache, block_tables,
                                    ctx_lens, scale)
        return out
    return ref.paged_attn_decode(out, q, kcache, vcache, block_tables,
                                 ctx_lens, scale)


def paged_attn_prefill(out, q, kcache, vcache, block_tables, tile_q0,
                       tile_qn, tile_seq, tile_pos0, scale: float,
                       max_kv: int = 0):
    """Chunked causal prefill over the paged cache.  Few tiles over a long
    cached context (grammar forced-run chunks, incremental history prefill)
    take the context-split path so the walk parallelizes across the chip."""
    if q.is_cuda:
        ntiles = tile_q0.shape[0]
        Hq = q.shape[1]
        D = q.shape[2]
        # D=128 prefill goes to the matrix cores; small grids additionally
        # context-split so the KV walk parallelizes across the chip
        ns = max(1, min(48, 4096 // max(1, ntiles * Hq)))
        if D == 128 and not os.environ.get("QUORACLE_NO_MFMA_ATTN"):
            if ns > 1 and max_kv >= 1024:
                QT = 16
                part_m = torch.empty((ntiles, Hq, ns, QT),
                                     dtype=torch.float32, device=q.device)
                part_l = torch.empty_like(part_m)
                part_acc = torch.empty((ntiles, Hq, ns, QT, D),
                                       dtype=torch.float32, device=q.device)
                ext().paged_attn_prefill_mfma_split(
                    out, q, kcache, vcache, block_tables, tile_q0, tile_qn,
                    tile_seq, tile_pos0, scale, part_m, part_l, part_acc)
                return out
            ext().paged_attn_prefill_mfma(out, q, kcache, vcache,
                                          block_tables, tile_q0, tile_qn,
                                          tile_seq, tile_pos0, scale)
            return out
        if ns > 1 and max_kv >= 1024:
            QT = 16
            part_m = torch.empty((ntiles, Hq, ns, QT), dtype=torch.float32,
                                 device=q.device)
            part_l = torch.empty_like(part_m)
            part_acc = torch.empty((ntiles, Hq, ns, QT, D),
                                   dtype=torch.float32, device=q.device)
            ext().paged_attn_prefill_split(out, q, kcache, vcache,
                                           block_tables, tile_q0, tile_qn,
                                           tile_seq, tile_pos0, scale,
                                           part_m, part_l, part_acc)
            return out
        ext().paged_attn_prefill(out, q, kcache, vcache, block_tables,
                                 tile_q0, tile_qn, tile_seq, tile_pos0, scale)
        return out
    return ref.paged_attn_prefill(out, q, kcache, vcache, block_tables,
                                  tile_q0, tile_qn, tile_seq, tile_pos0,
                                  scale)


def gather_rows(out, src, rows):
    if src.is_cuda:
        ext().gather_rows(out, src, rows)
        return out
    return ref.gather_rows(out, src, rows)


def cosine_sim_matrix(out, x):
    if x.is_cuda:
        ext().cosine_sim_matrix(out, x)
        return out
    out.copy_(ref.cosine_sim_matrix(x))
    return out
