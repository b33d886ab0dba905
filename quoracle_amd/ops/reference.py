"""Plain-PyTorch fp32 reference implementations of the HIP ops.

These are the numerics baselines the GPU tests compare the HIP kernels
against (tests/test_ops_gpu.py), and the compute path of the CPU reference
engine used by orchestrator tests.  Everything here is deliberately naive and
readable — correctness over speed.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch


def rmsnorm(x: torch.Tensor, w: torch.Tensor,
            residual: Optional[torch.Tensor] = None,
            eps: float = 1e-5) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
    """Returns (y, updated_residual)."""
    xf = x.float()
    if residual is not None:
        xf = xf + residual.float()
        residual = xf
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    y = xf * torch.rsqrt(var + eps) * w.float()
    return y, residual


def swiglu(gate_up: torch.Tensor) -> torch.Tensor:
    inter = gate_up.shape[-1] // 2
    gate = gate_up[..., :inter].float()
    up = gate_up[..., inter:].float()
    return torch.nn.functional.silu(gate) * up


def rope(x: torch.Tensor, pos: torch.Tensor,
         theta: float = 500000.0) -> torch.Tensor:
    """Rotate-half RoPE. x: [T, H, D]; pos: [T]."""
    T, H, D = x.shape
    half = D // 2
    xf = x.float()
    freqs = theta ** (-2.0 * torch.arange(half, dtype=torch.float32,
                                          device=x.device) / D)
    angles = pos.float()[:, None] * freqs[None, :]        # [T, half]
    cos = angles.cos()[:, None, :]
    sin = angles.sin()[:, None, :]
    x0, x1 = xf[..., :half], xf[..., half:]
    return torch.cat([x0 * cos - x1 * sin, x0 * sin + x1 * cos], dim=-1)


def attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
              scale: float, causal_offset: Optional[int] = None) -> torch.Tensor:
    """q: [Tq, Hq, D]; k/v: [Tk, Hkv, D] (full context, oldest first).

    causal_offset: absolute position of q row 0; q row i attends to
    positions <= causal_offset + i.  None = attend to everything (decode
    with Tq=1 uses causal_offset = Tk-1 equivalently).
    """
    Tq, Hq, D = q.shape
    Tk, Hkv, _ = k.shape
    gq = Hq // Hkv
    qf = q.float()
    kf = k.float().repeat_interleave(gq, dim=1)   # [Tk, Hq, D]
    vf = v.float().repeat_interleave(gq, dim=1)
    scores = torch.einsum("qhd,khd->hqk", qf, kf) * scale
    if causal_offset is not None:
        kv_pos = torch.arange(Tk, device=q.device)[None, :]
        q_pos = causal_offset + torch.arange(Tq, device=q.device)[:, None]
        scores = scores.masked_fill((kv_pos > q_pos)[None], float("-inf"))
    probs = torch.softmax(scores, dim=-1)
    return torch.einsum("hqk,khd->qhd", probs, vf)


def cosine_sim_matrix(x: torch.Tensor) -> torch.Tensor:
    xf = x.float()
    norms = xf.norm(dim=-1, keepdim=True)
    safe = torch.where(norms > 0, norms, torch.ones_like(norms))
    xn = xf / safe
    sim = xn @ xn.T
    zero = (norms.squeeze(-1) == 0)
    sim[zero, :] = 0.0
    sim[:, zero] = 0.0
    return sim


# -- paged-KV reference implementations (same layouts as the HIP kernels) ----
# kcache/vcache: [NB, Hkv, BS, D]; block_tables: [B, MAXB] int32;
# slots: [T] int32 global slot = block*BS + offset.

def kv_append(kcache: torch.Tensor, vcache: torch.Tensor,
              k: torch.Tensor, v: torch.Tensor, slots: torch.Tensor) -> None:
    NB, Hkv, BS, D = kcache.shape
    for t in range(slots.shape[0]):
        s = int(slots[t])
        if s < 0:
            continue
        blk, off = s // BS, s % BS
        kcache[blk, :, off, :] = k[t]
        vcache[blk, :, off, :] = v[t]


def _gather_kv(cache: torch.Tensor, block_table: torch.Tensor,
               length: int) -> torch.Tensor:
    """[len, Hkv, D] rows 0..len-1 of one sequence from the paged cache."""
    NB, Hkv, BS, D = cache.shape
    toks = torch.arange(length, device=cache.device)
    blocks = block_table[toks // BS].long()
    offs = toks % BS
    return cache[blocks, :, offs, :]          # [len, Hkv, D]


def paged_attn_decode(out: torch.Tensor, q: torch.Tensor,
                      kcache: torch.Tensor, vcache: torch.Tensor,
                      block_tables: torch.Tensor, ctx_lens: torch.Tensor,
                      scale: float) -> torch.Tensor:
    B = q.shape[0]
    for b in range(B):
        length = int(ctx_lens[b])
        k = _gather_kv(kcache, block_tables[b], length)
        v = _gather_kv(vcache, block_tables[b], length)
        out[b] = attention(q[b:b + 1].float(), k, v, scale)[0].to(out.dtype)
    return out


def paged_attn_prefill(out: torch.Tensor, q: torch.Tensor,
                       kcache: torch.Tensor, vcache: torch.Tensor,
                       block_tables: torch.Tensor,
                       tile_q0: torch.Tensor, tile_qn: torch.Tensor,
                       tile_seq: torch.Tensor, tile_pos0: torch.Tensor,
                       scale: float) -> torch.Tensor:
    for t in range(tile_q0.shape[0]):
        q0, qn = int(tile_q0[t]), int(tile_qn[t])
        seq, pos0 = int(tile_seq[t]), int(tile_pos0[t])
        kv_len = pos0 + qn
        k = _gather_kv(kcache, block_tables[seq], kv_len)
        v = _gather_kv(vcache, block_tables[seq], kv_len)
        o = attention(q[q0:q0 + qn].float(), k, v, scale, causal_offset=pos0)
        out[q0:q0 + qn] = o.to(out.dtype)
    return out


def gather_rows(out: torch.Tensor, src: torch.Tensor,
                rows: torch.Tensor) -> torch.Tensor:
    out.copy_(src[rows.long()])
    return out
