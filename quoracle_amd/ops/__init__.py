"""quoracle_amd.ops — HIP/CDNA4 kernels (gfx950) with fail-loud loading.

On a GPU box the extension MUST be present: every op raises if the native
module failed to import, so a silent eager fallback can never masquerade as
the HIP path.  CPU-side tests use ops.reference (plain fp32 PyTorch
implementations of the same ops) as the numerics baseline.
"""

from __future__ import annotations

import importlib.util
import os
from typing import Optional

_HERE = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_HERE, "_quoracle_ops.so")

_ext = None
_load_error: Optional[str] = None


def _load():
    global _ext, _load_error
    if _ext is not None:
        return _ext
    if not os.path.exists(_SO):
        _load_error = f"native extension not built: {_SO} missing " \
                      "(run quoracle_amd/ops/build.py)"
        return None
    try:
        spec = importlib.util.spec_from_file_location("_quoracle_ops", _SO)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        _ext = mod
    except Exception as exc:  # noqa: BLE001
        _load_error = f"native extension failed to load: {exc}"
        return None
    return _ext


def available() -> bool:
    return _load() is not None


def ext():
    """The native module; raises loudly when missing (no silent fallback)."""
    mod = _load()
    if mod is None:
        raise RuntimeError(
            f"quoracle_amd HIP extension unavailable: {_load_error}")
    return mod


# -- thin wrappers -----------------------------------------------------------

def rmsnorm_fused(y, x, residual, w, eps: float = 1e-5):
    ext().rmsnorm_fused(y, x, residual, w, eps)
    return y


def swiglu(out, gate_up):
    ext().swiglu(out, gate_up)
    return out


def rope_inplace(q, k, pos, theta: float = 500000.0):
    ext().rope_inplace(q, k, pos, theta)


def kv_append(kcache, vcache, k, v, slots):
    ext().kv_append(kcache, vcache, k, v, slots)


def paged_attn_decode(out, q, kcache, vcache, block_tables, ctx_lens,
                      scale: float):
    ext().paged_attn_decode(out, q, kcache, vcache, block_tables, ctx_lens,
                            scale)
    return out


def paged_attn_prefill(out, q, kcache, vcache, block_tables, tile_q0, tile_qn,
                       tile_seq, tile_pos0, scale: float):
    ext().paged_attn_prefill(out, q, kcache, vcache, block_tables, tile_q0,
                             tile_qn, tile_seq, tile_pos0, scale)
    return out


def cosine_sim_matrix(out, x):
    ext().cosine_sim_matrix(out, x)
    return out


def gather_rows(out, src, rows):
    ext().gather_rows(out, src, rows)
    return out


def paged_attn_decode_split(out, q, kcache, vcache, block_tables, ctx_lens,
                            scale, part_m, part_l, part_acc):
    ext().paged_attn_decode_split(out, q, kcache, vcache, block_tables,
                                  ctx_lens, scale, part_m, part_l, part_acc)
    return out


def paged_attn_prefill_mfma(out, q, kcache, vcache, block_tables, tile_q0,
                            tile_qn, tile_seq, tile_pos0, scale):
    ext().paged_attn_prefill_mfma(out, q, kcache, vcache, block_tables,
                                  tile_q0, tile_qn, tile_seq, tile_pos0,
                                  scale)
    return out


def paged_attn_prefill_mfma32(out, q, kcache, vcache, block_tables, tile_q0,
                              tile_qn, tile_seq, tile_pos0, scale):
    ext().paged_attn_prefill_mfma32(out, q, kcache, vcache, block_tables,
                                    tile_q0, tile_qn, tile_seq, tile_pos0,
                                    scale)
    return out
