"""Llama-family transformer over the paged-KV HIP ops.

Design (MI355X-first, not a port):
  * Flat-token batching: one forward serves a MIXED batch of decode tokens
    (one per running sequence, rows 0..n_decode-1) and chunked-prefill tokens
    (rows n_decode..T-1) — the continuous batcher builds a ForwardBatch and
    the model runs ONE pass for everything pending on this GPU.
  * Plain GEMMs go to hipBLASLt/rocBLAS via torch.matmul (bf16); the fused
    hot ops (rmsnorm+residual, rope, paged attention, kv scatter, swiglu)
    are the hand-written gfx950 kernels in quoracle_amd/ops.
  * Weights are random-init bf16 (no network for checkpoints — BASELINE.json
    says synthetic data / random-init weights); each model key gets its own
    seed so pool members are decorrelated voters.
  * MoE (Mixtral-style): top-k router + per-expert GEMMs, dense fallback
    implementation first; grouped-GEMM kernel is a later optimization.

The reference has no model execution at all (HTTP providers —
reference: lib/quoracle/models/model_query.ex); this file is the MI355X
replacement for that entire layer.
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Dict, List, Optional

import torch

from ..ops import dispatch as D
from .config import ModelConfig, get_config, instance_seed
from ..parallel.tp import TPContext, shard_gate_up, shard_qkv, shard_rows


@dataclass
class ForwardBatch:
    """Everything one mixed decode+prefill forward needs.

    Token order: decode tokens first (one per decode sequence, same order as
    the first n_decode rows of block_tables), then prefill-chunk tokens.
    """
    tokens: torch.Tensor          # [T] int32
    positions: torch.Tensor       # [T] int32
    slots: torch.Tensor           # [T] int32  (global slot = blk*BS + off)
    block_tables: torch.Tensor    # [B_all, MAXB] int32 (decode seqs first)
    n_decode: int = 0
    ctx_lens: Optional[torch.Tensor] = None   # [n_decode] int32 (incl. new tok)
    max_ctx: int = 0                          # host-known max(ctx_lens)
    max_kv: int = 0                           # host-known max tile kv extent
    # prefill tiles (QT=16 query rows each); empty tensors when no prefill
    tile_q0: Optional[torch.Tensor] = None    # [ntiles] int32 row into tokens
    tile_qn: Optional[torch.Tensor] = None
    tile_seq: Optional[torch.Tensor] = None   # row into block_tables
    tile_pos0: Optional[torch.Tensor] = None
    # 32-row tiling of the same prefill rows (8-wave MFMA kernel)
    tile32_q0: Optional[torch.Tensor] = None
    tile32_qn: Optional[torch.Tensor] = None
    tile32_seq: Optional[torch.Tensor] = None
    tile32_pos0: Optional[torch.Tensor] = None
    sample_rows: Optional[torch.Tensor] = None  # [R] int64 rows needing logits

    @property
    def total_tokens(self) -> int:
        return self.tokens.shape[0]


class KVCache:
    """Per-layer paged K/V pools: [num_blocks, Hkv, BS, D] bf16 each."""

    def __init__(self, cfg: ModelConfig, num_blocks: int, block_size: int,
                 device: torch.device):
        self.block_size = block_size
        self.num_blocks = num_blocks
        shape = (num_blocks, cfg.n_kv_heads, block_size, cfg.head_dim)
        self.k = [torch.zeros(shape, dtype=torch.bfloat16, device=device)
                  for _ in range(cfg.n_layers)]
        self.v = [torch.zeros(shape, dtype=torch.bfloat16, device=device)
                  for _ in range(cfg.n_layers)]

    def bytes(self) -> int:
        return sum(t.numel() * 2 for t in self.k) * 2


def _rand(shape, std: float, gen: torch.Generator, device, dtype):
    w = torch.empty(shape, device=device, dtype=torch.float32)
    w.normal_(0.0, std, generator=gen)
    return w.to(dtype)


class LlamaModel:
    def __init__(self, key: str, device: torch.device,
                 dtype: torch.dtype = torch.bfloat16,
                 cfg: Optional[ModelConfig] = None,
                 tp: Optional[TPContext] = None):
        self.key = key
        self.cfg = cfg or get_config(key)
        self.device = device
        self.dtype = dtype
        self.tp = tp or TPContext(0, 1)
        if self.tp.world > 1:
            from dataclasses import replace
            c = self.cfg
            assert c.n_heads % self.tp.world == 0 \
                and c.n_kv_heads % self.tp.world == 0 \
                and c.intermediate % self.tp.world == 0, \
                f"{c.name} not divisible by tp={self.tp.world}"
            # local (per-shard) architecture: forward + KV cache use this
            self.lcfg = replace(c, n_heads=c.n_heads // self.tp.world,
                                n_kv_heads=c.n_kv_heads // self.tp.world,
                                intermediate=c.intermediate // self.tp.world)
        else:
            self.lcfg = self.cfg
        self._init_weights()

    def _init_weights(self) -> None:
        cfg, dev, dt = self.cfg, self.device, self.dtype
        # Small (test) configs generate on CPU so the same key yields
        # bit-identical weights on CPU and GPU (numerics parity tests);
        # big models generate device-side (CPU randn of 8B params is minutes).
        gen_dev = "cpu" if cfg.hidden <= 1024 or dev.type == "cpu" else dev
        gen = torch.Generator(device=gen_dev)
        gen.manual_seed(instance_seed(self.key))
        std = 0.02
        out_std = std / math.sqrt(2 * cfg.n_layers)
        self.embed = _rand((cfg.vocab_size, cfg.hidden), std, gen, gen_dev, dt).to(dev)
        self.layers: List[Dict[str, torch.Tensor]] = []
        for _ in range(cfg.n_layers):
            tp = self.tp
            wqkv = _rand((cfg.hidden, cfg.qkv_dim), std, gen, gen_dev, dt)
            wo = _rand((cfg.q_dim, cfg.hidden), out_std, gen, gen_dev, dt)
            if tp.world > 1:
                wqkv = shard_qkv(wqkv, cfg.n_heads, cfg.n_kv_heads,
                                 cfg.head_dim, tp.rank, tp.world)
                wo = shard_rows(wo, tp.rank, tp.world)
            layer = {
                "attn_norm": torch.ones(cfg.hidden, dtype=dt, device=dev),
                "wqkv": wqkv.to(dev),
                "wo": wo.to(dev),
                "ffn_norm": torch.ones(cfg.hidden, dtype=dt, device=dev),
            }
            if cfg.is_moe:
                layer["router"] = _rand((cfg.hidden, cfg.n_experts), std, gen, gen_dev, dt).to(dev)
                wgu = _rand((cfg.n_experts, cfg.hidden, 2 * cfg.intermediate),
                            std, gen, gen_dev, dt)
                wdn = _rand((cfg.n_experts, cfg.intermediate, cfg.hidden),
                            out_std, gen, gen_dev, dt)
                if tp.world > 1:
                    wgu = shard_gate_up(wgu, cfg.intermediate, tp.rank, tp.world)
                    step = cfg.intermediate // tp.world
                    wdn = wdn[:, tp.rank * step:(tp.rank + 1) * step, :].contiguous()
                layer["w_gate_up"] = wgu.to(dev)
                layer["w_down"] = wdn.to(dev)
            else:
                wgu = _rand((cfg.hidden, 2 * cfg.intermediate), std, gen, gen_dev, dt)
                wdn = _rand((cfg.intermediate, cfg.hidden), out_std, gen, gen_dev, dt)
                if tp.world > 1:
                    wgu = shard_gate_up(wgu, cfg.intermediate, tp.rank, tp.world)
                    wdn = shard_rows(wdn, tp.rank, tp.world)
                layer["w_gate_up"] = wgu.to(dev)
                layer["w_down"] = wdn.to(dev)
            self.layers.append(layer)
        self.final_norm = torch.ones(cfg.hidden, dtype=dt, device=dev)
        if cfg.tie_embeddings:
            self.lm_head = self.embed
        else:
            self.lm_head = _rand((cfg.vocab_size, cfg.hidden), std, gen, gen_dev, dt).to(dev)
        self.scale = 1.0 / math.sqrt(cfg.head_dim)
        self._register_cpp()

    def _register_cpp(self) -> None:
        """Hand the weights to the C++ forward driver (GPU, non-TP): the
        whole layer loop then runs in one native call instead of ~400
        Python op dispatches (ops/csrc/forward.h)."""
        self.use_cpp = False
        if self.device.type != "cuda" or self.tp.world > 1:
            return
        import os as _os
        if _os.environ.get("QUORACLE_NO_CPP_FWD"):
            return
        from .. import ops as _ops
        if not _ops.available():
            return
        lcfg = self.lcfg
        layer_lists = []
        for layer in self.layers:
            lw = [layer["attn_norm"], layer["wqkv"], layer["wo"],
                  layer["ffn_norm"], layer["w_gate_up"], layer["w_down"]]
            if lcfg.is_moe:
                lw.append(layer["router"])
            layer_lists.append(lw)
        _ops.ext().register_model(
            self.key, self.embed, self.lm_head, self.final_norm, layer_lists,
            lcfg.n_heads, lcfg.n_kv_heads, lcfg.head_dim, lcfg.intermediate,
            lcfg.rope_theta, lcfg.rmsnorm_eps, lcfg.is_moe,
            lcfg.top_k_experts)
        self.use_cpp = True

    def param_bytes(self) -> int:
        n = self.embed.numel() + self.final_norm.numel()
        if self.lm_head is not self.embed:
            n += self.lm_head.numel()
        for layer in self.layers:
            n += sum(t.numel() for t in layer.values())
        return n * self.embed.element_size()

    def new_kv_cache(self, num_blocks: int, block_size: int) -> KVCache:
        return KVCache(self.lcfg, num_blocks, block_size, self.device)

    # -- forward -------------------------------------------------------------

    def forward(self, batch: ForwardBatch, kv: KVCache) -> torch.Tensor:
        if getattr(self, "use_cpp", False):
            from .. import ops as _ops
            import os as _os
            return _ops.ext().llama_forward(
                self.key, batch.tokens, batch.positions, batch.slots,
                batch.block_tables, batch.n_decode, batch.ctx_lens,
                batch.max_ctx, batch.tile_q0, batch.tile_qn, batch.tile_seq,
                batch.tile_pos0, batch.tile32_q0, batch.tile32_qn,
                batch.tile32_seq, batch.tile32_pos0, batch.max_kv,
                kv.k, kv.v,
                bool(_os.environ.get("QUORACLE_NO_MFMA_ATTN")))
        cfg = self.lcfg     # per-shard head/intermediate dims under TP
        T = batch.total_tokens
        res = torch.empty((T, cfg.hidden), dtype=self.dtype, device=self.device)
        D.gather_rows(res, self.embed, batch.tokens)
        h = torch.empty_like(res)
        D.rmsnorm(h, res, None, self.layers[0]["attn_norm"], cfg.rmsnorm_eps)

        attn_out = torch.empty((T, cfg.n_heads, cfg.head_dim),
                               dtype=self.dtype, device=self.device)
        has_prefill = batch.tile_q0 is not None and batch.tile_q0.numel() > 0

        for li, layer in enumerate(self.layers):
            qkv = h @ layer["wqkv"]                     # [T, qkv_dim]
            q = qkv[:, :cfg.q_dim].view(T, cfg.n_heads, cfg.head_dim).contiguous()
            k = qkv[:, cfg.q_dim:cfg.q_dim + cfg.kv_dim] \
                .view(T, cfg.n_kv_heads, cfg.head_dim).contiguous()
            v = qkv[:, cfg.q_dim + cfg.kv_dim:] \
                .view(T, cfg.n_kv_heads, cfg.head_dim).contiguous()
            D.rope_inplace(q, k, batch.positions, cfg.rope_theta)
            D.kv_append(kv.k[li], kv.v[li], k, v, batch.slots)
            if batch.n_decode:
                D.paged_attn_decode(
                    attn_out[:batch.n_decode], q[:batch.n_decode],
                    kv.k[li], kv.v[li], batch.block_tables,
                    batch.ctx_lens, self.scale, max_ctx=batch.max_ctx)
            if has_prefill:
                D.paged_attn_prefill(
                    attn_out, q, kv.k[li], kv.v[li], batch.block_tables,
                    batch.tile_q0, batch.tile_qn, batch.tile_seq,
                    batch.tile_pos0, self.scale, max_kv=batch.max_kv)
            proj = attn_out.view(T, cfg.q_dim) @ layer["wo"]
            self.tp.all_reduce_(proj)
            D.rmsnorm(h, proj, res, layer["ffn_norm"], cfg.rmsnorm_eps)
            if cfg.is_moe:
                ffn = self._moe_ffn(h, layer)
            else:
                gu = h @ layer["w_gate_up"]
                act = torch.empty((T, cfg.intermediate), dtype=self.dtype,
                                  device=self.device)
                D.swiglu(act, gu)
                ffn = act @ layer["w_down"]
            self.tp.all_reduce_(ffn)
            next_norm = (self.layers[li + 1]["attn_norm"]
                         if li + 1 < cfg.n_layers else self.final_norm)
            D.rmsnorm(h, ffn, res, next_norm, cfg.rmsnorm_eps)
        return h        # final-normed hidden states [T, hidden]

    # rows at or below this take the capacity-padded batched path (no host
    # sync, hipGraph-capturable); above it the per-expert GEMMs are
    # compute-bound and padding waste costs more than one counts sync
    MOE_BMM_MAX_ROWS = 512

    def _moe_ffn(self, h: torch.Tensor, layer: Dict[str, torch.Tensor]) -> torch.Tensor:
        """MoE dispatch.  Decode-size batches: capacity-padded bins + ONE
        bmm per FFN GEMM, zero host syncs (weight streaming dominates, so
        the padding is free).  Big prefill chunks: token-sorted contiguous
        GEMM slices per expert with a single counts sync amortized over
        the chunk."""
        cfg = self.lcfg     # local expert intermediate under TP
        T = h.shape[0]
        K = cfg.top_k_experts
        logits = (h @ layer["router"]).float()                 # [T, E]
        weights, experts = torch.topk(torch.softmax(logits, dim=-1), K, dim=-1)
        weights = (weights / weights.sum(dim=-1, keepdim=True)).to(h.dtype)
        expert_flat = experts.reshape(-1)                      # [T*K]
        order = torch.argsort(expert_flat, stable=True)
        token_of = order // K                                  # source token
        if T * K <= self.MOE_BMM_MAX_ROWS:
            return self._moe_ffn_bmm(h, layer, weights, experts, order,
                                     token_of)
        h_sorted = h[token_of]                                 # [T*K, d]
        counts = torch.bincount(expert_flat, minlength=cfg.n_experts)
        counts_l = counts.tolist()
        down_sorted = torch.empty_like(h_sorted)
        start = 0
        for e, n in enumerate(counts_l):
            if n == 0:
                continue
            he = h_sorted[start:start + n]
            gu = he @ layer["w_gate_up"][e]
            act = torch.empty((n, cfg.intermediate), dtype=h.dtype,
                              device=h.device)
            D.swiglu(act, gu)
            down_sorted[start:start + n] = act @ layer["w_down"][e]
            start += n
        w_sorted = weights.reshape(-1)[order].unsqueeze(1)
        out = torch.zeros_like(h)
        out.index_add_(0, token_of, down_sorted * w_sorted)
        return out

    def _moe_ffn_bmm(self, h: torch.Tensor, layer: Dict[str, torch.Tensor],
                     weights: torch.Tensor, experts: torch.Tensor,
                     order: torch.Tensor, token_of: torch.Tensor) -> torch.Tensor:
        """Capacity-padded batched MoE FFN: each expert gets a fixed bin of
        T rows (an expert receives at most one row per token), tokens are
        scattered into their bins on the GPU, both GEMMs run as one bmm
        over [E, T, *] — no counts.cpu(), so decode steps stay sync-free
        and capture into hipGraphs."""
        cfg = self.lcfg
        T, hidden = h.shape
        K = cfg.top_k_experts
        E = cfg.n_experts
        S = T * K
        expert_flat = experts.reshape(-1)
        e_sorted = expert_flat[order]
        # scatter_add, not bincount: bincount syncs to size its output,
        # which breaks hipGraph capture of MoE decode steps
        counts = torch.zeros(E, dtype=expert_flat.dtype,
                             device=expert_flat.device)
        counts.scatter_add_(0, expert_flat, torch.ones_like(expert_flat))
        raw_off = torch.cumsum(counts, 0) - counts             # exclusive
        pos = torch.arange(S, device=h.device) - raw_off[e_sorted]
        idx = e_sorted * T + pos                               # bin slots
        bins = torch.zeros((E * T, hidden), dtype=h.dtype, device=h.device)
        bins.index_copy_(0, idx, h[token_of])
        gu = torch.bmm(bins.view(E, T, hidden), layer["w_gate_up"])
        act = torch.empty((E * T, cfg.intermediate), dtype=h.dtype,
                          device=h.device)
        D.swiglu(act, gu.view(E * T, -1))
        down = torch.bmm(act.view(E, T, cfg.intermediate), layer["w_down"])
        w_sorted = weights.reshape(-1)[order].unsqueeze(1)
        vals = down.view(E * T, hidden)[idx] * w_sorted
        out = torch.zeros_like(h)
        out.index_add_(0, token_of, vals)
        return out

    def _moe_ffn_naive(self, h: torch.Tensor, layer: Dict[str, torch.Tensor]) -> torch.Tensor:
        """Reference implementation kept for numerics tests."""
        cfg = self.lcfg
        logits = (h @ layer["router"]).float()
        weights, experts = torch.topk(torch.softmax(logits, dim=-1),
                                      cfg.top_k_experts, dim=-1)
        weights = (weights / weights.sum(dim=-1, keepdim=True)).to(h.dtype)
        out = torch.zeros_like(h)
        for e in range(cfg.n_experts):
            mask = (experts == e)
            rows = mask.any(dim=-1).nonzero(as_tuple=True)[0]
            if rows.numel() == 0:
                continue
            he = h[rows]
            gu = he @ layer["w_gate_up"][e]
            act = torch.empty((he.shape[0], cfg.intermediate),
                              dtype=h.dtype, device=h.device)
            D.swiglu(act, gu)
            down = act @ layer["w_down"][e]
            w_e = (weights * mask.to(weights.dtype)).sum(dim=-1)[rows]
            out.index_add_(0, rows, down * w_e[:, None])
        return out

    def compute_logits(self, hidden: torch.Tensor,
                       rows: torch.Tensor) -> torch.Tensor:
        """Logits only for the rows that need sampling: [R, vocab] fp32."""
        if getattr(self, "use_cpp", False):
            from .. import ops as _ops
            return _ops.ext().llama_logits(self.key, hidden, rows)
        return (hidden[rows] @ self.lm_head.T).float()

    # -- embedding-model path (consensus vote) -------------------------------

    def embed_texts_hidden(self, token_batches: List[torch.Tensor]) -> torch.Tensor:
        """Mean-pooled final hidden per text — the embedder forward.

        Runs each text as a self-contained prefill over a throwaway KV
        cache; used only by the small embed model so simplicity wins.
        """
        cfg = self.cfg
        outs = []
        bs = 16
        for toks in token_batches:
            L = toks.shape[0]
            nb = (L + bs - 1) // bs
            kvc = self.new_kv_cache(nb, bs)
            bt = torch.arange(nb, dtype=torch.int32,
                              device=self.device).unsqueeze(0)
            pos = torch.arange(L, dtype=torch.int32, device=self.device)
            ntiles = (L + 15) // 16
            t0 = torch.arange(ntiles, dtype=torch.int32,
                              device=self.device) * 16
            qn = torch.clamp(torch.full_like(t0, L) - t0, max=16)
            batch = ForwardBatch(
                tokens=toks.to(torch.int32), positions=pos, slots=pos.clone(),
                block_tables=bt, n_decode=0,
                tile_q0=t0, tile_qn=qn,
                tile_seq=torch.zeros_like(t0), tile_pos0=t0)
            hidden = self.forward(batch, kvc)
            outs.append(hidden.float().mean(dim=0))
        return torch.stack(outs)
