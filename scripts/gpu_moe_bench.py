"""MoE FFN micro-bench: capacity-padded bmm (no host sync) vs the
token-sorted per-expert loop, at decode and prefill sizes, on a
Mixtral-8x7B-shaped layer (hidden 4096, E=8, top-2, intermediate 14336
— TP=1 worth of experts).  Numerics are checked against the naive
per-expert reference first."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from dataclasses import replace

from quoracle_amd.models.config import PRESETS
from quoracle_amd.models.llama import LlamaModel

dev = torch.device("cuda:0")
cfg = replace(PRESETS["llama3-8b"], n_experts=8, top_k_experts=2,
              n_layers=1, intermediate=14336, vocab_size=1024)
torch.manual_seed(3)
m = LlamaModel("moe-bench", dev, dtype=torch.bfloat16, cfg=cfg)
layer = m.layers[0]


def timeit(fn, n=20):
    fn(); torch.cuda.synchronize()
    t = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t) / n * 1e3


for T in (8, 16, 64, 256, 2048):
    h = torch.randn(T, cfg.hidden, device=dev, dtype=torch.bfloat16)
    naive = m._moe_ffn_naive(h, layer).float()
    fast = m._moe_ffn(h, layer).float()
    rel = (naive - fast).norm() / max(1e-6, naive.norm())
    path = "bmm" if T * cfg.top_k_experts <= m.MOE_BMM_MAX_ROWS else "sorted"
    ms_auto = timeit(lambda: m._moe_ffn(h, layer))
    old = LlamaModel.MOE_BMM_MAX_ROWS
    try:
        LlamaModel.MOE_BMM_MAX_ROWS = 0      # force sorted loop (host sync)
        ms_loop = timeit(lambda: m._moe_ffn(h, layer))
    finally:
        LlamaModel.MOE_BMM_MAX_ROWS = old
    # weight-streaming floor: all experts' FFN weights once
    wbytes = 8 * (cfg.hidden * 2 * 14336 + 14336 * cfg.hidden) * 2
    print(f"T={T:5d} path={path:6s} {ms_auto:7.3f} ms  "
          f"(sorted-loop {ms_loop:7.3f} ms, {ms_loop/ms_auto:4.2f}x)  "
          f"weights {wbytes/ms_auto/1e9:5.2f} TB/s  rel {rel:.4f}")
