"""Tensor-parallel numerics on CPU (gloo, world 2): the TP=2 sharded
forward must reproduce the single-process forward bit-for-tolerance.
Covers dense (tiny) and MoE (tiny-moe via config override) blocks —
BASELINE config 5's mechanism without GPUs."""

import os

import pytest
import torch
import torch.multiprocessing as mp


def _mk_batch(device, n_prompt, vocab):
    from quoracle_amd.models.llama import ForwardBatch
    bs = 16
    nb = (n_prompt + bs - 1) // bs + 1
    toks = torch.arange(n_prompt, dtype=torch.int32, device=device) % vocab
    pos = torch.arange(n_prompt, dtype=torch.int32, device=device)
    ntiles = (n_prompt + 15) // 16
    t0 = torch.arange(ntiles, dtype=torch.int32, device=device) * 16
    qn = torch.clamp(torch.full_like(t0, n_prompt) - t0, max=16)
    return ForwardBatch(
        tokens=toks, positions=pos, slots=pos.clone(),
        block_tables=torch.arange(nb, dtype=torch.int32,
                                  device=device).unsqueeze(0),
        n_decode=0, tile_q0=t0, tile_qn=qn,
        tile_seq=torch.zeros_like(t0), tile_pos0=t0), nb


def _single_forward(cfg_name, moe):
    from quoracle_amd.models import LlamaModel
    from quoracle_amd.models.config import PRESETS
    from dataclasses import replace
    cfg = PRESETS["tiny"]
    if moe:
        cfg = replace(cfg, n_experts=4, top_k_experts=2)
    model = LlamaModel("tp-test", torch.device("cpu"), cfg=cfg)
    batch, nb = _mk_batch(torch.device("cpu"), 33, cfg.vocab_size)
    kv = model.new_kv_cache(nb, 16)
    return model.forward(batch, kv).float()


def _tp_worker(rank, world, port, moe, out_q):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from dataclasses import replace
        from quoracle_amd.models import LlamaModel
        from quoracle_amd.models.config import PRESETS
        from quoracle_amd.parallel.tp import TPContext
        cfg = PRESETS["tiny"]
        if moe:
            cfg = replace(cfg, n_experts=4, top_k_experts=2)
        tp = TPContext(rank, world)
        model = LlamaModel("tp-test", torch.device("cpu"), cfg=cfg, tp=tp)
        batch, nb = _mk_batch(torch.device("cpu"), 33, cfg.vocab_size)
        kv = model.new_kv_cache(nb, 16)
        hidden = model.forward(batch, kv).float()
        if rank == 0:
            # plain ndarray: tensors ride shared-memory FDs that die with
            # the producer process
            out_q.put(hidden.numpy())
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("moe", [False, True])
def test_tp2_matches_single(moe):
    single = _single_forward("tiny", moe)
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    port = 29531 + int(moe)
    procs = [ctx.Process(target=_tp_worker, args=(r, 2, port, moe, out_q))
             for r in range(2)]
    for p in procs:
        p.start()
    tp_hidden = torch.from_numpy(out_q.get(timeout=180))
    for p in procs:
        p.join(timeout=60)
    assert all(p.exitcode == 0 for p in procs)
    rel = (tp_hidden - single).norm() / single.norm()
    assert rel < 0.05, f"TP mismatch rel={rel}"


def test_moe_sorted_dispatch_matches_naive():
    """Token-sorted MoE == per-expert masked loop (CPU, fp32-ish check)."""
    import torch
    from dataclasses import replace
    from quoracle_amd.models import LlamaModel
    from quoracle_amd.models.config import PRESETS
    cfg = replace(PRESETS["tiny"], n_experts=4, top_k_experts=2)
    model = LlamaModel("moe-sort", torch.device("cpu"), cfg=cfg)
    torch.manual_seed(3)
    h = torch.randn(37, cfg.hidden, dtype=torch.bfloat16)
    layer = model.layers[0]
    a = model._moe_ffn(h, layer).float()
    b = model._moe_ffn_naive(h, layer).float()
    assert torch.allclose(a, b, atol=2e-2, rtol=2e-2), \
        (a - b).abs().max().item()
