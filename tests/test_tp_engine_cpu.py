"""Lockstep TP engine on CPU (gloo, world 2): a constrained generate
through TP=2 replicas must produce EXACTLY the same text as the
single-process engine (deterministic lockstep)."""

import os

import pytest
import torch
import torch.multiprocessing as mp


def _single_text():
    from quoracle_amd.engine.api import GenerateRequest
    from quoracle_amd.engine.engine import LocalEngine
    eng = LocalEngine(["tiny#tp"], device=torch.device("cpu"),
                      kv_blocks_override=256, embed_model_key=None,
                      prefill_chunk=64)
    r = eng.generate_sync(GenerateRequest(
        model_key="tiny#tp",
        messages=[{"role": "user", "content": "hello tp"}],
        temperature=0.7, max_tokens=400, seed=21,
        action_grammar=True, session_id="tp1"), timeout=300)
    assert r.ok, r.error
    return r.text


def _tp_worker(rank, world, port, out_q):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from quoracle_amd.engine.api import GenerateRequest
        from quoracle_amd.engine.tp_engine import TPEngine, serve_tp_replica
        from quoracle_amd.parallel.tp import TPContext
        tp = TPContext(rank, world)
        eng = TPEngine(["tiny#tp"], tp, device=torch.device("cpu"),
                       kv_blocks_override=256, embed_model_key=None,
                       prefill_chunk=64)
        if rank == 0:
            r = eng.generate_sync(GenerateRequest(
                model_key="tiny#tp",
                messages=[{"role": "user", "content": "hello tp"}],
                temperature=0.7, max_tokens=400, seed=21,
                action_grammar=True, session_id="tp1"), timeout=300)
            eng.request_stop()
            eng.step()
            out_q.put((r.error, r.text))
        else:
            serve_tp_replica(eng)
    finally:
        dist.destroy_process_group()


def test_tp_engine_lockstep_matches_single():
    single = _single_text()
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    procs = [ctx.Process(target=_tp_worker, args=(r, 2, 29541, out_q))
             for r in range(2)]
    for p in procs:
        p.start()
    err, text = out_q.get(timeout=240)
    for p in procs:
        p.join(timeout=120)
    assert err is None, err
    assert all(p.exitcode == 0 for p in procs)
    assert text == single, f"TP text diverged:\n{text}\nvs\n{single}"
