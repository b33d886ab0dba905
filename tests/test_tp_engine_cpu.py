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


def _tp_reset_worker(rank, world, port, out_q):
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
        mgr_before = id(eng.models["tiny#tp"].mgr)
        if rank == 1:
            # rank-local fault: must NOT reset immediately (would desync)
            eng.reset_model("tiny#tp")
            assert id(eng.models["tiny#tp"].mgr) == mgr_before
            assert eng._reset_pending == {"tiny#tp"}
        if rank == 0:
            r = eng.generate_sync(GenerateRequest(
                model_key="tiny#tp",
                messages=[{"role": "user", "content": "after fault"}],
                temperature=0.7, max_tokens=200, seed=5,
                action_grammar=True, session_id="tpr"), timeout=300)
            eng.request_stop()
            eng.step()
            out_q.put(("r0", r.error, id(eng.models["tiny#tp"].mgr) != mgr_before))
        else:
            serve_tp_replica(eng)
            out_q.put(("r1", None, id(eng.models["tiny#tp"].mgr) != mgr_before))
    finally:
        dist.destroy_process_group()


def test_tp_group_watchdog_resets_all_ranks_together():
    """A fault on ONE TP rank defers its KV reset to the step sync point,
    where the MAX-all-reduced mask makes EVERY rank rebuild — lockstep
    block allocation survives and generation still completes."""
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    procs = [ctx.Process(target=_tp_reset_worker, args=(r, 2, 29547, out_q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        who, err, reset_applied = out_q.get(timeout=240)
        results[who] = (err, reset_applied)
    for p in procs:
        p.join(timeout=120)
    assert all(p.exitcode == 0 for p in procs)
    assert results["r0"][0] is None, results["r0"][0]
    # BOTH ranks rebuilt their BlockManager (not just the faulty one)
    assert results["r0"][1] and results["r1"][1], results


def test_tp_engine_world1_serves_requests():
    """TP=1 TPEngine (config-5 mechanism at trivial world) must serve
    through the normal inbox — found by the GPU rehearsal: _tp_direct
    suppressed admission entirely at world=1."""
    import torch.distributed as dist
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ["MASTER_PORT"] = "29553"
    dist.init_process_group("gloo", rank=0, world_size=1)
    try:
        from quoracle_amd.engine.api import GenerateRequest
        from quoracle_amd.engine.tp_engine import TPEngine
        from quoracle_amd.parallel.tp import TPContext
        eng = TPEngine(["tiny#tp1"], TPContext(0, 1),
                       device=torch.device("cpu"), kv_blocks_override=256,
                       embed_model_key=None, prefill_chunk=64).start()
        try:
            r = eng.generate_sync(GenerateRequest(
                model_key="tiny#tp1",
                messages=[{"role": "user", "content": "world one"}],
                temperature=0.7, max_tokens=120, seed=9,
                action_grammar=True, session_id="w1"), timeout=120)
            assert r.ok, r.error
            assert r.output_tokens > 0
        finally:
            eng.stop()
    finally:
        dist.destroy_process_group()
