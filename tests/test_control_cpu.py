"""Control-plane round trip on CPU (gloo, world 2): rank 0 drives a
ControlClient against rank 1's serve_engine(LocalEngine) — generate
round-trips with reply matching, barrier + reduce_max bracketing, and a
clean stop/bye shutdown (quoracle_amd/parallel/control.py; the reference's
only distribution is BEAM messaging — SURVEY.md §5.8)."""

import asyncio
import os

import pytest
import torch
import torch.multiprocessing as mp


def _worker(rank, world, port, out_q):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from quoracle_amd.engine.api import GenerateRequest
        from quoracle_amd.engine.engine import LocalEngine
        from quoracle_amd.parallel.control import (ControlClient,
                                                   RemoteEngine,
                                                   serve_engine,
                                                   shard_models)
        assert shard_models(["a", "b", "c"], 2) == [["a", "c"], ["b"]]
        if rank == 1:
            eng = LocalEngine(["tiny"], device=torch.device("cpu"))
            serve_engine(eng)            # returns after rank 0's shutdown()
            return
        client = ControlClient(remote_ranks=[1])
        remote = RemoteEngine(1, client)
        assert remote.count_tokens("hello") > 0
        assert remote.context_limit("tiny") > 0

        async def drive():
            reqs = [GenerateRequest(model_key="tiny",
                                    messages=[{"role": "user",
                                               "content": f"say {i}"}],
                                    max_tokens=4, temperature=0.0, seed=i)
                    for i in range(3)]
            return await asyncio.gather(*[remote.generate(r) for r in reqs])

        client.barrier_all()
        results = asyncio.run(drive())
        client.barrier_all()
        elapsed = client.reduce_max_elapsed(0.001)
        client.shutdown()
        ok = (len(results) == 3
              and all(r.error is None and r.output_tokens >= 1
                      for r in results)
              and elapsed >= 0.0)
        out_q.put((ok, [getattr(r, "error", "?") for r in results], elapsed))
    finally:
        dist.destroy_process_group()


def test_control_plane_generate_roundtrip():
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, 29553, out_q))
             for r in range(2)]
    for p in procs:
        p.start()
    ok, errors, elapsed = out_q.get(timeout=240)
    for p in procs:
        p.join(timeout=120)
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]
    assert ok, errors


def _worker3(rank, world, port, out_q):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from quoracle_amd.engine.api import GenerateRequest
        from quoracle_amd.engine.engine import LocalEngine
        from quoracle_amd.parallel.control import (ControlClient,
                                                   RemoteEngine, serve_engine)
        if rank != 0:
            eng = LocalEngine(["tiny"], device=torch.device("cpu"),
                              embed_model_key=None)
            serve_engine(eng)
            return
        client = ControlClient(remote_ranks=[1, 2])
        remotes = {r: RemoteEngine(r, client) for r in (1, 2)}

        async def drive():
            reqs = [(r, GenerateRequest(
                model_key="tiny",
                messages=[{"role": "user", "content": f"m{r}-{i}"}],
                max_tokens=3, temperature=0.0, seed=i))
                for r in (1, 2) for i in range(3)]
            return await asyncio.gather(
                *[remotes[r].generate(q) for r, q in reqs])

        results = asyncio.run(drive())
        client.shutdown()
        out_q.put((all(r.error is None for r in results), len(results)))
    finally:
        dist.destroy_process_group()


def test_control_plane_two_servers():
    """World 3: rank 0 fans out to two independent engine servers with
    concurrent in-flight requests (the shape of the driver's 8-GPU run)."""
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    procs = [ctx.Process(target=_worker3, args=(r, 3, 29554, out_q))
             for r in range(3)]
    for p in procs:
        p.start()
    ok, n = out_q.get(timeout=240)
    for p in procs:
        p.join(timeout=120)
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]
    assert ok and n == 6
