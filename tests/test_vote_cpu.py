"""Distributed embedding vote on CPU (gloo, world 2): round-robin shard +
all-gather must reproduce the local embedding order and values."""

import os

import pytest
import torch
import torch.multiprocessing as mp

TEXTS = [f"candidate action text number {i}" for i in range(7)]


def _worker(rank, world, port, out_q):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from quoracle_amd.engine.engine import LocalEngine
        from quoracle_amd.parallel.vote import (all_gather_vote,
                                                assign_round_robin,
                                                compute_local_embeddings)
        eng = LocalEngine([], device=torch.device("cpu"),
                          embed_model_key="embed-small")
        shards = assign_round_robin(TEXTS, world)
        counts = [len(s) for s in shards]
        local = compute_local_embeddings(eng, shards[rank])
        blocks = all_gather_vote(local, counts)
        if rank == 0:
            out = [None] * len(TEXTS)
            for r, block in enumerate(blocks):
                for j, row in enumerate(block.cpu().tolist()):
                    out[r + j * world] = row
            direct = eng.embed_sync(TEXTS)
            import numpy as np
            out_q.put((np.array(out, dtype=np.float32),
                       np.array(direct, dtype=np.float32)))
    finally:
        dist.destroy_process_group()


def test_vote_all_gather_matches_local():
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, 29551, out_q))
             for r in range(2)]
    for p in procs:
        p.start()
    gathered, direct = out_q.get(timeout=240)
    for p in procs:
        p.join(timeout=120)
    assert all(p.exitcode == 0 for p in procs)
    # same embed model replica everywhere -> identical vectors
    d = gathered[:, :direct.shape[1]]
    assert abs(d - direct).max() < 1e-4


def _demb_worker(rank, world, port, out_q):
    import numpy as np
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from quoracle_amd.engine.engine import LocalEngine
        from quoracle_amd.parallel.control import ControlClient, serve_engine
        from quoracle_amd.parallel.vote import DistributedEmbedder
        eng = LocalEngine([], device=torch.device("cpu"),
                          embed_model_key="embed-small")
        if rank != 0:
            serve_engine(eng)
            return
        client = ControlClient([1])
        demb = DistributedEmbedder(eng, client, world)
        vecs = demb(TEXTS)                      # distributed path
        direct = eng.embed_sync(TEXTS)          # local reference
        client.shutdown()
        out_q.put((np.array(vecs, dtype=np.float32),
                   np.array(direct, dtype=np.float32)))
    finally:
        dist.destroy_process_group()


def test_distributed_embedder_matches_local():
    """The bench's rank-0 embedder: shard fan-out over the control plane +
    all-gather merge must reproduce local embeddings in input order."""
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    procs = [ctx.Process(target=_demb_worker, args=(r, 2, 29556, out_q))
             for r in range(2)]
    for p in procs:
        p.start()
    vecs, direct = out_q.get(timeout=240)
    for p in procs:
        p.join(timeout=120)
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]
    assert vecs.shape[0] == len(TEXTS)
    assert abs(vecs[:, :direct.shape[1]] - direct).max() < 1e-4
