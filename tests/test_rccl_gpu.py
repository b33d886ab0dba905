"""RCCL on a real MI355X box (single-rank world): library init, basic
collectives, and — the part the TP decode path depends on — an RCCL
collective captured inside a hipGraph and replayed.

Multi-rank xGMI behavior needs >1 GPU (the driver's round-end scale run);
these tests pin down everything provable on one device: RCCL initializes
on this ROCm stack, enqueues collectives on the current stream, and those
enqueues are hipGraph-capturable and replay correctly
(engine/tp_engine.py keeps graphs disabled for world>1 until a multi-GPU
run confirms the same holds across ranks).
"""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def nccl_pg():
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    import torch.distributed as dist
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29713")
    dist.init_process_group("nccl", rank=0, world_size=1)
    yield dist
    dist.destroy_process_group()


def test_rccl_world1_collectives(nccl_pg):
    dist = nccl_pg
    dev = torch.device("cuda:0")
    x = torch.arange(1024, dtype=torch.float32, device=dev)
    dist.all_reduce(x)
    torch.cuda.synchronize()
    assert torch.equal(x.cpu(), torch.arange(1024, dtype=torch.float32))

    y = torch.full((256,), 3.0, device=dev)
    out = [torch.empty_like(y)]
    dist.all_gather(out, y)
    torch.cuda.synchronize()
    assert torch.equal(out[0].cpu(), y.cpu())


def test_rccl_allreduce_inside_hip_graph(nccl_pg):
    """Capture a compute + all_reduce + compute chain into a hipGraph and
    replay it with fresh inputs: the collective must participate in the
    graph (correct results on every replay), which is what TP decode
    capture requires."""
    dist = nccl_pg
    dev = torch.device("cuda:0")
    inp = torch.zeros(4096, device=dev)
    outp = torch.zeros(4096, device=dev)

    def body():
        t = inp * 2.0
        dist.all_reduce(t)        # world 1: semantic identity, real enqueue
        outp.copy_(t + 1.0)

    # warmup on a side stream (required before capture)
    s = torch.cuda.Stream(dev)
    s.wait_stream(torch.cuda.current_stream(dev))
    with torch.cuda.stream(s):
        for _ in range(2):
            body()
    torch.cuda.current_stream(dev).wait_stream(s)
    torch.cuda.synchronize(dev)

    graph = torch.cuda.CUDAGraph()
    with torch.cuda.graph(graph, capture_error_mode="thread_local"):
        body()

    for fill in (1.0, 5.0, -2.0):
        inp.fill_(fill)
        graph.replay()
        torch.cuda.synchronize(dev)
        expect = fill * 2.0 + 1.0
        assert torch.allclose(outp, torch.full_like(outp, expect)), \
            f"replay with fill={fill}: got {outp[:4].tolist()}"
