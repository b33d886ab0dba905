"""Two-plane distributed runtime.

Control plane — torch.distributed p2p over gloo (CPU tensors): rank 0 runs
the orchestrator (agent tree, consensus, actions); every other rank runs an
EngineServer hosting its shard of the model pool.  Generate requests travel
rank0 -> host rank as framed pickles; results travel back tagged by request
id, so many agent turns are in flight against every GPU at once and each
server's LocalEngine continuous-batches them into one forward per step.

Data plane — RCCL over xGMI (backend "nccl" on ROCm) for tensor collectives:
the consensus vote's embedding exchange and TP collectives use the default
device group, never this channel.

The reference's only distribution is BEAM messaging (Phoenix.PubSub /
GenServer casts, reference: lib/quoracle/pubsub/agent_events.ex); this file
is its MI355X replacement, built for one-process-per-GPU torchrun.
"""

from __future__ import annotations

import pickle
import threading
from typing import Any, Dict, List, Sequence

import torch
import torch.distributed as dist

REQ_TAG = 11
REP_TAG = 12


def _send_obj(obj: Any, dst: int, tag: int, lock: threading.Lock) -> None:
    payload = pickle.dumps(obj, protocol=pickle.HIGHEST_PROTOCOL)
    buf = torch.frombuffer(bytearray(payload), dtype=torch.uint8)
    header = torch.tensor([buf.numel()], dtype=torch.int64)
    with lock:
        dist.send(header, dst=dst, tag=tag)
        dist.send(buf, dst=dst, tag=tag)


def _recv_obj(src: int, tag: int) -> Any:
    header = torch.zeros(1, dtype=torch.int64)
    dist.recv(header, src=src, tag=tag)
    buf = torch.zeros(int(header.item()), dtype=torch.uint8)
    dist.recv(buf, src=src, tag=tag)
    return pickle.loads(buf.numpy().tobytes())


def shard_models(model_keys: Sequence[str], world: int) -> List[List[str]]:
    """Round-robin model placement: model i -> rank i % world."""
    shards: List[List[str]] = [[] for _ in range(world)]
    for i, key in enumerate(model_keys):
        shards[i % world].append(key)
    return shards


class RemoteEngine:
    """Engine-protocol proxy for models hosted by another rank.

    Token accounting is local (deterministic tokenizer + static configs), so
    only generate() round-trips.
    """

    def __init__(self, rank: int, client: "ControlClient"):
        self.rank = rank
        self.client = client
        from ..engine.tokenizer import ByteTokenizer
        self._tok = ByteTokenizer()

    async def generate(self, request):
        return await self.client.call(self.rank, request)

    async def embed(self, texts):
        raise RuntimeError("embedding runs on the orchestrator rank")

    def count_tokens(self, text: str) -> int:
        return self._tok.count(text)

    def context_limit(self, model_key: str) -> int:
        from ..models import get_config
        return get_config(model_key).max_context

    def output_limit(self, model_key: str) -> int:
        from ..models import get_config
        return get_config(model_key).max_output


class ControlClient:
    """Rank-0 side: request fan-out + reply matching."""

    def __init__(self, remote_ranks: Sequence[int]):
        self._locks: Dict[int, threading.Lock] = {
            r: threading.Lock() for r in remote_ranks}
        self._pending: Dict[str, Any] = {}
        self._plock = threading.Lock()
        self._next_id = 0
        self._receivers = [
            threading.Thread(target=self._recv_loop, args=(r,), daemon=True,
                             name=f"ctl-recv-{r}")
            for r in remote_ranks]
        for t in self._receivers:
            t.start()

    def _recv_loop(self, rank: int) -> None:
        while True:
            try:
                msg = _recv_obj(rank, REP_TAG)
            except Exception:
                return
            if msg is None or msg.get("kind") == "bye":
                return
            rid = msg["rid"]
            with self._plock:
                entry = self._pending.pop(rid, None)
            if entry is not None:
                loop, fut = entry
                loop.call_soon_threadsafe(
                    lambda f=fut, r=msg["result"]: f.done() or f.set_result(r))

    async def call(self, rank: int, request) -> Any:
        import asyncio
        loop = asyncio.get_running_loop()
        fut: "asyncio.Future" = loop.create_future()
        with self._plock:
            self._next_id += 1
            rid = f"r{self._next_id}"
            self._pending[rid] = (loop, fut)
        await loop.run_in_executor(
            None, _send_obj, {"kind": "gen", "rid": rid, "request": request},
            rank, REQ_TAG, self._locks[rank])
        return await fut

    def shutdown(self, join_timeout: float = 30.0) -> None:
        for rank, lock in self._locks.items():
            try:
                _send_obj({"kind": "stop"}, rank, REQ_TAG, lock)
            except Exception:
                pass
        # each server replies "bye" after engine.stop(); wait for the
        # receiver threads to drain it — tearing down the process group
        # while a thread is blocked in dist.recv aborts the process
        for t in self._receivers:
            t.join(timeout=join_timeout)

    def embed_gather(self, shards, counts) -> None:
        """Fan the per-rank text shards to the servers; every rank then
        joins one RCCL all-gather of the embedding blocks (parallel/vote.py)."""
        for rank, lock in self._locks.items():
            _send_obj({"kind": "embed_gather", "texts": shards[rank],
                       "counts": counts}, rank, REQ_TAG, lock)

    def barrier_all(self) -> None:
        """Device-synchronized barrier across orchestrator + all servers
        (brackets the bench's timed region)."""
        for rank, lock in self._locks.items():
            _send_obj({"kind": "barrier"}, rank, REQ_TAG, lock)
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        dist.barrier()

    def reduce_max_elapsed(self, elapsed: float) -> float:
        """MAX over ranks of the time between the last two barriers."""
        for rank, lock in self._locks.items():
            _send_obj({"kind": "reduce_max"}, rank, REQ_TAG, lock)
        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        return float(t.item())


def serve_engine(engine, orchestrator_rank: int = 0) -> None:
    """Rank>0 main loop: host this rank's models until 'stop' arrives.

    The LocalEngine's own thread continuous-batches everything in flight;
    this loop only moves requests/replies across the control plane.
    """
    engine.start()
    send_lock = threading.Lock()

    def _complete(rid: str, result) -> None:
        _send_obj({"kind": "rep", "rid": rid, "result": result},
                  orchestrator_rank, REP_TAG, send_lock)

    import time
    barrier_times: List[float] = []
    while True:
        msg = _recv_obj(orchestrator_rank, REQ_TAG)
        if msg is None or msg.get("kind") == "stop":
            break
        if msg.get("kind") == "barrier":
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            dist.barrier()
            barrier_times.append(time.perf_counter())
            continue
        if msg.get("kind") == "embed_gather":
            from .vote import all_gather_vote, compute_local_embeddings
            local = compute_local_embeddings(engine, msg["texts"])
            all_gather_vote(local, msg["counts"])
            continue
        if msg.get("kind") == "reduce_max":
            elapsed = (barrier_times[-1] - barrier_times[-2]
                       if len(barrier_times) >= 2 else 0.0)
            t = torch.tensor([elapsed], dtype=torch.float64)
            dist.all_reduce(t, op=dist.ReduceOp.MAX)
            continue
        if msg.get("kind") == "gen":
            rid = msg["rid"]
            err = engine._submit(msg["request"], ("cb",
                                                  lambda r, rid=rid: _complete(rid, r)))
            if err is not None:
                _complete(rid, err)
    engine.stop()
    try:
        _send_obj({"kind": "bye"}, orchestrator_rank, REP_TAG, send_lock)
    except Exception:
        pass
