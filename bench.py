#!/usr/bin/env python3
"""Flagship benchmark: consensus agent-steps/sec on an N-GPU model pool.

Measures the BASELINE.json metric — consensus agent-steps per second (and
p50 step latency) with a 3-model Llama-3-8B pool — through the REAL
pipeline: every agent step runs per-model constrained decoding on the local
HIP/CDNA4 engine (random-init weights, synthetic prompts), fingerprint
clustering, winner/merge selection and action execution.

Scaling is weak: each GPU hosts its own 3-model pool shard and a fixed
number of agents; rank 0 orchestrates, ranks 1..N-1 serve their models over
the gloo control plane (one process per GPU, torchrun).

One bench "step" is a fleet round: every agent completes exactly one
consensus agent-step (all pool models decode, votes clustered, action
executed).  value = agents * steps / elapsed = whole-job agent-steps/sec.
"""

from __future__ import annotations

import argparse
import asyncio
import json
import os
import statistics
import sys
import time


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=3)
    p.add_argument("--warmup", type=int, default=1)
    p.add_argument("--model", default="llama3-8b")
    p.add_argument("--pool-size", type=int, default=3)
    p.add_argument("--agents-per-gpu", type=int, default=8)
    p.add_argument("--rounds", type=int, default=4,
                   help="max refinement rounds (reference default 4)")
    p.add_argument("--kv-gb", type=float, default=24.0)
    p.add_argument("--device", default=None, help="override (e.g. cpu)")
    p.add_argument("--pool-scope", choices=["shard", "global"],
                   default="shard",
                   help="shard: each agent votes over its own rank's models "
                        "(weak scaling). global: ONE pool spanning every "
                        "rank, one model per GPU (BASELINE config 4; use "
                        "--model a+b+... for a heterogeneous pool)")
    p.add_argument("--scenario", choices=["flat", "tree"], default="flat",
                   help="flat: independent agents (weak scaling); tree: "
                        "depth-2 recursive spawn tree per shard "
                        "(BASELINE config 3)")
    p.add_argument("--fanout", type=int, default=4)
    p.add_argument("--tp", type=int, default=1,
                   help="tensor-parallel degree: the whole job is ONE "
                        "lockstep TP group hosting a '+'-joined model pool "
                        "(BASELINE config 5: --tp 4 --model "
                        "llama3-70b+mixtral-8x7b)")
    return p.parse_args()


def rank_model_keys(model: str, pool_size: int, rank: int):
    """Shard mode: rank r hosts pool_size instances of the preset.
    '+'-joined presets round-robin over the instances (heterogeneous)."""
    presets = model.split("+")
    return [f"{presets[j % len(presets)]}#r{rank}m{j}"
            for j in range(pool_size)]


T_START = time.perf_counter()


def main():
    args = parse_args()
    import torch
    print(f"[bench t={time.perf_counter() - T_START:.1f}s] torch imported",
          file=sys.stderr, flush=True)

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if args.gpus > 1 and world == 1:
        print(f"[bench] WARNING: --gpus {args.gpus} but WORLD_SIZE=1 — "
              f"multi-GPU runs must be launched via torchrun "
              f"(--nproc-per-node {args.gpus}); continuing single-process",
              file=sys.stderr, flush=True)
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    use_cuda = torch.cuda.is_available() and args.device != "cpu"
    if use_cuda:
        torch.cuda.set_device(local_rank)
        device = torch.device(f"cuda:{local_rank}")
    else:
        device = torch.device(args.device or "cpu")

    distributed = world > 1
    if distributed:
        import torch.distributed as dist
        backend = "cpu:gloo,cuda:nccl" if use_cuda else "gloo"
        dist.init_process_group(backend=backend)

    from quoracle_amd.engine.engine import LocalEngine

    if args.tp > 1:
        if world != args.tp:
            raise SystemExit(f"--tp {args.tp} needs WORLD_SIZE={args.tp}")
        from quoracle_amd.engine.tp_engine import TPEngine, serve_tp_replica
        from quoracle_amd.parallel.tp import TPContext
        tp = TPContext(rank, world)
        keys = [f"{m}#tp{i}" for i, m in enumerate(args.model.split("+"))]
        engine = TPEngine(keys, tp, device=device,
                          kv_gb_per_model=args.kv_gb,
                          embed_model_key="embed-small" if rank == 0 else None)
        print(f"[bench t={time.perf_counter() - T_START:.1f}s] rank {rank}: "
              f"TP{world} engine up, pool {keys}", file=sys.stderr, flush=True)
        if rank != 0:
            serve_tp_replica(engine)
            return
        result = asyncio.run(orchestrate(args, engine, device, 1,
                                         pool_keys=keys,
                                         parallelism=f"tp{world}"))
        result["n_gpus"] = world
        print(json.dumps(result), flush=True)
        return

    if args.pool_scope == "global":
        # config 4: one model per GPU, all of them in ONE consensus pool
        presets = args.model.split("+")
        local_keys = [f"{presets[rank % len(presets)]}#r{rank}"]
    else:
        local_keys = rank_model_keys(args.model, args.pool_size, rank)
    engine = LocalEngine(
        local_keys, device=device, kv_gb_per_model=args.kv_gb,
        embed_model_key="embed-small")   # every rank joins the RCCL vote
    print(f"[bench t={time.perf_counter() - T_START:.1f}s] rank {rank}: "
          f"engine up, models {local_keys} on {device}",
          file=sys.stderr, flush=True)

    if rank != 0:
        from quoracle_amd.parallel.control import serve_engine
        serve_engine(engine)
        return

    result = asyncio.run(orchestrate(args, engine, device, world))
    print(json.dumps(result), flush=True)


async def orchestrate(args, engine, device, world, pool_keys=None,
                      parallelism=None):
    import torch
    from quoracle_amd.agent.core import AgentActor, MESSAGE_TYPES
    from quoracle_amd.agent.state import AgentState
    from quoracle_amd.agent.supervisor import Supervisor
    from quoracle_amd.engine.pool import EnginePool
    from quoracle_amd.governance.profiles import Profile, VALID_GROUPS
    from quoracle_amd.tasks.runtime import RuntimeConfig, TaskRuntime
    from quoracle_amd.utils import ids

    engine.start()
    client = None
    pool = EnginePool(embedder=engine)
    if world > 1:
        from quoracle_amd.parallel.control import ControlClient, RemoteEngine
        from quoracle_amd.parallel.vote import DistributedEmbedder
        client = ControlClient(list(range(1, world)))
        # consensus vote embeddings: spread over all ranks, merged by ONE
        # RCCL all-gather over xGMI
        pool = EnginePool(embedder=DistributedEmbedder(engine, client, world))
        pool._embedder_engine = engine
    presets = args.model.split("+")

    def keys_for_rank(r):
        if args.pool_scope == "global":
            return [f"{presets[r % len(presets)]}#r{r}"]
        return rank_model_keys(args.model, args.pool_size, r)

    for r in range(world):
        for key in keys_for_rank(r):
            if r == 0:
                pool.assign(key, engine)
            else:
                from quoracle_amd.parallel.control import RemoteEngine
                pool.assign(key, RemoteEngine(r, client))

    runtime = TaskRuntime(engines=pool, config=RuntimeConfig())
    Supervisor(runtime)
    global_pool = [k for r in range(world) for k in keys_for_rank(r)]
    for r in range(world):
        runtime.profiles.put(Profile(
            name=f"bench-r{r}", description="bench pool shard",
            model_pool=pool_keys or (
                global_pool if args.pool_scope == "global"
                else keys_for_rank(r)),
            capability_groups=list(VALID_GROUPS),
            max_refinement_rounds=args.rounds))
    if pool_keys:
        for key in pool_keys:
            pool.assign(key, engine)

    # Build the agent fleet, lockstep-driven (the actor loop is not
    # started — the bench owns cycle timing).
    actors = []

    def make_agent(name, shard, parent_id, prompt):
        profile = runtime.profiles.resolve(f"bench-r{shard}")
        state = AgentState(
            agent_id=ids.agent_id(name),
            task_id=f"bench-task-{shard}", parent_id=parent_id,
            profile=f"bench-r{shard}",
            model_pool=list(profile.model_pool),
            capability_groups=list(profile.capability_groups),
            max_refinement_rounds=profile.max_refinement_rounds,
        )
        state.init_model_maps()
        actor = AgentActor(state, runtime)
        runtime.registry.register(state.agent_id, actor, state.task_id,
                                  parent_id=parent_id)
        actor.state.message_queue.append({"type": "user_message",
                                          "content": prompt})
        actors.append(actor)
        return actor

    if args.scenario == "tree":
        # depth-2 recursive tree per shard (BASELINE config 3): the bench
        # steps EVERY agent of the tree each round; parent<->child messages
        # flow through the real send_message executor
        for shard in range(world):
            root = make_agent(f"root{shard}", shard, None,
                              f"Coordinate benchmark tree {shard}.")
            for c in range(args.fanout):
                child = make_agent(
                    f"c{shard}-{c}", shard, root.state.agent_id,
                    f"Subtask {c}: analyze and report to your parent.")
                root.state.children[child.state.agent_id] = {
                    "status": "running", "task_description": f"subtask {c}"}
                for g in range(args.fanout):
                    gc = make_agent(
                        f"g{shard}-{c}-{g}", shard, child.state.agent_id,
                        f"Leaf task {c}.{g}: gather one detail and report.")
                    child.state.children[gc.state.agent_id] = {
                        "status": "running",
                        "task_description": f"leaf {c}.{g}"}
    else:
        for i in range(args.agents_per_gpu * world):
            make_agent(f"bench{i}", i % world, None,
                       f"Benchmark task {i}: assess the situation, plan the "
                       f"work, and coordinate results. Iteration seed {i}.")

    step_latencies = []

    async def one_round(actor, timed):
        t0 = time.perf_counter()
        while not actor.inbox.empty():
            msg = actor.inbox.get_nowait()
            t = msg.get("type")
            if t == "action_result":
                actor._handle_action_result(msg)
            elif t in MESSAGE_TYPES:
                actor.state.message_queue.append(msg)
        await actor._run_cycle()
        if timed:
            step_latencies.append((time.perf_counter() - t0) * 1e3)

    async def fleet_round(timed=False):
        await asyncio.gather(*[one_round(a, timed) for a in actors])
        # let dispatched action tasks deliver their results
        for _ in range(4):
            await asyncio.sleep(0)

    def barrier():
        if world > 1:
            client.barrier_all()
        elif torch.cuda.is_available():
            torch.cuda.synchronize()

    def note(msg):
        print(f"[bench t={time.perf_counter() - T_START:.1f}s] {msg}",
              file=sys.stderr, flush=True)

    note(f"agents built: {len(actors)}; engine stats {engine.stats}")
    for i in range(args.warmup):
        tw = time.perf_counter()
        await fleet_round()
        note(f"warmup round {i}: {time.perf_counter() - tw:.2f}s "
             f"stats={engine.stats}")
    barrier()
    t0 = time.perf_counter()
    for i in range(args.steps):
        tw = time.perf_counter()
        await fleet_round(timed=True)
        note(f"timed round {i}: {time.perf_counter() - tw:.2f}s")
    barrier()
    elapsed = time.perf_counter() - t0
    note(f"timed region done: {elapsed:.2f}s stats={engine.stats}")
    if world > 1:
        elapsed = client.reduce_max_elapsed(elapsed)
        client.shutdown()
    engine.stop()

    n_agents = len(actors)
    agent_steps = n_agents * args.steps
    decisions = sum(a.steps_completed for a in actors)
    return {
        "metric": "consensus agent-steps/sec",
        "value": round(agent_steps / elapsed, 3),
        "unit": "agent-steps/sec",
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(elapsed * 1e3 / args.steps, 2),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16",
        "data": "synthetic",
        "config": {
            "model": args.model,
            "pool_size": args.pool_size,
            "agents_per_gpu": args.agents_per_gpu,
            "scenario": args.scenario,
            "pool_scope": args.pool_scope,
            "fanout": args.fanout if args.scenario == "tree" else None,
            "agents_total": n_agents,
            "parallelism": parallelism or f"pool-sharded dp{world}",
            "p50_step_latency_ms": round(
                statistics.median(step_latencies), 2) if step_latencies else None,
            "decisions_completed": decisions,
            "constrained_decoding": True,
            "max_refinement_rounds": args.rounds,
            "engine_stats": dict(engine.stats),
        },
    }


if __name__ == "__main__":
    main()
