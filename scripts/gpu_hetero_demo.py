"""Heterogeneous consensus on one MI355X: a llama3-8b + mixtral-8x7b
pool (two DIFFERENT architectures, 141 GB of weights together) hosted by
one engine, driving a REAL agent decision end to end — task API,
constrained decoding on both models, fingerprint clustering across
architectures, action execution.  The single-box flavor of BASELINE
config 4's heterogeneous pool."""
import asyncio
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch


async def main():
    from quoracle_amd.agent.supervisor import Supervisor
    from quoracle_amd.engine.engine import LocalEngine
    from quoracle_amd.engine.pool import EnginePool
    from quoracle_amd.governance.profiles import Profile, ProfileStore
    from quoracle_amd.persistence.store import Store
    from quoracle_amd.tasks.manager import TaskManager
    from quoracle_amd.tasks.runtime import RuntimeConfig, TaskRuntime

    t0 = time.perf_counter()
    pool_keys = ["llama3-8b#h0", "mixtral-8x7b#h0"]
    engine = LocalEngine(pool_keys, device=torch.device("cuda:0"),
                         kv_gb_per_model=8.0).start()
    free, total = torch.cuda.mem_get_info()
    print(f"hetero pool up in {time.perf_counter()-t0:.1f}s "
          f"({(total-free)/(1<<30):.0f} GiB HBM in use)", flush=True)

    pool = EnginePool(default=engine)
    store = Store(":memory:")
    profiles = ProfileStore(store)
    profiles.put(Profile(name="hetero", description="two architectures",
                         model_pool=pool_keys,
                         capability_groups=["file_read"],
                         max_refinement_rounds=3))
    runtime = TaskRuntime(store=store, engines=pool, profiles=profiles,
                          config=RuntimeConfig())
    Supervisor(runtime)
    manager = TaskManager(runtime)

    t0 = time.perf_counter()
    result = await manager.create_task(
        "Orient yourself and report the situation.", "hetero")
    root = runtime.registry.lookup(result["root_agent_id"]).actor
    for _ in range(1200):
        if root.steps_completed >= 1:
            break
        await asyncio.sleep(0.1)
    el = time.perf_counter() - t0
    assert root.steps_completed >= 1, "no decision completed"
    decisions = [e for m in pool_keys
                 for e in root.state.model_histories[m]
                 if e.get("type") == "decision"]
    print(f"heterogeneous consensus decision in {el:.1f}s: "
          f"action={decisions[0]['content'].get('action')}  "
          f"stats={engine.stats}")
    await manager.supervisor.terminate_tree(root.state.agent_id)
    engine.stop()


if __name__ == "__main__":
    asyncio.run(main())
