"""quoracle-amd CLI.

  python -m quoracle_amd serve --models llama3-8b#0,llama3-8b#1,llama3-8b#2
      Boot the full stack on this GPU: local engine, task runtime (with
      boot-time revival of running tasks, reference: boot/agent_revival.ex)
      and the web monitor on http://HOST:PORT.
  python -m quoracle_amd show-prompts [scenario]
      Render verbatim LLM prompts (reference: mix quoracle.show_llm_prompts).
"""

from __future__ import annotations

import argparse
import sys


def cmd_serve(args) -> None:
    import torch
    import uvicorn

    from .agent.supervisor import Supervisor
    from .engine.engine import LocalEngine
    from .engine.pool import EnginePool
    from .governance.profiles import Profile
    from .persistence.store import Store
    from .tasks.manager import TaskManager
    from .tasks.runtime import RuntimeConfig, TaskRuntime
    from .ui.server import create_app

    device = torch.device(args.device) if args.device else None
    keys = [k for k in args.models.split(",") if k]
    if args.fake:
        from .engine.fake import FakeEngine
        engine = FakeEngine()
    else:
        engine = LocalEngine(keys, device=device,
                             kv_gb_per_model=args.kv_gb).start()
    pool = EnginePool(default=engine, embedder=engine)
    for k in keys:
        pool.assign(k, engine)
    store = Store(args.db)
    runtime = TaskRuntime(engines=pool, store=store, config=RuntimeConfig(
        groves_dir=args.groves_dir, skills_dir=args.skills_dir))
    Supervisor(runtime)
    if not runtime.profiles.exists("default"):
        runtime.profiles.put(Profile(
            name="default", description="default pool",
            model_pool=keys or ["fake-a", "fake-b"],
            capability_groups=["hierarchy", "local_execution", "file_read",
                               "file_write", "external_api"]))
    manager = TaskManager(runtime)
    app = create_app(manager)

    @app.on_event("startup")
    async def _revive():
        # boot-time restoration of tasks left running
        result = await manager.restore_running_tasks()
        if result.get("restored"):
            print(f"[serve] revived tasks: {result}", file=sys.stderr)

    print(f"[serve] monitor on http://{args.host}:{args.port} "
          f"(models: {keys or 'fake'})", file=sys.stderr)
    uvicorn.run(app, host=args.host, port=args.port, log_level="warning")


def main() -> None:
    p = argparse.ArgumentParser(prog="quoracle_amd")
    sub = p.add_subparsers(dest="cmd", required=True)

    s = sub.add_parser("serve", help="engine + runtime + web monitor")
    s.add_argument("--models", default="llama3-8b#0,llama3-8b#1,llama3-8b#2")
    s.add_argument("--db", default="quoracle.db")
    s.add_argument("--host", default="127.0.0.1")
    s.add_argument("--port", type=int, default=8800)
    s.add_argument("--kv-gb", type=float, default=8.0)
    s.add_argument("--device", default=None)
    s.add_argument("--groves-dir", default=None)
    s.add_argument("--skills-dir", default=None)
    s.add_argument("--fake", action="store_true",
                   help="FakeEngine backend (no GPU; orchestrator demo)")
    s.set_defaults(fn=cmd_serve)

    sp = sub.add_parser("show-prompts", help="render verbatim LLM prompts")
    sp.add_argument("scenario", nargs="?", default="all")
    sp.set_defaults(fn=lambda a: __import__(
        "quoracle_amd.tools.show_prompts", fromlist=["main"]).main(
            [a.scenario]))

    stk = sub.add_parser("show-task", help="render a task transcript")
    stk.add_argument("db")
    stk.add_argument("task_id", nargs="?")
    stk.set_defaults(fn=lambda a: __import__(
        "quoracle_amd.tools.show_task", fromlist=["main"]).main(
            [a.db] + ([a.task_id] if a.task_id else [])))

    args = p.parse_args()
    args.fn(args)


if __name__ == "__main__":
    main()
