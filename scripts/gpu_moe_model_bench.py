"""Mixtral-8x7B on one MI355X through the real engine: agent-bench style
short run proving the MoE architecture (capacity-padded sync-free FFN,
graph-captured decode) serves on hardware.  Also a TP=1 TPEngine pass —
the config-5 engine mechanism on a real GPU."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from quoracle_amd.engine.api import GenerateRequest
from quoracle_amd.engine.engine import LocalEngine

dev = torch.device("cuda:0")
t0 = time.perf_counter()
eng = LocalEngine(["mixtral-8x7b#0"], device=dev, kv_gb_per_model=8.0,
                  embed_model_key=None).start()
print(f"mixtral up in {time.perf_counter()-t0:.1f}s", flush=True)
outs = 0
t0 = time.perf_counter()
for i in range(3):
    r = eng.generate_sync(GenerateRequest(
        model_key="mixtral-8x7b#0",
        messages=[{"role": "user", "content": f"decide {i}: " + "ctx " * 500}],
        temperature=0.8, max_tokens=256, seed=i, action_grammar=True,
        session_id=f"moe{i}"), timeout=600)
    assert r.ok, r.error
    outs += r.output_tokens
el = time.perf_counter() - t0
print(f"mixtral-8x7b: 3 grammar generations, {outs} output tokens in "
      f"{el:.1f}s ({outs/el:.1f} tok/s incl. prefill)  stats={eng.stats}")
eng.stop()

# TP=1 TPEngine on GPU (RCCL PG world 1): the lockstep engine mechanism
import torch.distributed as dist
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29719")
dist.init_process_group("nccl", rank=0, world_size=1)
from quoracle_amd.engine.tp_engine import TPEngine
from quoracle_amd.parallel.tp import TPContext
tpe = TPEngine(["llama3-8b#tp"], TPContext(0, 1), device=dev,
               kv_gb_per_model=4.0, embed_model_key=None).start()
r = tpe.generate_sync(GenerateRequest(
    model_key="llama3-8b#tp",
    messages=[{"role": "user", "content": "tp check"}],
    temperature=0.7, max_tokens=128, seed=3, action_grammar=True,
    session_id="tp-gpu"), timeout=300)
assert r.ok, r.error
print(f"TPEngine(world=1) on GPU: ok, {r.output_tokens} tokens "
      f"in {r.latency_ms:.0f}ms")
tpe.stop()
dist.destroy_process_group()
