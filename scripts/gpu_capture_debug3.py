"""Bisect the dual-preset capture failure: which op class fails?"""
import sys
import traceback

sys.path.insert(0, "/root/repo")
import torch

from quoracle_amd.engine.engine import LocalEngine

dev = torch.device("cuda:0")
eng = LocalEngine(["mixtral-8x7b#h0", "llama3-8b#h0"], device=dev,
                  kv_gb_per_model=8.0)
mix = eng.models["mixtral-8x7b#h0"]
lla = eng.models["llama3-8b#h0"]


def try_capture(name, fn):
    s = torch.cuda.Stream(dev)
    s.wait_stream(torch.cuda.current_stream(dev))
    with torch.cuda.stream(s):
        for _ in range(2):
            fn()
    torch.cuda.current_stream(dev).wait_stream(s)
    torch.cuda.synchronize(dev)
    g = torch.cuda.CUDAGraph()
    try:
        with torch.cuda.graph(g, capture_error_mode="thread_local"):
            fn()
        print(f"{name}: CAPTURE OK")
        return True
    except Exception as exc:  # noqa: BLE001
        print(f"{name}: FAILED — {str(exc)[:90]}")
        return False


# 1. bare mixtral-shaped GEMM (hipBLASLt)
h = torch.randn(1, 4096, device=dev, dtype=torch.bfloat16)
w = torch.randn(4096, 6144, device=dev, dtype=torch.bfloat16)
try_capture("bare matmul (mixtral qkv shape)", lambda: h @ w)

# 2. the python MoE bmm path ops alone
layer = mix.model.layers[0]
try_capture("moe_ffn_bmm python path",
            lambda: mix.model._moe_ffn(h, layer))

# 3. our extension ops alone (rmsnorm + decode attention shapes)
from quoracle_amd import ops
y = torch.empty_like(h)
wn = torch.ones(4096, device=dev, dtype=torch.bfloat16)
try_capture("rmsnorm kernel", lambda: ops.ext().rmsnorm_fused(
    y, h, None, wn, 1e-5))

# 4. llama capture FIRST (no mixtral capture attempted)
lla.graphs.allow_capture = True
ok = lla.graphs._capture(1) is not None
print(f"llama capture first in dual engine: {'OK' if ok else 'FAILED'}")

# 5. then mixtral
mix.graphs.allow_capture = True
ok = mix.graphs._capture(1) is not None
print(f"mixtral capture after llama: {'OK' if ok else 'FAILED'}")
