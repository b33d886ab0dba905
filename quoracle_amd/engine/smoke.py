"""Driver smoke entry: one constrained decode of the flagship model on GPU."""

from __future__ import annotations

import json

import torch


def run_smoke(model_key: str = "llama3-8b") -> dict:
    assert torch.cuda.is_available(), "smoke needs a GPU"
    from .api import GenerateRequest
    from .engine import LocalEngine

    engine = LocalEngine([model_key], kv_gb_per_model=1.0)
    req = GenerateRequest(
        model_key=model_key,
        messages=[{"role": "system", "content": "You are a consensus agent."},
                  {"role": "user", "content": "Assess the task."}],
        temperature=0.8, max_tokens=1024, seed=7,
        action_grammar=True, session_id="smoke")
    result = engine.generate_sync(req, timeout=600)
    assert result.ok, f"smoke generate failed: {result.error}"
    parsed = json.loads(result.text)
    assert parsed["action"] in {"orient", "send_message", "todo", "wait"}
    torch.cuda.synchronize()
    print(f"[smoke] ok: action={parsed['action']} "
          f"in={result.input_tokens} out={result.output_tokens} "
          f"latency_ms={result.latency_ms:.1f}")
    return parsed
