"""Token sampling + constrained action-JSON decoding.

With random-init weights (no network for checkpoints) a free-running decode
emits byte noise that would never parse as an action, so the engine supports
grammar-constrained decoding: the sampler walks a per-sequence template of
the action JSON the consensus parser expects (parser.py / actions/schema.py),
forcing structural tokens and letting the model's logits drive the free
parts (reasoning text, param values, the action CHOICE itself).  Every
generated token is still a full model forward — the constraint only masks
what is *emitted*, so benchmark compute is identical to real decoding.

Choice semantics: at the action-choice step the sampler draws one token from
the model's (temperature-scaled) distribution over printable bytes and maps
it onto a candidate action — model- and temperature-dependent, so consensus
across decorrelated pool members genuinely disagrees at high temperature and
converges as the refinement schedule cools (reference behavior:
lib/quoracle/consensus/temperature.ex).
"""

from __future__ import annotations

import json
from dataclasses import dataclass
from typing import Any, Dict, List, Optional, Sequence, Tuple

import torch

from .tokenizer import EOS

# bytes allowed in free-text spans inside JSON strings (no '"' or '\\')
_SAFE_TEXT = [ord(c) for c in
              "abcdefghijklmnopqrstuvwxyz ABCDEFGHIJKLMNOPQRSTUVWXYZ"
              "0123456789.,:;!?()- "]
_SAFE_SET = frozenset(_SAFE_TEXT)


@dataclass
class SamplingParams:
    temperature: float = 1.0
    top_p: float = 1.0
    max_tokens: int = 256
    seed: Optional[int] = None
    # None = free-running decode; list = grammar-constrained to these actions
    allowed_actions: Optional[List[str]] = None


def _safe_mask(vocab_size: int, device) -> torch.Tensor:
    mask = torch.full((vocab_size,), float("-inf"), device=device)
    mask[torch.tensor(_SAFE_TEXT, device=device)] = 0.0
    return mask


class _MaskCache:
    _cache: Dict[Tuple[int, str], torch.Tensor] = {}

    @classmethod
    def get(cls, vocab_size: int, device) -> torch.Tensor:
        key = (vocab_size, str(device))
        if key not in cls._cache:
            cls._cache[key] = _safe_mask(vocab_size, device)
        return cls._cache[key]


# -- grammar -----------------------------------------------------------------

# ops: ("forced", id) | ("free",) | ("choice",)
FORCED, FREE, CHOICE = "forced", "free", "choice"

# Parameter plans: how to fill each action's required params.
#   ("free", n)  -> JSON string with n model-sampled tokens
#   ("lit", x)   -> json literal
#   ("ctx", key, default) -> JSON string literal from the grammar context
_PARAM_PLANS: Dict[str, List[Tuple[str, Any]]] = {
    "orient": [("current_situation", ("free", 24)),
               ("goal_clarity", ("free", 12)),
               ("available_resources", ("free", 12)),
               ("key_challenges", ("free", 16)),
               ("delegation_consideration", ("free", 12))],
    "send_message": [("to", ("lit", "parent")),
                     ("content", ("free", 32))],
    "todo": [("items", ("todo_items", 2))],
    "wait": [("wait", ("lit", True))],
    "spawn_child": [("task_description", ("free", 24)),
                    ("success_criteria", ("free", 12)),
                    ("immediate_context", ("free", 12)),
                    ("approach_guidance", ("free", 12)),
                    ("profile", ("ctx", "spawn_profile", "default"))],
    "file_read": [("path", ("ctx", "file_read_path", "/tmp/notes.txt"))],
}


def _encode(text: str) -> List[int]:
    return list(text.encode("utf-8"))


class ActionGrammar:
    """Per-sequence state machine emitting the next-token constraint."""

    def __init__(self, allowed_actions: Sequence[str],
                 reasoning_tokens: int = 24,
                 context: Optional[Dict[str, Any]] = None):
        self.context = context or {}
        self.candidates = [a for a in allowed_actions if a in _PARAM_PLANS] \
            or ["wait"]
        self._ops: List[Tuple] = []
        self._pos = 0
        self.done = False
        self._emit_forced('{"reasoning": "')
        for _ in range(reasoning_tokens):
            self._ops.append((FREE,))
        self._emit_forced('", "action": "')
        self._ops.append((CHOICE,))
        # remainder of the template is appended when the choice resolves

    def _emit_forced(self, text: str) -> None:
        for b in _encode(text):
            self._ops.append((FORCED, b))

    def _emit_params(self, action: str) -> None:
        self._emit_forced('", "params": {')
        plans = _PARAM_PLANS[action]
        for i, (name, plan) in enumerate(plans):
            if i:
                self._emit_forced(", ")
            self._emit_forced(json.dumps(name) + ": ")
            kind = plan[0]
            if kind == "free":
                self._emit_forced('"')
                for _ in range(plan[1]):
                    self._ops.append((FREE,))
                self._emit_forced('"')
            elif kind == "lit":
                self._emit_forced(json.dumps(plan[1]))
            elif kind == "ctx":
                self._emit_forced(json.dumps(
                    self.context.get(plan[1], plan[2])))
            elif kind == "todo_items":
                self._emit_forced('[')
                for j in range(plan[1]):
                    if j:
                        self._emit_forced(', ')
                    self._emit_forced('{"content": "')
                    for _ in range(10):
                        self._ops.append((FREE,))
                    self._emit_forced('", "state": "todo"}')
                self._emit_forced(']')
        self._emit_forced('}, "wait": false}')
        self._ops.append((FORCED, EOS))

    def current(self) -> Tuple:
        if self._pos >= len(self._ops):
            return (FORCED, EOS)
        return self._ops[self._pos]

    def advance(self, sampled_id: int) -> int:
        """Given the raw sampled id for this step, return the id actually
        emitted (and fed back to the model next step)."""
        op = self.current()
        self._pos += 1
        if op[0] == FORCED:
            emitted = op[1]
            if emitted == EOS and self._pos >= len(self._ops):
                self.done = True
            return emitted
        if op[0] == FREE:
            # defensive: the engine masks sampling to _SAFE_TEXT, but any
            # out-of-set id (misbehaving driver, replayed stream) must
            # still yield valid JSON
            if sampled_id in _SAFE_SET:
                return sampled_id
            return _SAFE_TEXT[sampled_id % len(_SAFE_TEXT)]
        # CHOICE: map the model's draw onto a candidate action
        action = self.candidates[sampled_id % len(self.candidates)]
        name_bytes = _encode(action)
        emitted = name_bytes[0]
        # force the rest of the name, then the param template
        rest = self._ops[self._pos:]
        self._ops = self._ops[:self._pos]
        for b in name_bytes[1:]:
            self._ops.append((FORCED, b))
        self._emit_params(action)
        self._ops.extend(rest)       # (normally empty)
        return emitted


class Sampler:
    """Batched sampling over mixed constrained/unconstrained sequences."""

    def __init__(self, vocab_size: int, device):
        self.vocab_size = vocab_size
        self.device = device

    def sample(self, logits: torch.Tensor, params: List[SamplingParams],
               grammars: List[Optional[ActionGrammar]],
               generators: List[Optional[torch.Generator]]) -> List[int]:
        """logits: [R, V] fp32 — one row per sequence needing a token.
        Returns emitted token ids (grammar-adjusted)."""
        R = logits.shape[0]
        safe = _MaskCache.get(self.vocab_size, logits.device)
        # Forced-token rows need no GPU work; the rest get ONE batched
        # masked/temperature softmax, then a per-row multinomial (per-row
        # generators keep per-request determinism) and ONE host sync.
        emitted: List[Optional[int]] = [None] * R
        masked_rows: List[int] = []     # grammar FREE/CHOICE: printable only
        plain_rows: List[int] = []      # unconstrained decode
        for r in range(R):
            g = grammars[r]
            if g is not None:
                if g.current()[0] == FORCED:
                    emitted[r] = g.advance(0)
                else:
                    masked_rows.append(r)
            else:
                plain_rows.append(r)

        pending: List[Tuple[int, torch.Tensor]] = []

        def _batch(rows: List[int], mask: Optional[torch.Tensor]) -> None:
            if not rows:
                return
            temps = torch.tensor(
                [max(params[r].temperature, 1e-4) for r in rows],
                dtype=logits.dtype, device=logits.device).unsqueeze(1)
            sub = logits[rows]
            if mask is not None:
                sub = sub + mask
            probs = torch.softmax(sub / temps, dim=-1)
            for i, r in enumerate(rows):
                row_probs = probs[i]
                if params[r].top_p < 1.0:
                    row_probs = _top_p_filter(row_probs, params[r].top_p)
                pending.append((r, torch.multinomial(
                    row_probs, 1, generator=generators[r])))

        _batch(masked_rows, safe)
        _batch(plain_rows, None)
        if pending:
            ids = torch.cat([t for _, t in pending]).cpu()   # single sync
            for (r, _), idx in zip(pending, ids.tolist()):
                g = grammars[r]
                emitted[r] = g.advance(idx) if g is not None else idx
        return emitted


def _top_p_filter(probs: torch.Tensor, top_p: float) -> torch.Tensor:
    sorted_probs, order = torch.sort(probs, descending=True)
    cum = torch.cumsum(sorted_probs, dim=-1)
    keep = cum - sorted_probs < top_p
    keep[0] = True
    filtered = torch.zeros_like(probs)
    filtered[order[keep]] = probs[order[keep]]
    return filtered / filtered.sum()
