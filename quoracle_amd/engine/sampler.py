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

# ops: ("forced", id) | ("free",) | ("choice",) | ("digit",) | ("word",)
#      | ("enum", values) | ("bchoice", candidates)
FORCED, FREE, CHOICE = "forced", "free", "choice"
DIGIT, WORD, ENUM, BCHOICE = "digit", "word", "enum", "bchoice"

_WORD_CHARS = "abcdefghijklmnopqrstuvwxyz0123456789_"

# Parameter plans cover ALL 22 actions, derived from the schema registry
# (actions/schema.py) with per-param overrides for values that must come
# from runtime context (spawn profile, live child ids, confinement-safe
# paths).  Plan kinds:
#   ("free", n)   -> JSON string, n model-sampled printable tokens
#   ("word", n)   -> JSON string, n tokens from [a-z0-9_] (identifiers)
#   ("digits", n) -> bare number of n model-sampled digits
#   ("lit", x)    -> json literal
#   ("ctx", key, default) -> JSON string literal from the grammar context
#   ("prefix", text, n)   -> JSON string: forced prefix + n free tokens
#   ("enum", [..]) -> model-chosen enum value (temperature-dependent)
#   ("list_free", k, n)   -> JSON list of k free strings of n tokens
#   ("todo_items", k)     -> todo list items
#   ("batch", k)  -> k batchable sub-action specs, each a model CHOICE
_FREE_LONG = {"content", "prompt", "task_description", "current_situation",
              "description", "query", "command"}


def _string_plan(action: str, name: str) -> Tuple:
    n = 24 if name in _FREE_LONG else 12
    return ("free", n)


# (action, param) -> plan overrides
_PARAM_OVERRIDES: Dict[Tuple[str, str], Tuple] = {
    ("send_message", "to"): ("lit", "parent"),
    ("wait", "wait"): ("lit", True),
    ("spawn_child", "profile"): ("ctx", "spawn_profile", "default"),
    ("file_read", "path"): ("ctx", "file_read_path", "/tmp/notes.txt"),
    ("file_write", "path"): ("ctx", "file_write_path", "/tmp/scratch.txt"),
    ("file_write", "mode"): ("lit", "write"),
    ("dismiss_child", "child_id"): ("ctx", "child_id", "no-child"),
    ("adjust_budget", "child_id"): ("ctx", "child_id", "no-child"),
    ("adjust_budget", "new_budget"): ("digit_string", 2),
    ("record_cost", "amount"): ("digit_string", 2),
    ("execute_shell", "command"): ("prefix", "echo ", 8),
    ("generate_secret", "name"): ("word", 8),
    ("create_skill", "name"): ("word", 8),
    ("fetch_web", "url"): ("ctx", "fetch_url", "http://127.0.0.1:9/unreachable"),
    ("call_api", "url"): ("ctx", "api_url", "http://127.0.0.1:9/unreachable"),
    ("call_api", "api_type"): ("lit", "rest"),
    ("call_api", "method"): ("lit", "GET"),
    ("call_mcp", "transport"): ("lit", "stdio"),
    ("call_mcp", "command"): ("ctx", "mcp_command", "false"),
    ("learn_skills", "skills"): ("list_word", 1, 8),
    ("search_secrets", "search_terms"): ("list_word", 1, 6),
    ("todo", "items"): ("todo_items", 2),
    ("batch_sync", "actions"): ("batch", 2),
    ("batch_async", "actions"): ("batch", 2),
}

# xor'd optionals the grammar must still template (schema xor group choice)
_XOR_PICKS: Dict[str, List[str]] = {
    "execute_shell": ["command"],
    "file_write": ["content"],
    "call_mcp": ["transport", "command"],
    "call_api": ["method"],       # api_type=rest needs a verb
}


def _type_plan(action: str, name: str, ptype: Any) -> Optional[Tuple]:
    if isinstance(ptype, tuple):
        kind = ptype[0]
        if kind == "enum":
            return ("enum", list(ptype[1]))
        if kind == "list":
            if ptype[1] == "string":
                return ("list_free", 1, 8)
            return None          # spec lists handled via overrides
        if kind in ("union", "map_shape"):
            return None
    if ptype == "string":
        return _string_plan(action, name)
    if ptype in ("integer", "number"):
        return ("digits", 2)
    if ptype == "boolean":
        return ("lit", False)
    if ptype == "map":
        return ("lit", {})
    return None


def _build_param_plans() -> Dict[str, List[Tuple[str, Any]]]:
    from ..actions import schema as schema_mod
    plans: Dict[str, List[Tuple[str, Any]]] = {}
    for action in schema_mod.ACTIONS:
        s = schema_mod.get_schema(action)
        names = list(s.required_params) + [
            p for p in _XOR_PICKS.get(action, ())
            if p not in s.required_params]
        if action == "spawn_child" and "profile" not in names:
            names.append("profile")
        entries: List[Tuple[str, Any]] = []
        for name in names:
            plan = _PARAM_OVERRIDES.get((action, name))
            if plan is None:
                plan = _type_plan(action, name, s.param_types.get(name, "string"))
            if plan is None:
                plan = ("lit", "")
            entries.append((name, plan))
        plans[action] = entries
    return plans


_PARAM_PLANS: Dict[str, List[Tuple[str, Any]]] = _build_param_plans()

# simple batchable sub-actions the batch grammar can nest (full plans,
# no further nesting)
_BATCH_CANDIDATES = ["orient", "todo", "record_cost", "search_secrets",
                     "file_read", "send_message"]


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

    def _emit_plan_value(self, plan: Tuple) -> None:
        kind = plan[0]
        if kind == "free":
            self._emit_forced('"')
            for _ in range(plan[1]):
                self._ops.append((FREE,))
            self._emit_forced('"')
        elif kind == "word":
            self._emit_forced('"')
            for _ in range(plan[1]):
                self._ops.append((WORD,))
            self._emit_forced('"')
        elif kind == "digits":
            # leading digit 1-9 so the JSON number stays valid
            self._ops.append((DIGIT, True))
            for _ in range(plan[1] - 1):
                self._ops.append((DIGIT, False))
        elif kind == "digit_string":
            self._emit_forced('"')
            self._ops.append((DIGIT, True))
            for _ in range(plan[1] - 1):
                self._ops.append((DIGIT, False))
            self._emit_forced('"')
        elif kind == "lit":
            self._emit_forced(json.dumps(plan[1]))
        elif kind == "ctx":
            self._emit_forced(json.dumps(
                str(self.context.get(plan[1]) or plan[2])))
        elif kind == "prefix":
            self._emit_forced('"' + plan[1])
            for _ in range(plan[2]):
                self._ops.append((FREE,))
            self._emit_forced('"')
        elif kind == "enum":
            self._emit_forced('"')
            self._ops.append((ENUM, list(plan[1])))
            # closing quote emitted when the choice resolves
        elif kind in ("list_free", "list_word"):
            tok = FREE if kind == "list_free" else WORD
            self._emit_forced('[')
            for j in range(plan[1]):
                if j:
                    self._emit_forced(', ')
                self._emit_forced('"')
                for _ in range(plan[2]):
                    self._ops.append((tok,))
                self._emit_forced('"')
            self._emit_forced(']')
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
        elif kind == "batch":
            self._emit_forced('[')
            for j in range(plan[1]):
                if j:
                    self._emit_forced(', ')
                self._emit_forced('{"action": "')
                self._ops.append((BCHOICE,))
                # sub-action name tail + params + '}' inserted at resolution
            self._emit_forced(']')
        else:                        # pragma: no cover — unknown plan kind
            self._emit_forced('""')

    def _emit_param_entries(self, action: str) -> None:
        for i, (name, plan) in enumerate(_PARAM_PLANS[action]):
            if i:
                self._emit_forced(", ")
            self._emit_forced(json.dumps(name) + ": ")
            self._emit_plan_value(plan)

    def _emit_params(self, action: str) -> None:
        self._emit_forced('", "params": {')
        self._emit_param_entries(action)
        self._emit_forced('}, "wait": false}')
        self._ops.append((FORCED, EOS))

    def current(self) -> Tuple:
        if self._pos >= len(self._ops):
            return (FORCED, EOS)
        return self._ops[self._pos]

    def _splice(self, emit_fn) -> None:
        """Insert ops generated by emit_fn at the current position."""
        rest = self._ops[self._pos:]
        self._ops = self._ops[:self._pos]
        emit_fn()
        self._ops.extend(rest)

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
        if op[0] == WORD:
            ch = _WORD_CHARS[sampled_id % len(_WORD_CHARS)]
            return ord(ch)
        if op[0] == DIGIT:
            digits = "123456789" if op[1] else "0123456789"
            return ord(digits[sampled_id % len(digits)])
        if op[0] == ENUM:
            value = op[1][sampled_id % len(op[1])]
            emitted = ord(value[0])
            self._splice(lambda: self._emit_forced(value[1:] + '"'))
            return emitted
        if op[0] == BCHOICE:
            action = _BATCH_CANDIDATES[sampled_id % len(_BATCH_CANDIDATES)]
            emitted = ord(action[0])

            def _emit_sub():
                self._emit_forced(action[1:] + '", "params": {')
                self._emit_param_entries(action)
                self._emit_forced('}}')
            self._splice(_emit_sub)
            return emitted
        # CHOICE: map the model's draw onto a candidate action
        action = self.candidates[sampled_id % len(self.candidates)]
        name_bytes = _encode(action)
        emitted = name_bytes[0]

        def _emit_main():
            for b in name_bytes[1:]:
                self._ops.append((FORCED, b))
            self._emit_params(action)
        self._splice(_emit_main)
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
