"""Deterministic byte-level tokenizer + chat template.

There is no network for real BPE vocab files, so the engine uses a
byte-level tokenizer: ids 0-255 are raw UTF-8 bytes, specials follow.  The
model's vocab (e.g. 128256 for llama3-8b) stays full-size — the lm_head /
embedding GEMMs keep their real shapes — the tokenizer just never *produces*
ids above the byte+special range, and decoding folds out-of-range sampled
ids back into printable bytes.

Token counts from this tokenizer are exact for the hosted models, replacing
the reference's tiktoken estimation NIF (reference: agent/token_manager.ex:19-24).
"""

from __future__ import annotations

from typing import Dict, List, Sequence

BYTE_VOCAB = 256
BOS = 256
EOS = 257
ROLE_SYSTEM = 258
ROLE_USER = 259
ROLE_ASSISTANT = 260
EOT = 261            # end of turn
N_SPECIAL = 6
VOCAB_FLOOR = BYTE_VOCAB + N_SPECIAL

_ROLE_IDS: Dict[str, int] = {
    "system": ROLE_SYSTEM,
    "user": ROLE_USER,
    "assistant": ROLE_ASSISTANT,
}

# ids sampled above the produced range decode to a printable byte so random
# weight models still emit valid UTF-8
_PRINTABLE = [ord(c) for c in
              "abcdefghijklmnopqrstuvwxyz ABCDEFGHIJKLMNOPQRSTUVWXYZ"
              "0123456789.,:;!?()[]{}\"'"]


class ByteTokenizer:
    def __init__(self, vocab_size: int = VOCAB_FLOOR):
        if vocab_size < VOCAB_FLOOR:
            raise ValueError(f"vocab_size must be >= {VOCAB_FLOOR}")
        self.vocab_size = vocab_size

    def encode(self, text: str) -> List[int]:
        return list(text.encode("utf-8", errors="replace"))

    def decode(self, ids: Sequence[int]) -> str:
        out = bytearray()
        for i in ids:
            if i < BYTE_VOCAB:
                out.append(i)
            elif i < VOCAB_FLOOR:
                continue          # specials render as nothing
            else:
                out.append(_PRINTABLE[i % len(_PRINTABLE)])
        return out.decode("utf-8", errors="replace")

    def count(self, text: str) -> int:
        return len(text.encode("utf-8", errors="replace"))

    # -- chat template -------------------------------------------------------

    def render_messages(self, messages: List[Dict[str, str]]) -> List[int]:
        """[BOS] ( [ROLE] bytes [EOT] )* [ROLE_ASSISTANT]  — generation
        continues after the trailing assistant marker."""
        ids: List[int] = [BOS]
        for msg in messages:
            ids.append(_ROLE_IDS.get(msg.get("role", "user"), ROLE_USER))
            ids.extend(self.encode(msg.get("content", "")))
            ids.append(EOT)
        ids.append(ROLE_ASSISTANT)
        return ids

    def count_messages(self, messages: List[Dict[str, str]]) -> int:
        return 2 + sum(2 + self.count(m.get("content", "")) for m in messages)
