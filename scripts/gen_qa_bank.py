#!/usr/bin/env python3
"""Generate the qa-benchmark grove's question bank.

The reference ships mmlu-pro (12,032 downloaded questions) and livebench
as quality harnesses (reference: priv/groves/, README.md:548-551).  This
environment has no network, so the local bank is procedurally generated:
objectively-scoreable multiple-choice questions (10 options A-J,
mmlu-pro style) across six subjects, deterministic under a fixed seed so
the bank is reproducible and the grove's accuracy scoring is exact.

Usage: python scripts/gen_qa_bank.py [n_per_subject] > questions.json
"""

import json
import random
import sys

LETTERS = "ABCDEFGHIJ"


def _options(rng, correct, pool):
    """10 options containing the correct value once."""
    wrong = []
    seen = {str(correct)}
    for cand in pool:
        if str(cand) not in seen:
            wrong.append(cand)
            seen.add(str(cand))
        if len(wrong) >= 9:
            break
    while len(wrong) < 9:           # top up when the pool ran short
        cand = rng.randint(0, 999)
        if str(cand) not in seen:
            wrong.append(cand)
            seen.add(str(cand))
    opts = wrong[:9] + [correct]
    rng.shuffle(opts)
    return [str(o) for o in opts], LETTERS[opts.index(correct)]


def q_arithmetic(rng):
    a, b, c = rng.randint(2, 60), rng.randint(2, 30), rng.randint(2, 12)
    val = a + b * c
    opts, ans = _options(rng, val, [val + d for d in
                                    (-c, c, b, -b, 1, -1, 10, -10, a, 2 * c)])
    return {"q": f"What is {a} + {b} * {c}?", "options": opts, "answer": ans}


def q_sequence(rng):
    start, step = rng.randint(1, 20), rng.randint(2, 9)
    seq = [start + i * step for i in range(4)]
    val = start + 4 * step
    opts, ans = _options(rng, val, [val + d for d in
                                    (-step, step, 1, -1, 2, -2, step + 1,
                                     -step - 1, 3, -3)])
    return {"q": f"What number comes next: {', '.join(map(str, seq))}, ...?",
            "options": opts, "answer": ans}


def q_logic(rng):
    # three named propositions keep the question space large enough for a
    # unique 100-question bank
    names = rng.sample(["P", "Q", "R", "S", "T", "U", "V", "W"], 3)
    vals = [rng.choice([True, False]) for _ in range(3)]
    op1, op2 = rng.sample(["AND", "OR", "XOR"], 2)

    def apply(op, a, b):
        return {"AND": a and b, "OR": a or b, "XOR": a != b}[op]

    val = apply(op2, apply(op1, vals[0], vals[1]), vals[2])
    correct = "true" if val else "false"
    distract = ["false" if val else "true", "undefined", "both",
                "neither", names[0], names[1], names[2],
                f"not {names[0]}", f"not {names[1]}", "invalid"]
    opts, ans = _options(rng, correct, distract)
    given = ", ".join(f"{n} is {str(v).lower()}"
                      for n, v in zip(names, vals))
    return {"q": f"If {given}, what is ({names[0]} {op1} {names[1]}) "
                 f"{op2} {names[2]}?", "options": opts, "answer": ans}


_NAMES = ["Ada", "Brahe", "Curie", "Darwin", "Euler", "Fermi", "Gauss",
          "Hopper", "Ising", "Joule"]
_ITEMS = ["lanterns", "maps", "coins", "books", "seeds", "tiles",
          "ropes", "flasks", "gears", "shells"]


def q_reading(rng):
    names = rng.sample(_NAMES, 3)
    counts = rng.sample(range(3, 40), 3)
    items = rng.sample(_ITEMS, 3)
    passage = ". ".join(f"{n} collected {c} {it}"
                        for n, c, it in zip(names, counts, items))
    pick = rng.randrange(3)
    val = counts[pick]
    opts, ans = _options(rng, val, counts + [v + 1 for v in counts]
                         + [v - 1 for v in counts] + [50, 2])
    return {"q": f"{passage}. How many {items[pick]} did {names[pick]} "
                 f"collect?", "options": opts, "answer": ans}


def q_comparison(rng):
    vals = rng.sample(range(100, 999), 5)
    kind, val = rng.choice([("largest", max(vals)), ("smallest", min(vals)),
                            ("median", sorted(vals)[2])])
    opts, ans = _options(rng, val, vals + [v + 1 for v in vals] + [500])
    return {"q": f"Which is the {kind} of {', '.join(map(str, vals))}?",
            "options": opts, "answer": ans}


def q_modular(rng):
    a, m = rng.randint(20, 400), rng.randint(3, 12)
    val = a % m
    opts, ans = _options(rng, val, list(range(0, m)) + [val + m, m, a // m])
    return {"q": f"What is {a} mod {m}?", "options": opts, "answer": ans}


SUBJECTS = {
    "arithmetic": q_arithmetic,
    "sequences": q_sequence,
    "logic": q_logic,
    "reading": q_reading,
    "comparison": q_comparison,
    "modular": q_modular,
}


def generate(n_per_subject: int = 100, seed: int = 20260914):
    rng = random.Random(seed)
    bank = {}
    for subject, gen in SUBJECTS.items():
        seen, out = set(), []
        while len(out) < n_per_subject:
            item = gen(rng)
            if item["q"] in seen:
                continue
            assert item["answer"] in LETTERS
            assert len(item["options"]) == 10
            seen.add(item["q"])
            out.append(item)
        bank[subject] = out
    return bank


if __name__ == "__main__":
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 100
    json.dump(generate(n), sys.stdout, indent=1)
