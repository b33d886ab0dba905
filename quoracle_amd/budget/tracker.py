"""Budget tracking, escrow and enforcement.

Behavior-parity with the reference (reference: lib/quoracle/budget/tracker.ex:
6,42-149, escrow.ex:1-60, enforcer.ex): available = allocated − spent −
committed; statuses ok / warning (≤20% left) / over_budget / na; parent→child
escrow locks the child's allocation in the parent's committed column; budget
decreases must not go below what the child has already spent+committed.
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Optional


class BudgetError(Exception):
    def __init__(self, reason: str):
        super().__init__(reason)
        self.reason = reason


@dataclass
class BudgetView:
    mode: str                  # "root" | "allocated" | "na"
    allocated: Optional[float]
    spent: float
    committed: float

    @property
    def available(self) -> Optional[float]:
        if self.allocated is None:
            return None
        return self.allocated - self.spent - self.committed

    @property
    def status(self) -> str:
        if self.mode == "na":
            return "na"
        if self.allocated is None:
            return "ok"  # unlimited
        avail = self.available
        if avail < 0:
            return "over_budget"
        if self.allocated > 0 and avail / self.allocated <= 0.2:
            return "warning"
        return "ok"


def parse_amount(value) -> float:
    """Budget strings are positive decimals ('50.00')."""
    try:
        amount = float(value)
    except (TypeError, ValueError):
        raise BudgetError("invalid_amount") from None
    # NaN/inf would poison every downstream escrow comparison
    if not math.isfinite(amount) or amount <= 0:
        raise BudgetError("invalid_amount")
    return amount


def check_can_spend(view: BudgetView, amount: float = 0.0) -> None:
    """Pre-action enforcement: raises over_budget when exhausted."""
    if view.mode == "na" or view.allocated is None:
        return
    if (view.available or 0.0) - amount < 0:
        raise BudgetError("over_budget")


def lock_allocation(view: BudgetView, amount: float) -> float:
    """Escrow `amount` for a child: returns the parent's new committed total.
    Raises when the parent lacks available budget."""
    if view.allocated is not None:
        if (view.available or 0.0) < amount:
            raise BudgetError("insufficient_budget")
    return view.committed + amount


def release_allocation(committed: float, amount: float) -> float:
    return max(0.0, committed - amount)


def validate_decrease(new_budget: float, child_spent: float,
                      child_committed: float) -> None:
    """A child's budget cannot drop below what it already used or escrowed."""
    if new_budget < child_spent + child_committed:
        raise BudgetError("budget_below_usage")
