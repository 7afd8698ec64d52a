"""Priority-points economy stub (parity: reference
client/routing/spending_policy.py:1-17 — NoSpendingPolicy returns 0; the
interface exists so schedulers can price requests later)."""
from __future__ import annotations

from abc import ABC, abstractmethod


class SpendingPolicyBase(ABC):
    @abstractmethod
    def get_points(self, protocol: str, *args, **kwargs) -> float:
        ...


class NoSpendingPolicy(SpendingPolicyBase):
    def get_points(self, protocol: str, *args, **kwargs) -> float:
        return 0.0
