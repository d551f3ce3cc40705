"""Spending policy stub (parity: reference client/routing/spending_policy.py —
the reference's point system is explicitly NOT IMPLEMENTED there either)."""

from abc import ABC, abstractmethod


class SpendingPolicyBase(ABC):
    @abstractmethod
    def get_points(self, protocol: str, *args, **kwargs) -> float: ...


class NoSpendingPolicy(SpendingPolicyBase):
    def get_points(self, protocol: str, *args, **kwargs) -> float:
        return 0.0
