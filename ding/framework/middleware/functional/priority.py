"""Priority middleware. Parity: reference functional/priority.py
(priority_calculator)."""
from typing import Callable

from ...context import OnlineRLContext


def priority_calculator(priority_calculation_fn: Callable) -> Callable:
    """Annotate freshly collected trajectories with initial priorities."""

    def _calculate(ctx: OnlineRLContext):
        if ctx.trajectories is not None:
            priorities = priority_calculation_fn(ctx.trajectories)
            for t, p in zip(ctx.trajectories, priorities):
                t['priority'] = float(p)

    return _calculate
