from .greedy import (  # noqa: F401
    ServerEntry,
    allocate_equally,
    allocate_maximally,
    make_priority_groups,
    solve_greedy,
)
from .solver import Manager, Optimizer, Solver  # noqa: F401
