from .allocation import (  # noqa: F401
    Allocation,
    AllocationDiff,
    allocation_from_data,
    create_allocation,
    create_allocation_diff,
    reallocate,
    scale_allocation,
)
from .system import (  # noqa: F401
    Accelerator,
    AllocationByType,
    Model,
    Server,
    ServiceClass,
    System,
    Target,
)
