from .engine import EngineStats, SweepEngine  # noqa: F401
from .fastpath import FastSweep, WinnerRecord  # noqa: F401
from .snapshot import CellSnapshot, build_cell_snapshot, compute_batch_size  # noqa: F401
