from .files import dir_size, is_dir
from .copy import CopyEngine, copy_tree, move_tree_contents
from .timing import PhaseTimer, Metrics, METRICS

__all__ = [
    "dir_size",
    "is_dir",
    "CopyEngine",
    "copy_tree",
    "move_tree_contents",
    "PhaseTimer",
    "Metrics",
    "METRICS",
]
