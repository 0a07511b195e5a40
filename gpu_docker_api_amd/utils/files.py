"""Filesystem helpers (reference: utils/file.go:13-59)."""
from __future__ import annotations

import os


def dir_size(path: str) -> int:
    """Total bytes under ``path`` (the volume shrink-safety check,
    reference utils/file.go:13-22). Uses st_blocks*512 when available so
    sparse files are measured by allocation, not apparent size."""
    total = 0
    for root, _dirs, files in os.walk(path):
        for f in files:
            fp = os.path.join(root, f)
            try:
                st = os.lstat(fp)
            except OSError:
                continue
            blocks = getattr(st, "st_blocks", None)
            total += blocks * 512 if blocks is not None else st.st_size
    return total


def is_dir(path: str) -> bool:
    return os.path.isdir(path)
