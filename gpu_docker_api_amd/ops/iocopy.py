"""Python facade over the _iocopy extension (io_uring copy engine)."""
from __future__ import annotations

import importlib.util
import os
from typing import Optional

_ext = None
_ext_err: Optional[str] = None


def _load():
    global _ext, _ext_err
    if _ext is not None:
        return _ext
    if _ext_err is not None:
        raise ImportError(_ext_err)
    so = os.path.join(os.path.dirname(os.path.abspath(__file__)), "_iocopy.so")
    if not os.path.exists(so):
        _ext_err = f"{so} not built (run ops.build)"
        raise ImportError(_ext_err)
    spec = importlib.util.spec_from_file_location("_iocopy", so)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    _ext = mod
    return mod


def copy_tree(src: str, dst: str) -> dict:
    return _load().copy_tree(src, dst)


def uring_available() -> bool:
    try:
        return bool(_load().uring_available())
    except ImportError:
        return False
