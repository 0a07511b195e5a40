"""Bulk data movers for rolling replacement.

The reference migrates a container's writable layer with a shell tar pipe
(``(cd src; tar c .) | (cd dest; tar x)`` — /root/reference/utils/copy.go:17-27)
and migrates volumes by spinning up a throwaway ubuntu container that runs
``mv`` inside, fire-and-forget (utils/copy.go:74-128). Here:

* the preferred engine is the native io_uring copier (csrc/iocopy.cpp →
  ops._iocopy): queued reads/writes, sparse-aware, preserves
  mode/mtime/symlinks/xattrs (overlayfs whiteouts are device nodes + xattrs,
  which it replicates);
* the fallback is a shell-free tar pipe (two connected subprocesses with
  --xattrs) — still awaited, never fire-and-forget;
* volume migration is a plain host-side move (the daemon can see both
  mountpoints; no helper container needed), also awaited.
"""
from __future__ import annotations

import asyncio
import logging
import os
import shutil
import subprocess
from typing import Optional

log = logging.getLogger(__name__)

_iocopy = None
_iocopy_err: Optional[str] = None


def _load_iocopy():
    global _iocopy, _iocopy_err
    if _iocopy is not None or _iocopy_err is not None:
        return _iocopy
    try:
        from ..ops import iocopy as mod  # noqa: WPS433

        _iocopy = mod
    except Exception as exc:  # extension not built
        _iocopy_err = str(exc)
        log.info("native iocopy unavailable (%s); using tar fallback", exc)
    return _iocopy


def _tar_supports(flag: str) -> bool:
    try:
        out = subprocess.run(["tar", "--help"], capture_output=True, text=True, timeout=10)
        return flag in out.stdout
    except Exception:
        return False


_TAR_XATTRS: Optional[bool] = None


async def _tar_pipe_copy(src: str, dest: str) -> None:
    """tar -C src -c . | tar -C dest -x  (no shell, awaited, xattrs kept)."""
    global _TAR_XATTRS
    if _TAR_XATTRS is None:
        _TAR_XATTRS = _tar_supports("--xattrs")
    extra = ["--xattrs", "--acls"] if _TAR_XATTRS else []
    r_fd, w_fd = os.pipe()
    try:
        reader = await asyncio.create_subprocess_exec(
            "tar", "-C", src, *extra, "-cf", "-", ".", stdout=w_fd
        )
    finally:
        os.close(w_fd)
    try:
        writer = await asyncio.create_subprocess_exec(
            "tar",
            "-C",
            dest,
            *(extra + ["--xattrs-include=*"] if extra else []),
            "-xf",
            "-",
            stdin=r_fd,
        )
    finally:
        os.close(r_fd)
    rc_w = await writer.wait()
    rc_r = await reader.wait()
    if rc_r != 0 or rc_w != 0:
        raise RuntimeError(f"tar pipe copy {src} -> {dest} failed (tar rc {rc_r}/{rc_w})")


class CopyEngine:
    """Engine selection: 'auto' (io_uring if built, else tar), 'iouring',
    'tar', 'python' (shutil; tests)."""

    def __init__(self, kind: str = "auto") -> None:
        self.kind = kind

    # trees at or below this size are copied inline: engine setup (io_uring
    # ring init / tar fork+exec / thread hop) costs ~1-2 ms, which dominates
    # the rolling-replace latency when the writable layer is near-empty
    SMALL_TREE_BYTES = 1 << 20
    SMALL_TREE_FILES = 32

    @classmethod
    def _small_tree_entries(cls, src: str):
        """Top-level regular files/symlinks totaling <= SMALL_TREE_BYTES,
        or None when the tree needs a real engine (subdirs/large/special)."""
        entries = []
        total = 0
        try:
            with os.scandir(src) as it:
                for e in it:
                    if len(entries) >= cls.SMALL_TREE_FILES:
                        return None
                    if e.is_symlink():
                        entries.append(e)
                        continue
                    if not e.is_file(follow_symlinks=False):
                        return None
                    total += e.stat(follow_symlinks=False).st_size
                    if total > cls.SMALL_TREE_BYTES:
                        return None
                    entries.append(e)
        except OSError:
            return None
        return entries

    async def copy_dir(self, src: str, dest: str) -> None:
        if not os.path.isdir(src):
            raise FileNotFoundError(src)
        os.makedirs(dest, exist_ok=True)
        small = self._small_tree_entries(src)
        if small is not None:
            for e in small:
                target = os.path.join(dest, e.name)
                if e.is_symlink():
                    link = os.readlink(e.path)
                    if os.path.lexists(target):
                        os.unlink(target)
                    os.symlink(link, target)
                else:
                    shutil.copy2(e.path, target)
            return
        kind = self.kind
        if kind == "auto":
            kind = "iouring" if _load_iocopy() is not None else "tar"
        if kind == "iouring":
            mod = _load_iocopy()
            if mod is None:
                raise RuntimeError(f"io_uring engine requested but unavailable: {_iocopy_err}")
            # run in a thread: the native engine blocks its calling thread
            await asyncio.get_running_loop().run_in_executor(
                None, mod.copy_tree, src, dest
            )
        elif kind == "tar":
            await _tar_pipe_copy(src, dest)
        elif kind == "python":
            await asyncio.get_running_loop().run_in_executor(
                None,
                lambda: shutil.copytree(src, dest, symlinks=True, dirs_exist_ok=True),
            )
        else:
            raise ValueError(f"unknown copy engine {kind!r}")

    async def move_contents(self, src: str, dest: str) -> None:
        """Move everything under src into dest (volume resize migration).
        Same-filesystem moves are renames; cross-fs falls back to copy+rm."""
        if not os.path.isdir(src):
            raise FileNotFoundError(src)
        os.makedirs(dest, exist_ok=True)

        def _move() -> None:
            for entry in os.listdir(src):
                shutil.move(os.path.join(src, entry), os.path.join(dest, entry))

        await asyncio.get_running_loop().run_in_executor(None, _move)


async def copy_tree(src: str, dest: str, engine: str = "auto") -> None:
    await CopyEngine(engine).copy_dir(src, dest)


async def move_tree_contents(src: str, dest: str) -> None:
    await CopyEngine().move_contents(src, dest)
