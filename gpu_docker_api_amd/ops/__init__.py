"""Native ops: HIP/CDNA4 kernels and the io_uring copy engine.

Built in-tree by :mod:`.build` (the .so files live next to this file so they
travel to GPU boxes with the repo snapshot). On a GPU box these MUST load —
ops fail loudly rather than falling back to a silent eager path.
"""
