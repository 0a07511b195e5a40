"""MI355X-native GPU container control plane.

A from-scratch rebuild of the capabilities of XShengTech/gpu-docker-api
 for ROCm / 8xMI355X nodes: etcd-semantics MVCC state,
topology-aware GPU scheduling, rolling-replace container lifecycle, and a
native HIP/C++ layer (xGMI bandwidth probe, MFMA warm-up, RCCL smoke test,
io_uring copy engine).
"""

__version__ = "0.1.0"
