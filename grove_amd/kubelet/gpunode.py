"""GPU node agent: runs scheduled pods' payloads on real MI355X GPUs.

The scheduler assigns each GPU pod device indices via the scheduling.amd.com/gpu-ids
annotation (the amdgpu-device-plugin analog); this agent executes the pod's payload on
that device using the native _gpuwork kernels (MFMA bf16 GEMM + HBM stream). On a GPU
box the native extension is REQUIRED — a missing .so raises instead of silently falling
back to eager PyTorch.

Payload spec comes from pod annotations:
  grove.io/payload       = "gemm" (default) | "stream" | "none"
  grove.io/payload-shape = "MxNxK[xiters]" for gemm, bytes for stream
"""
from __future__ import annotations

import logging
from typing import Any, Dict, Optional, Tuple

from ..api import constants as c
from ..kubecore.store import Obj

log = logging.getLogger("grove.gpunode")

PAYLOAD_ANNOTATION = "grove.io/payload"
PAYLOAD_SHAPE_ANNOTATION = "grove.io/payload-shape"
GPU_IDS_ANNOTATION = "scheduling.amd.com/gpu-ids"

_gpuwork = None


def load_gpuwork():
    """Import the native kernel extension; fail loudly when CUDA/ROCm is up."""
    global _gpuwork
    if _gpuwork is not None:
        return _gpuwork
    import torch
    try:
        from ..ops import _gpuwork as ext
    except ImportError as e:
        if torch.cuda.is_available():
            raise RuntimeError(
                "grove_amd.ops._gpuwork native extension is missing on a GPU box — "
                "build it with python -m grove_amd.ops.build") from e
        return None
    _gpuwork = ext
    return ext


def parse_payload(pod: Obj) -> Tuple[str, Tuple[int, ...]]:
    ann = pod["metadata"].get("annotations") or {}
    kind = ann.get(PAYLOAD_ANNOTATION, "gemm")
    shape = ann.get(PAYLOAD_SHAPE_ANNOTATION, "")
    if kind == "gemm":
        dims = tuple(int(x) for x in shape.split("x")) if shape else (1024, 1024, 1024, 1)
        if len(dims) == 3:
            dims = dims + (1,)
        return "gemm", dims
    if kind == "stream":
        return "stream", (int(shape) if shape else 1 << 24, 2)
    if kind == "decode":
        dims = tuple(int(v) for v in shape.split("x")) if shape else (8192, 8192, 4)
        if len(dims) == 2:
            dims = dims + (1,)
        return "decode", dims
    return "none", ()


def assigned_gpu(pod: Obj) -> Optional[int]:
    ids = (pod["metadata"].get("annotations") or {}).get(GPU_IDS_ANNOTATION, "")
    if not ids:
        return None
    return int(ids.split(",")[0])


def run_payload_descriptor(kind: str, dims: Tuple[int, ...], device: int) -> Dict[str, Any]:
    """Execute a payload on a local GPU device; returns metrics. Called by node-agent
    ranks (bench.py) and by the in-process GPU kubelet (smoke)."""
    import torch
    if not torch.cuda.is_available():
        # CPU development fallback: tiny matmul keeps the control flow identical
        if kind == "gemm":
            m, n, k, iters = dims
            a = torch.randn(min(m, 128), min(k, 128))
            b = torch.randn(min(k, 128), min(n, 128))
            for _ in range(iters):
                a @ b
        return {"device": "cpu", "kind": kind}
    ext = load_gpuwork()
    # logical device index -> physical (oversubscribed protocol rehearsals run
    # more ranks than GPUs; real N-GPU runs are 1:1)
    device = device % max(1, torch.cuda.device_count())
    with torch.cuda.device(device):
        if kind == "gemm":
            m, n, k, iters = dims
            tflops = ext.burn_gemm(m, n, k, iters)
            return {"device": device, "kind": kind, "tflops": tflops}
        if kind == "stream":
            n_floats = dims[0]
            iters = dims[1] if len(dims) > 1 else 2
            gbps = ext.stream_triad(n_floats, iters)
            return {"device": device, "kind": kind, "gbps": gbps}
        if kind == "decode":
            n, k, iters = dims
            gbps = ext.burn_decode(n, k, iters)
            return {"device": device, "kind": kind, "weight_stream_gbps": gbps}
    return {"device": device, "kind": kind}


def gpu_pod_payload(pod: Obj) -> None:
    """VirtualKubelet payload hook: run the pod's work on its assigned GPU."""
    gpu = assigned_gpu(pod)
    kind, dims = parse_payload(pod)
    if kind == "none":
        return
    if gpu is None:
        gpu = 0
    run_payload_descriptor(kind, dims, gpu)
