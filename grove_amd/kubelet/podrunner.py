"""Pod runner — the container entrypoint for process-backed pods.

When the ProcessKubelet (kubelet/process.py) runs a pod as a real OS process, this
module is the "container image": it receives the injected GROVE_* env contract
(api/common/constants parity), runs grove-initc against the HTTP apiserver when the
pod's clique has startup dependencies, executes the payload on the assigned GPU
(HIP_VISIBLE_DEVICES set by the kubelet), and exits 0 on success — exactly the
lifecycle a real inference container sees under the reference stack.
"""
from __future__ import annotations

import json
import os
import sys


def main() -> int:
    pcs = os.environ.get("GROVE_PCS_NAME", "?")
    pclq = os.environ.get("GROVE_PCLQ_NAME", "?")
    idx = os.environ.get("GROVE_PCLQ_POD_INDEX", "?")
    server = os.environ.get("GROVE_API_SERVER")
    podgang = os.environ.get("GROVE_PODGANG_NAME")
    parents = os.environ.get("GROVE_STARTS_AFTER", "")  # "fqn:min,fqn:min"
    namespace = os.environ.get("GROVE_NAMESPACE", "default")

    print(f"podrunner: {pclq}/{idx} of {pcs} starting", flush=True)

    if parents and server and podgang:
        from ..initc import wait_for_parents, parse_podcliques
        specs = parse_podcliques(parents.split(","))
        ok = wait_for_parents(namespace, podgang, specs, server=server,
                              timeout=float(os.environ.get("GROVE_INITC_TIMEOUT",
                                                           "120")),
                              poll=0.1)
        if not ok:
            print("podrunner: initc timed out", file=sys.stderr, flush=True)
            return 1
        print("podrunner: startup dependencies satisfied", flush=True)

    payload = os.environ.get("GROVE_PAYLOAD", "gemm")
    shape = os.environ.get("GROVE_PAYLOAD_SHAPE", "512x512x512x1")
    if payload != "none":
        from .gpunode import run_payload_descriptor
        dims = tuple(int(x) for x in shape.split("x"))
        if len(dims) == 3:
            dims = dims + (1,)
        # HIP_VISIBLE_DEVICES is set by the kubelet → device 0 is OUR gpu
        metrics = run_payload_descriptor("gemm" if payload == "gemm" else payload,
                                         dims, 0)
        print("podrunner: payload done", json.dumps(metrics), flush=True)
    return 0


if __name__ == "__main__":
    sys.exit(main())
