# grove-amd operator / agent image (MI355X-native).
#
# Build context = repo root. The ROCm PyTorch base ships hipcc + rocm_smi +
# PyTorch-ROCm; the native extensions (_sched C++ placement core, _topo rocm_smi/KFD
# topology probe, _gpuwork CDNA4 MFMA kernels) are compiled for gfx950 at build time
# so the image needs no toolchain at runtime.
#
#   docker build -t grove-amd:latest .
#   kubectl apply -k deploy/
FROM rocm/pytorch:rocm7.0_ubuntu22.04_py3.10_pytorch_release_2.10.0

ENV PYTHONUNBUFFERED=1 \
    PYTORCH_ROCM_ARCH=gfx950 \
    HSA_ENABLE_IPC_MODE_LEGACY=0

WORKDIR /opt/grove

# dependency layer (everything needed is in the base image or stdlib; fastapi +
# uvicorn serve the kube-style apiserver)
RUN pip install --no-cache-dir fastapi uvicorn pyyaml

COPY grove_amd/ grove_amd/
COPY crds/ crds/
COPY samples/ samples/

# build the native extensions in-tree for gfx950 (hipcc cross-compiles without a GPU)
RUN python -c "from grove_amd.ops.build import build_all; build_all(force=True)"

# non-root runtime (rocm_smi needs the video/render groups on GPU nodes)
RUN useradd -r -u 10001 -G video,render grove || useradd -r -u 10001 grove
USER 10001

ENTRYPOINT ["python", "-m", "grove_amd"]
CMD ["operator", "--config-file", "/etc/grove/config.yaml"]
