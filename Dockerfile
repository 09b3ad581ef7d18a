# Controller image: ROCm + PyTorch base with the in-tree HIP kernel library
# built for gfx950 at image build time (hipcc cross-compiles without a GPU).
FROM rocm/pytorch:rocm7.0_ubuntu22.04_py3.10_pytorch_release_2.10.0

WORKDIR /app
COPY inferno_amd/ inferno_amd/
COPY deploy/ deploy/
COPY bench.py .

ENV PYTORCH_ROCM_ARCH=gfx950 \
    HSA_ENABLE_IPC_MODE_LEGACY=0 \
    PYTHONPATH=/app

RUN pip install --no-cache-dir fastapi uvicorn httpx prometheus_client pyyaml \
    && python -m inferno_amd.ops.build

USER 1001
ENTRYPOINT ["python", "-m", "inferno_amd.controller.main"]
