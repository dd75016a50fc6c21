# GridLLM-AMD — one image for every process type (bus / server / gateway /
# worker). Reference analog: server/Dockerfile + client/Dockerfile
# (multi-stage node:24-alpine); here the base is the ROCm PyTorch image and
# the build step compiles the gfx950 HIP kernel library in-tree.
#
# Build:  docker build -t gridllm-amd .
# The image needs no GPU to BUILD (hipcc cross-compiles gfx950); running a
# worker needs /dev/kfd + /dev/dri mapped (see docker-compose.yml).

FROM rocm/pytorch:rocm7.2_ubuntu22.04_py3.10_pytorch

WORKDIR /app

# Python deps are part of the ROCm image (torch); the control plane needs
# only stdlib + fastapi/uvicorn/msgpack/aiohttp/prometheus_client.
RUN pip install --no-cache-dir fastapi uvicorn msgpack aiohttp \
    prometheus_client httpx

COPY csrc/ csrc/
COPY gridllm_amd/ gridllm_amd/
COPY scripts/ scripts/
COPY pytest.ini README.md ./

# compile the CDNA4 kernel library for gfx950 (no GPU required)
ENV PYTORCH_ROCM_ARCH=gfx950
RUN python -m gridllm_amd.ops.build

# dmabuf IPC for multi-process GPU work (RCCL / tensor sharing)
ENV HSA_ENABLE_IPC_MODE_LEGACY=0

# non-root, like the reference images
RUN useradd -m gridllm && chown -R gridllm /app
USER gridllm

EXPOSE 4000 46379 3000

# default command = server; compose overrides per service
CMD ["python", "-m", "gridllm_amd.server"]
