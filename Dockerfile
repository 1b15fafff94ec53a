# Worker / consumer image (reference parity: Dockerfile.crowdllama —
# but ROCm-native: the runtime needs the HIP stack and a gfx950 GPU).
# Build args stamp the version the same way the reference injects ldflags
# (Dockerfile.crowdllama:28-32); crowdllama_amd.version picks them up from
# the environment at import.
FROM rocm/dev-ubuntu-22.04:6.4-complete

ARG VERSION=dev
ARG COMMIT=unknown
ENV CROWDLLAMA_VERSION=${VERSION} \
    CROWDLLAMA_COMMIT=${COMMIT} \
    PYTORCH_ROCM_ARCH=gfx950

WORKDIR /opt/crowdllama-amd
COPY pyproject.toml setup.py ./
COPY crowdllama_amd ./crowdllama_amd
COPY bench.py ./

RUN pip install --no-cache-dir numpy aiohttp pybind11 && \
    python3 crowdllama_amd/ops/build.py

EXPOSE 9001 14001
ENTRYPOINT ["python3", "-m", "crowdllama_amd.cli"]
CMD ["start", "--worker-mode"]
