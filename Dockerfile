# ROCm runtime image for the MI355X-native trader (reference parity:
# Dockerfile — python-slim + CMD run_trader; here the base must carry the
# ROCm 7.x runtime and the gfx950 extension is built at image build time).
FROM rocm/pytorch:rocm7.0_ubuntu22.04_py3.10_pytorch_2.10
WORKDIR /app
# netcat for the per-service nc -z healthchecks
# (docker-compose.distributed.yml)
RUN apt-get update && apt-get install -y --no-install-recommends \
    netcat-openbsd && rm -rf /var/lib/apt/lists/*
COPY . /app
ENV PYTORCH_ROCM_ARCH=gfx950 \
    HSA_ENABLE_IPC_MODE_LEGACY=0
RUN python -m ai_crypto_trader_amd.ops.build
EXPOSE 8050
CMD ["python", "run_trader.py", "--minutes", "0", "--speed", "1"]
