# dts_amd server image — ROCm base with PyTorch-ROCm; builds the gfx950
# extension at image build time (parity: reference 2-stage Dockerfile, but
# the "frontend build" stage is replaced by the kernel build).
FROM rocm/pytorch:latest

WORKDIR /app
COPY . /app

ENV PYTORCH_ROCM_ARCH=gfx950
RUN python -m dts_amd.ops.build && python -m dts_amd.core.build

EXPOSE 8000
HEALTHCHECK --interval=30s --timeout=5s CMD curl -sf http://localhost:8000/health || exit 1
CMD ["python", "-m", "dts_amd.server", "--host", "0.0.0.0", "--port", "8000"]
