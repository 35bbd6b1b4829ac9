# LLM API Gateway (MI355X) — deployment image.
# Multi-stage (parity with the reference's python-slim multi-stage build,
# /root/reference/Dockerfile:1-93, on a ROCm base for the local engines):
# the builder compiles the gfx950 HIP extension in-tree (hipcc
# cross-compiles without a GPU); the runtime stage carries only the app,
# the built extension and the runtime deps.

FROM rocm/pytorch:latest AS builder

WORKDIR /build
COPY requirements.txt setup.py ./
COPY llmapigateway_amd/ llmapigateway_amd/
RUN PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

FROM rocm/pytorch:latest AS runtime

WORKDIR /app
COPY requirements.txt .
RUN pip install --no-cache-dir -r requirements.txt

# app + the prebuilt extension from the builder stage
COPY --from=builder /build/llmapigateway_amd/ llmapigateway_amd/
COPY static/ static/
COPY main.py ./

COPY docker/entrypoint.sh /entrypoint.sh
COPY docker/healthcheck.py /healthcheck.py
RUN chmod +x /entrypoint.sh

# non-root user; GPU access needs the video/render groups
RUN useradd -m -u 1000 gateway && \
    usermod -aG video gateway 2>/dev/null || true && \
    mkdir -p /app/db /app/logs && chown -R gateway /app
USER gateway

ENV GATEWAY_PORT=9100
EXPOSE 9100

HEALTHCHECK --interval=30s --timeout=10s --retries=3 \
    CMD python /healthcheck.py || exit 1

ENTRYPOINT ["/entrypoint.sh"]
