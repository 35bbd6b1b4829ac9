# LLM API Gateway (MI355X) — deployment image.
# Parity: the reference's multi-stage python-slim Dockerfile (+ the ROCm
# runtime needed for local engines). Build on a host with ROCm 7.x and the
# gfx950 extension prebuilt in-tree (setup.py build_ext --inplace), or use
# a rocm/pytorch base to build inside the image.
FROM rocm/pytorch:latest AS base

WORKDIR /app

COPY requirements.txt .
RUN pip install --no-cache-dir -r requirements.txt

COPY llmapigateway_amd/ llmapigateway_amd/
COPY static/ static/
COPY setup.py main.py ./

# compile the gfx950 HIP extension in-tree (cross-compiles without a GPU)
RUN PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

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
