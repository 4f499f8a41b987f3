# opsagent-amd — MI355X-native Kubernetes ops agent.
# Mirrors the reference's 2-stage image (ref /root/reference/Dockerfile:1-54:
# runtime gets python3 + jq + kubectl + a kubernetes-SDK venv) on a ROCm base
# with the gfx950 kernels built at image build time (hipcc cross-compiles
# without a GPU).

FROM rocm/pytorch:latest AS build
WORKDIR /app
COPY opsagent_amd ./opsagent_amd
COPY configs ./configs
COPY __graft_entry__.py bench.py ./
ENV PYTORCH_ROCM_ARCH=gfx950
RUN python -m opsagent_amd.ops.build --force

FROM rocm/pytorch:latest
WORKDIR /app

# tools the agent shells out to (ref Dockerfile:24-38)
RUN apt-get update && apt-get install -y --no-install-recommends jq curl ca-certificates \
    && curl -fsSL -o /usr/local/bin/kubectl \
       "https://dl.k8s.io/release/$(curl -fsSL https://dl.k8s.io/release/stable.txt)/bin/linux/amd64/kubectl" \
    && chmod +x /usr/local/bin/kubectl \
    && curl -fsSL https://raw.githubusercontent.com/aquasecurity/trivy/main/contrib/install.sh \
       | sh -s -- -b /usr/local/bin \
    && pip install --no-cache-dir kubernetes pyyaml pandas fastapi uvicorn typer rich safetensors \
    && rm -rf /var/lib/apt/lists/*

COPY --from=build /app /app
ENV PYTHONPATH=/app
ENV HSA_ENABLE_IPC_MODE_LEGACY=0

EXPOSE 8080
ENTRYPOINT ["python", "-m", "opsagent_amd.cli"]
CMD ["serve", "--port", "8080"]
