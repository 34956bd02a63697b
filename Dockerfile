# kubetorch-amd image: one image for controller, data-store and worker pods
# (the chart's `image.repository`). Base is PyTorch-ROCm for gfx950; the
# HIP extensions are compiled at build time (hipcc cross-compiles without a
# GPU, so this works on CPU-only CI builders).
FROM rocm/pytorch:rocm7.2_ubuntu22.04_py3.10_pytorch_2.10

ENV PYTORCH_ROCM_ARCH=gfx950 \
    HSA_ENABLE_IPC_MODE_LEGACY=0 \
    PYTHONUNBUFFERED=1

WORKDIR /opt/kubetorch-amd
COPY pyproject.toml README.md ./
COPY kubetorch_amd ./kubetorch_amd
COPY __graft_entry__.py bench.py ./

# build the gfx950 HIP extensions in-tree and install the package
RUN python -c "from kubetorch_amd.ops.build import build; build()" \
    && pip install --no-cache-dir -e .

# pod server port, controller port, data-store port, metrics
EXPOSE 32300 8081 8873 9090

# default: worker pod server (the chart overrides command per component:
# controller -> `kt server start --kind controller`, data store ->
# `kt server start --kind store`)
CMD ["kt", "server", "start", "--port", "32300"]
