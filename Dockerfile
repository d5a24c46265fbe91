# gordo-amd — reproducible environment spec (parity: the reference ships
# Dockerfile-ModelBuilder / Dockerfile-GordoDeploy / Dockerfile-Watchman;
# here one image serves every role, the CLI selects it).
#
# Base: ROCm 7.2 + PyTorch-ROCm (the image this framework is developed and
# benchmarked on). The HIP extension targets gfx950 (MI355X) ONLY.
FROM rocm/pytorch:rocm7.2_ubuntu22.04_py3.10_pytorch_2.10

WORKDIR /opt/gordo-amd
COPY requirements.txt .
RUN pip install --no-cache-dir -r requirements.txt

COPY setup.py .
COPY gordo_amd/ gordo_amd/
# Build the CDNA4 kernels in-tree for gfx950 (cross-compiles without a GPU)
ENV PYTORCH_ROCM_ARCH=gfx950
RUN python setup.py build_ext --inplace && pip install --no-deps -e .

# dmabuf IPC is required for RCCL / cross-process CUDA-tensor sharing on
# this driver stack
ENV HSA_ENABLE_IPC_MODE_LEGACY=0

# Roles (reference: one Dockerfile per role; here one image):
#   builder:   gordo-amd build --machine-config ... --output-dir ...
#   server:    gordo-amd run-server --model-dir ...
#   fleet:     gordo-amd fleet build ... && gordo-amd fleet serve ...
#   watchman:  gordo-amd run-watchman --project-config ...
#   client:    gordo-amd client predict ...
ENTRYPOINT ["gordo-amd"]
CMD ["--help"]
