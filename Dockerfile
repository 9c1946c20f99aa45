# MI355X (gfx950) runtime image: ROCm 7.x + PyTorch-ROCm.
# (Parity role of the reference's CUDA Dockerfile; the base tag tracks the
# ROCm release the kernels are built against.)
FROM rocm/pytorch:rocm7.0_ubuntu22.04_py3.10_pytorch_release_2.10
WORKDIR /workspace/stoix_amd
COPY . .
ENV PYTORCH_ROCM_ARCH=gfx950 \
    HSA_ENABLE_IPC_MODE_LEGACY=0
RUN python -m stoix_amd.ops.build
CMD ["python", "bench.py", "--gpus", "1"]
