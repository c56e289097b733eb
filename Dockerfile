# persia_amd runtime image — ROCm 7.x + PyTorch-ROCm on MI355X (gfx950).
# The reference ships a CUDA/conda Dockerfile (reference Dockerfile:1-60);
# on ROCm the official pytorch image already carries hipcc/rocBLAS/
# hipBLASLt/RCCL/rocprofv3, so the build is just the in-tree extension.
ARG BASE_IMAGE=rocm/pytorch:latest
FROM ${BASE_IMAGE}

ENV PYTORCH_ROCM_ARCH=gfx950 \
    HSA_ENABLE_IPC_MODE_LEGACY=0

WORKDIR /workspace/persia_amd
COPY . .

# gfx950-only HIP extension, built in-tree (the .so ships with the source
# tree; no site-packages install so multi-node snapshots stay consistent)
RUN python setup.py build_ext --inplace && \
    python -m pytest tests -q -m "not gpu"

# trainer entry: one rank per GPU over RCCL
# docker run --device=/dev/kfd --device=/dev/dri ... \
#   python -m torch.distributed.run --nnodes=1 --nproc-per-node=8 \
#     --master-addr=127.0.0.1 your_train.py
CMD ["python", "bench.py", "--steps", "100"]
