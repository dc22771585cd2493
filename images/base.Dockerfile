# Shared base: ROCm 7.2 + PyTorch-ROCm + the runbooks_amd package with the
# gfx950 HIP extension built in (PYTORCH_ROCM_ARCH=gfx950).
FROM rocm/pytorch:rocm7.2_ubuntu22.04_py3.10_pytorch_release_2.10.0
WORKDIR /opt/runbooks-amd
COPY setup.py ./
COPY runbooks_amd ./runbooks_amd
RUN PYTORCH_ROCM_ARCH=gfx950 python3 setup.py build_ext --inplace && \
    pip install --no-deps -e .
ENV HSA_ENABLE_IPC_MODE_LEGACY=0
WORKDIR /content
