# stoke-amd runtime image: ROCm 7 + PyTorch-ROCm for MI355X (gfx950)
FROM rocm/pytorch:rocm7.0_ubuntu22.04_py3.10_pytorch_release_2.10
ENV PYTORCH_ROCM_ARCH=gfx950 \
    HSA_ENABLE_IPC_MODE_LEGACY=0
WORKDIR /workspace/stoke-amd
COPY . .
RUN pip install --no-cache-dir attrs pyyaml numpy pytest && \
    python setup.py build_ext --inplace && \
    python -m pytest tests -m "not gpu" -q
CMD ["/bin/bash"]
