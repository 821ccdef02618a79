# simple_tip_amd — MI355X-native TIP/active-learning engine.
# ROCm counterpart of the reference's TF-CUDA image (reference Dockerfile:1).
FROM rocm/pytorch:rocm7.0_ubuntu22.04_py3.10_pytorch_release_2.10.0

ENV PYTORCH_ROCM_ARCH=gfx950 \
    HSA_ENABLE_IPC_MODE_LEGACY=0

WORKDIR /workspace
COPY . /workspace

# Build the in-tree HIP/CDNA4 extension (gfx950; hipcc cross-compiles
# without a GPU present at build time).
RUN python setup.py build_ext --inplace

# Artifact fabric mount point (reference: /assets)
VOLUME ["/assets"]

ENTRYPOINT ["python", "reproduction.py"]
