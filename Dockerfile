# MI355X (gfx950) runtime image for sparktorch_amd.
# Counterpart of the reference's Ubuntu+conda+Spark image
# (/root/reference/Dockerfile:29-31), rebuilt on the ROCm PyTorch base: the
# JVM/Spark layer is optional (the local barrier engine needs none), so the
# image stays a pure ROCm+torch stack with the gfx950 extension prebuilt.

FROM rocm/pytorch:rocm7.2_ubuntu22.04_py3.10_pytorch_release_2.10

ENV PYTORCH_ROCM_ARCH=gfx950 \
    HSA_ENABLE_IPC_MODE_LEGACY=0 \
    PYTHONUNBUFFERED=1

WORKDIR /opt/sparktorch_amd

COPY requirements.txt setup.py Makefile __graft_entry__.py bench.py ./
RUN pip install --no-cache-dir dill pytest pytest-timeout numpy

COPY sparktorch_amd ./sparktorch_amd
COPY tests ./tests
COPY examples ./examples
COPY vendor ./vendor

# cross-compile the gfx950 extension at image build time (no GPU needed)
RUN make clean && make build

# optional: real Spark execution — uncomment to bake pyspark in
# RUN pip install --no-cache-dir pyspark>=3.4

CMD ["python", "-m", "pytest", "tests/", "-x", "-q", "-m", "not gpu"]
