# Single-node 8xMI355X training image (reference shipped docker/standalone
# CPU/GPU Dockerfiles with a Hadoop+Spark pseudo-cluster; this framework
# needs only ROCm + PyTorch-ROCm).
FROM rocm/pytorch:latest
WORKDIR /opt/caffeonspark-amd
COPY . .
ENV PYTORCH_ROCM_ARCH=gfx950 \
    HSA_ENABLE_IPC_MODE_LEGACY=0
RUN python setup.py build_ext --inplace && \
    python -m pytest tests -q -m "not gpu"
ENTRYPOINT ["python", "-m", "caffeonspark_amd.tools.mini_cluster"]
