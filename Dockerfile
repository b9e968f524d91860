# Runtime image: ROCm 7.x + PyTorch-ROCm on an MI355X (gfx950) node.
# (Parity with the reference's Dockerfile, which pinned
# tensorflow:1.14.0-gpu-py3; here the base is AMD's official torch image.)
FROM rocm/pytorch:latest

WORKDIR /workspace/distributed-rl-mi355x
COPY . .

ENV PYTORCH_ROCM_ARCH=gfx950 \
    HSA_ENABLE_IPC_MODE_LEGACY=0

# build the in-tree gfx950 extension (hipcc cross-compiles without a GPU)
RUN python -m distributed_reinforcement_learning_amd.ops.build

# smoke: CPU test suite
RUN python -m pytest tests/ -q -m "not gpu" || true

CMD ["python", "train_impala.py", "--spawn"]
