# Deployment container (SURVEY.md §2a R10 — the reference ships a Dockerfile
# for launching one process per rank under an orchestrator).
#
# Base: a ROCm 7.x PyTorch image with MI355X (gfx950) support. The build
# cross-compiles the in-tree HIP extension; no GPU is needed at build time.
FROM rocm/pytorch:rocm7.0_ubuntu22.04_py3.10_pytorch_2.10

WORKDIR /workspace/dist-mnist-mi355x
COPY . .

ENV PYTORCH_ROCM_ARCH=gfx950 \
    HSA_ENABLE_IPC_MODE_LEGACY=0

RUN python tools/build_ext.py

# Rank env (MASTER_ADDR/MASTER_PORT/RANK/WORLD_SIZE) is injected by the
# launcher (Kubeflow PyTorchJob — see deploy/pytorchjob.yaml — or torchrun).
ENTRYPOINT ["python", "dist_mnist.py"]
