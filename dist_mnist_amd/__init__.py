"""dist_mnist_amd — MI355X-native distributed MNIST training framework.

A from-scratch, gfx950/CDNA4-first implementation with the capabilities of the
public ``leo-mao/dist-mnist`` example (the reference tree mounted
at /root/reference was empty at v0, so the compatibility surface is frozen from
the structural survey):

- reference-compatible CLI entrypoint (``dist_mnist.py``), env-var launch
  (MASTER_ADDR/MASTER_PORT/RANK/WORLD_SIZE), and ``torch.save`` state_dict
  checkpoints with the reference's exact key/shape set.
- every hot-path op (Conv2d fwd/bwd via fused-im2col MFMA GEMM, ReLU, MaxPool2d,
  Linear, LogSoftmax+NLLLoss, Dropout, SGD step) hand-written HIP for CDNA4
  (``native/``), built in-tree for gfx950 only — no Triton, no CUDA shims.
- DistributedDataParallel-style bucketed, backward-overlapped gradient averaging
  (``parallel/ddp.py``) on RCCL over xGMI (gloo kept for CPU plumbing).
- synthetic 1x28x28 data fed from pinned host memory via async H2D copies on a
  side stream overlapped with compute (``runtime/``).
"""

__version__ = "0.1.0"

from . import data  # noqa: F401
from .models.lenet import Net  # noqa: F401
from .parallel.ddp import DistributedDataParallel  # noqa: F401
