"""Checkpoint write/read.

Reference behavior: ``torch.save(model.state_dict(), path)`` at the end of
training, model moved to CPU first if on GPU — a plain state_dict pickle with
the §3.4 key/shape set. ``--resume`` (an extension; absent in the reference)
loads the same format back.
"""

from __future__ import annotations

import torch


def save_checkpoint(model, path: str):
    module = model.module if hasattr(model, "module") else model
    # fp32 cast applies to floating tensors only: integer buffers (e.g.
    # ResNet BatchNorm num_batches_tracked, int64) keep their dtype so the
    # file stays interchangeable with stock torch state_dicts.
    sd = {
        k: (v.detach().to("cpu", torch.float32) if v.is_floating_point()
            else v.detach().to("cpu"))
        for k, v in module.state_dict().items()
    }
    torch.save(sd, path)


def load_checkpoint(model, path: str, device=None):
    module = model.module if hasattr(model, "module") else model
    sd = torch.load(path, map_location="cpu", weights_only=True)
    module.load_state_dict(sd)
    if device is not None:
        module.to(device)
    return model
