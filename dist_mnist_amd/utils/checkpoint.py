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
    sd = {k: v.detach().to("cpu", torch.float32) for k, v in module.state_dict().items()}
    torch.save(sd, path)


def load_checkpoint(model, path: str, device=None):
    module = model.module if hasattr(model, "module") else model
    sd = torch.load(path, map_location="cpu", weights_only=True)
    module.load_state_dict(sd)
    if device is not None:
        module.to(device)
    return model
