"""Model wrapper utilities (parity: reference handyrl/model.py).

Keeps worker-side CPU inference single-threaded (the GPU path batches
inference in handyrl_amd/actor.py instead of widening CPU threads).
"""

import os
os.environ.setdefault('OMP_NUM_THREADS', '1')

import numpy as np
import torch
torch.set_num_threads(1)

import torch.nn as nn

from .util import map_r


def to_torch(x):
    return map_r(x, lambda v: torch.from_numpy(np.array(v)).contiguous() if v is not None else None)


def to_numpy(x):
    return map_r(x, lambda v: v.detach().numpy() if v is not None else None)


def to_gpu(x, device=None, non_blocking=False):
    return map_r(x, lambda v: v.cuda(device, non_blocking=non_blocking) if v is not None else None)


class ModelWrapper(nn.Module):
    """Uniform numpy-in/numpy-out single-sample inference over an nn.Module,
    plus inference-vs-training hidden-state shape handling."""

    def __init__(self, model):
        super().__init__()
        self.model = model

    def init_hidden(self, batch_size=None):
        if not hasattr(self.model, 'init_hidden'):
            return None
        if batch_size is None:     # inference: no batch dims, numpy leaves
            hidden = self.model.init_hidden([])
            return map_r(hidden, lambda h: h.detach().numpy() if isinstance(h, torch.Tensor) else h)
        return self.model.init_hidden(batch_size)

    def forward(self, *args, **kwargs):
        return self.model.forward(*args, **kwargs)

    def inference(self, x, hidden, **kwargs):
        if hasattr(self.model, 'inference'):
            return self.model.inference(x, hidden, **kwargs)
        self.eval()
        with torch.no_grad():
            xt = map_r(x, lambda v: torch.from_numpy(np.array(v)).contiguous().unsqueeze(0) if v is not None else None)
            ht = map_r(hidden, lambda h: torch.from_numpy(np.array(h)).contiguous().unsqueeze(0) if h is not None else None)
            outputs = self.forward(xt, ht, **kwargs)
        return map_r(outputs, lambda o: o.detach().numpy().squeeze(0) if o is not None else None)


class RandomModel(nn.Module):
    """Uniform-policy / zero-value stand-in, shaped by probing a real model
    once (used as the ``model_id == 0`` opponent)."""

    def __init__(self, model, x):
        super().__init__()
        wrapped = ModelWrapper(model)
        outputs = wrapped.inference(x, wrapped.init_hidden())
        self.output_dict = {k: np.zeros_like(v) for k, v in outputs.items() if k != 'hidden'}

    def inference(self, *args, **kwargs):
        return self.output_dict
