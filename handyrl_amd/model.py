"""Model wrapper utilities.

Semantics contract (reference handyrl/model.py): ``ModelWrapper`` gives any
``nn.Module`` a uniform numpy-in/numpy-out single-sample ``inference`` and
handles the inference-vs-training hidden-state shape split; ``RandomModel``
is the uniform-policy epoch-0 opponent.  Worker-side CPU inference stays
single-threaded — the MI355X path batches inference on the GPU
(handyrl_amd/actor.py) instead of widening CPU threads.
"""

import os
os.environ.setdefault('OMP_NUM_THREADS', '1')

import numpy as np
import torch
torch.set_num_threads(1)

import torch.nn as nn

from .util import map_r


def to_torch(x):
    """Nested numpy/scalars -> contiguous CPU tensors (None passes through)."""
    return map_r(x, lambda v: None if v is None
                 else torch.from_numpy(np.asarray(v)).contiguous())


def to_numpy(x):
    """Nested tensors -> numpy arrays (None passes through)."""
    return map_r(x, lambda v: None if v is None else v.detach().numpy())


def to_gpu(x, device=None, non_blocking=False):
    """Nested tensors -> CUDA (H2D is async when sources are pinned)."""
    return map_r(x, lambda v: None if v is None
                 else v.cuda(device, non_blocking=non_blocking))


def _batched(v):
    """numpy leaf -> (1, ...) tensor."""
    return torch.from_numpy(np.asarray(v)).contiguous().unsqueeze(0)


def _unbatched(v):
    """(1, ...) tensor leaf -> numpy."""
    return v.detach().numpy().squeeze(0)


class ModelWrapper(nn.Module):
    """Uniform inference surface over an ``nn.Module``.

    ``inference(x, hidden)`` feeds one observation (numpy leaves, no batch
    dim), runs the wrapped net under no_grad with a temporary batch axis,
    and strips that axis from every output.  ``init_hidden()`` without a
    batch size returns numpy-leaf hidden state for the same single-sample
    regime; with a batch-size list it defers to the net's training shapes.
    """

    def __init__(self, model):
        super().__init__()
        self.model = model

    def init_hidden(self, batch_size=None):
        make = getattr(self.model, 'init_hidden', None)
        if make is None:
            return None
        if batch_size is not None:
            return make(batch_size)
        return map_r(make([]), lambda h: h.detach().numpy()
                     if isinstance(h, torch.Tensor) else h)

    def forward(self, *args, **kwargs):
        return self.model(*args, **kwargs)

    def inference(self, x, hidden, **kwargs):
        # nets with their own batched inference (e.g. ONNX-style stubs)
        # keep full control
        native = getattr(self.model, 'inference', None)
        if native is not None:
            return native(x, hidden, **kwargs)
        self.eval()
        with torch.no_grad():
            out = self.forward(map_r(x, lambda v: None if v is None
                                     else _batched(v)),
                               map_r(hidden, lambda h: None if h is None
                                     else _batched(h)),
                               **kwargs)
        return map_r(out, lambda o: None if o is None else _unbatched(o))


class RandomModel(nn.Module):
    """Uniform-policy / zero-value opponent (``model_id == 0``).

    Output keys and shapes are discovered by probing the real model on one
    observation; every later call returns those zeros (zero logits = a
    uniform policy after masking+softmax).
    """

    def __init__(self, model, x):
        super().__init__()
        probe = ModelWrapper(model)
        shaped = probe.inference(x, probe.init_hidden())
        self.output_dict = {k: np.zeros_like(v) for k, v in shaped.items()
                            if k != 'hidden'}

    def inference(self, *args, **kwargs):
        return self.output_dict
