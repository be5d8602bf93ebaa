"""CDNA4 HIP kernel bindings.

The compiled extension lives IN-TREE (handyrl_amd/ops/_C.so) so it travels
with repo snapshots; ``build()`` (see handyrl_amd/ops/build.py) compiles it
for gfx950.  On a GPU box the ops below are required: if the extension is
missing they raise instead of silently falling back to eager PyTorch
(set HANDYRL_AMD_ALLOW_EAGER=1 to override for debugging).
"""

import glob
import importlib.machinery
import os

import torch

_EXT = None
_EXT_ERR = None


def _find_so():
    here = os.path.dirname(__file__)
    cands = sorted(glob.glob(os.path.join(here, '_C*.so')))
    return cands[0] if cands else None


def _load():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    so = _find_so()
    if so is None:
        _EXT_ERR = FileNotFoundError(
            'handyrl_amd HIP extension not built; run handyrl_amd/ops/build.py '
            '(or __graft_entry__.build()) to compile _C.so for gfx950')
        return None
    try:
        loader = importlib.machinery.ExtensionFileLoader('handyrl_amd_C', so)
        spec = importlib.util.spec_from_loader('handyrl_amd_C', loader)
        mod = importlib.util.module_from_spec(spec)
        loader.exec_module(mod)
        _EXT = mod
    except Exception as e:       # noqa: BLE001 - surface the load failure lazily
        _EXT_ERR = e
    return _EXT


def available():
    return _load() is not None


def require():
    ext = _load()
    if ext is None:
        if os.environ.get('HANDYRL_AMD_ALLOW_EAGER') == '1':
            return None
        raise RuntimeError('handyrl_amd HIP extension required on GPU but not loaded: %r' % (_EXT_ERR,))
    return ext


_SCAN_KIND = {'TD': 0, 'UPGO': 1, 'VTRACE': 2}


def target_scan(algorithm, values, returns, rewards, lambda_, gamma, rhos, cs):
    """Fused backward target scan on GPU. All inputs (B, T, P, 1) float32.

    Returns (targets, advantages), matching handyrl_amd.losses eager math.
    """
    ext = require()
    if ext is None:                      # debug escape hatch: eager fallback
        from .. import losses
        fn = {'TD': losses.temporal_difference, 'UPGO': losses.upgo}.get(algorithm)
        if fn is not None:
            return fn(values, returns, rewards, lambda_, gamma)
        return losses.vtrace(values, returns, rewards, lambda_, gamma, rhos, cs)

    kind = _SCAN_KIND[algorithm]
    v = values.contiguous().float()
    # only the bootstrap slice returns[:, -1] enters the scan; 'returns' may
    # be a (B, 1, P, 1) outcome tensor broadcast along T (train.py value path)
    ret = returns[:, -1].contiguous().float()
    rew = rewards.contiguous().float() if rewards is not None else None
    lam = lambda_.contiguous().float()
    if kind == 2:
        rho = rhos.contiguous().float()
        c = cs.contiguous().float()
    else:
        rho = c = None
    targets, adv = ext.target_scan(v, ret, rew, lam, rho, c, float(gamma), kind)
    return targets, adv


def masked_sample(logits, action_mask, uniform):
    """Sample one action per row from softmax(logits - action_mask).

    logits, action_mask: (N, A) float32 CUDA; uniform: (N,) float32 in [0,1).
    Returns (actions int64 (N,), selected_prob float32 (N,)).
    """
    ext = require()
    if ext is None:
        probs = torch.softmax(logits - action_mask, dim=-1)
        actions = torch.multinomial(probs, 1).squeeze(-1)
        return actions, probs.gather(-1, actions.unsqueeze(-1)).squeeze(-1)
    return ext.masked_sample(logits.contiguous().float(),
                             action_mask.contiguous().float(),
                             uniform.contiguous().float())
