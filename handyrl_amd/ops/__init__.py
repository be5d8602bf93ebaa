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


def mfma_probe(a_bf16, b_bf16):
    """Test-only: D(16,16) = A(16,32) @ B(32,16) via one MFMA."""
    return require().mfma_probe(a_bf16.contiguous(), b_bf16.contiguous())


def obs_to_nhwc(obs_u8):
    """(N,17,7,11) uint8 -> (N,77,32) bf16 NHWC, channels 17..31 zero."""
    return require().obs_to_nhwc(obs_u8.reshape(obs_u8.shape[0], 17, 77).contiguous())


def obs_to_nhwc_rot(obs_u8):
    """Canonical (G,17,7,11) uint8 -> seat-expanded (4G,77,32) bf16 NHWC
    (row g*4+k = seat k of game g; CHMAP channel rotation done in-kernel)."""
    return require().obs_to_nhwc_rot(
        obs_u8.reshape(obs_u8.shape[0], 17, 77).contiguous())


def torus_conv_fused(x, wfrag, shift, nbr, residual=None, relu=False):
    """y = act(torus_conv3x3(x) * scale + shift [+ residual]) on NHWC bf16.

    wfrag: (9,2,4,16,8) bf16 from pack_torus_weights; shift fp32 (32,);
    nbr: (77,9) int32 wrap-around neighbor table (see torus_neighbor_table);
    residual: optional (N,77,32) bf16 tensor added before the activation.
    """
    return require().torus_conv_fused(x, wfrag, shift, nbr, residual, relu)


def pack_weights_hip(w, scale=None, dgrad=False):
    """GPU packing of (32, ci<=32, 3, 3) fp32 conv weights into the MFMA
    fragment layout; dgrad=True produces the transposed/flipped weights for
    the data-gradient conv."""
    return require().pack_torus_weights_hip(w.contiguous(), scale, dgrad)


def torus_wgrad(x, dy, nbr):
    """dW[tap][ci][co] for the torus conv, fp32 (9,32,32)."""
    return require().torus_wgrad(x, dy, nbr)


def bn_nhwc_fwd(x, res, weight, bias, running_mean, running_var,
                momentum, eps, relu):
    """NHWC bf16 BN training fwd (+ optional residual add + relu).
    Returns (y, save_mean, save_rstd)."""
    return require().bn_nhwc_fwd(x, res, weight, bias, running_mean,
                                 running_var, momentum, eps, relu)


def bn_nhwc_bwd(x, dy, y, weight, save_mean, save_rstd, had_relu, want_dres):
    """Backward of bn_nhwc_fwd. Returns (dx, dweight, dbias, dres)."""
    return require().bn_nhwc_bwd(x, dy.contiguous(), y, weight,
                                 save_mean, save_rstd, had_relu, want_dres)


def torus_neighbor_table(device=None):
    """(77, 9) int32: nbr[cell][ky*3+kx] on the 7x11 torus (conv2d
    cross-correlation tap order, padding=1 circular)."""
    rows, cols = 7, 11
    tbl = torch.empty(rows * cols, 9, dtype=torch.int32)
    for r in range(rows):
        for c in range(cols):
            for ky in range(3):
                for kx in range(3):
                    rr = (r + ky - 1) % rows
                    cc = (c + kx - 1) % cols
                    tbl[r * cols + c, ky * 3 + kx] = rr * cols + cc
    return tbl.to(device) if device is not None else tbl


def pack_torus_weights(conv_weight, scale):
    """Pack (32, Cin<=32, 3, 3) conv weights (pre-scaled per out-channel)
    into the MFMA B-fragment layout the fused kernel reads:
    frag[tap][cotile][khi][lane_lo][e] = W[k = khi*8+e (ci), co = cotile*16+lane_lo],
    with k tap-major (k = tap*32 + ci), ci zero-padded to 32."""
    co, ci = conv_weight.shape[0], conv_weight.shape[1]
    assert co == 32 and ci <= 32
    w = conv_weight.float() * scale.float().view(-1, 1, 1, 1)
    wt = w.permute(2, 3, 1, 0).reshape(9, ci, co)          # (tap, ci, co)
    if ci < 32:
        pad = torch.zeros(9, 32 - ci, co, device=w.device, dtype=w.dtype)
        wt = torch.cat([wt, pad], dim=1)
    wt = wt.view(9, 4, 8, 2, 16)                           # (tap, khi, e, cotile, lo)
    frag = wt.permute(0, 3, 1, 4, 2).contiguous()          # (tap, cotile, khi, lo, e)
    return frag.to(torch.bfloat16)


def convlstm_cell(x, h, c, wfrag, bias, nbr, h_out, c_out):
    """Fused ConvLSTM cell (Geister DRC core, one kernel per cell eval):
    implicit-GEMM 3x3 zero-pad conv over the K-ordered (x | h) halves with
    the i/f/o/g gate + state update fused into the epilogue.

    x, h, h_out: (B, 36, 32) bf16 NHWC; c, c_out: (B, 36, 32) fp32;
    wfrag: pack_convlstm_weights output; bias: (128,) fp32;
    nbr: (36, 9) int32 from convlstm_neighbor_table (-1 = zero pad).
    Writes h_out/c_out in place and returns them."""
    return require().convlstm_cell(x, h, c, wfrag, bias, nbr, h_out, c_out)


def convlstm_neighbor_table(device=None):
    """(36, 9) int32 neighbor table for a 6x6 ZERO-padded 3x3 conv
    (cross-correlation tap order); -1 marks out-of-board taps."""
    rows = cols = 6
    tbl = torch.empty(rows * cols, 9, dtype=torch.int32)
    for r in range(rows):
        for c in range(cols):
            for ky in range(3):
                for kx in range(3):
                    rr, cc = r + ky - 1, c + kx - 1
                    ok = 0 <= rr < rows and 0 <= cc < cols
                    tbl[r * cols + c, ky * 3 + kx] = rr * cols + cc if ok else -1
    return tbl.to(device) if device is not None else tbl


def pack_convlstm_weights(conv_weight):
    """Pack a ConvLSTM conv weight (128, 64, 3, 3) fp32 into the fused
    cell kernel's B-fragment layout (2, 9, 8, 4, 16, 8) bf16:
    frag[src][tap][cotile][khi][lo][e] = W[n = cotile*16+lo]
                                          [ci = src*32 + khi*8+e][tap]."""
    co, ci = conv_weight.shape[0], conv_weight.shape[1]
    assert co == 128 and ci == 64, (co, ci)
    w9 = conv_weight.float().reshape(128, 64, 9)
    wv = w9.reshape(8, 16, 2, 4, 8, 9)        # (cotile, lo, src, khi, e, tap)
    frag = wv.permute(2, 5, 0, 3, 1, 4).contiguous()
    return frag.to(torch.bfloat16)


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
