"""GeeseNet — torus-convolution residual policy-value network.

Architecture parity with the reference Hungry Geese net (reference
envs/kaggle/hungry_geese.py:23-57): 17->32 torus conv stem, 12 residual
torus conv blocks with BatchNorm, a policy head over the head-cell feature
vector and a value head over [head-cell, board-average] features.

MI355X path: on GPU the torus conv + BN + ReLU (+residual) stack dispatches
to the fused CDNA4 HIP kernels in handyrl_amd/ops (implicit-GEMM on MFMA,
wrap-around indexing folded into the tile gather) instead of
pad+conv+bn+relu eager chains.
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from .common import apply_bn


class TorusConv2d(nn.Module):
    """3x3 convolution with wrap-around (circular) padding + optional BN."""

    def __init__(self, ch_in, ch_out, bn=True):
        super().__init__()
        self.conv = nn.Conv2d(ch_in, ch_out, 3, padding=1,
                              padding_mode='circular', bias=not bn)
        self.bn = nn.BatchNorm2d(ch_out) if bn else None

    def forward(self, x):
        return apply_bn(self.bn, self.conv(x))


class _TorusBlockFn(torch.autograd.Function):
    """One GeeseNet block on hand-written CDNA4 kernels, trainable:
    y = act(BN(torus_conv3x3(x)) [+ x]) over NHWC bf16 activations.

    forward: pack-weights kernel -> MFMA implicit-GEMM conv -> NHWC BN
    (+fused residual/relu).  backward: fused BN/relu/residual backward ->
    data-grad as the SAME MFMA conv with flipped/transposed packed weights
    -> weight-grad as gather + batched GEMM (hipBLASLt).  All nodes are
    hipGraph-capturable.
    """

    @staticmethod
    def forward(ctx, x, conv_w, bn_w, bn_b, r_mean, r_var, nbr, zero_shift,
                momentum, eps, residual, relu):
        from .. import ops
        wfrag = ops.pack_weights_hip(conv_w)
        conv_out = ops.torus_conv_fused(x, wfrag, zero_shift, nbr, None, False)
        y, mean, rstd = ops.bn_nhwc_fwd(conv_out, x if residual else None,
                                        bn_w, bn_b, r_mean, r_var,
                                        momentum, eps, relu)
        ctx.save_for_backward(x, conv_w, bn_w, conv_out, y, mean, rstd,
                              nbr, zero_shift)
        ctx.flags = (residual, relu)
        return y

    @staticmethod
    def backward(ctx, dy):
        from .. import ops
        (x, conv_w, bn_w, conv_out, y, mean, rstd,
         nbr, zero_shift) = ctx.saved_tensors
        residual, relu = ctx.flags

        dconv, dbnw, dbnb, dres = ops.bn_nhwc_bwd(
            conv_out, dy, y, bn_w, mean, rstd, relu, residual)

        dx = None
        if ctx.needs_input_grad[0]:
            wfrag_d = ops.pack_weights_hip(conv_w, dgrad=True)
            dx = ops.torus_conv_fused(dconv, wfrag_d, zero_shift, nbr,
                                      dres if residual else None, False)

        # weight grad: MFMA kernel (LDS-staged per image, x^T shared
        # across all 9 taps)
        ci = conv_w.shape[1]
        dw_t = ops.torus_wgrad(x, dconv, nbr)                  # (9, ci32, co)
        dW = dw_t.permute(2, 1, 0)[:, :ci].reshape(32, ci, 3, 3).contiguous()
        return (dx, dW, dbnw, dbnb, None, None, None, None,
                None, None, None, None)


class GeeseNet(nn.Module):
    # NOTE: channels_last was tried and REVERTED: on ROCm 7.x it routes
    # convs to composable_kernel paths whose weight-grad kernel costs
    # 7.4ms/call at these shapes (vs 33us for MIOpen igemm_wrw in NCHW) —
    # measured in profiles/bench_kernel_stats_r01_nhwc.txt.
    prefers_channels_last = False

    def __init__(self, layers=12, filters=32, ch_in=17, actions=4):
        super().__init__()
        # named conv0 for reference-checkpoint layout compatibility
        self.conv0 = TorusConv2d(ch_in, filters)
        self.blocks = nn.ModuleList(TorusConv2d(filters, filters) for _ in range(layers))
        self.head_p = nn.Linear(filters, actions, bias=False)
        self.head_v = nn.Linear(filters * 2, 1, bias=False)

    def reference_state_dict(self):
        """Reference-layout export hook (used by Learner.update_model)."""
        return export_reference_state_dict(self)

    def load_reference_state_dict(self, sd):
        """Load EITHER layout exactly: reference checkpoints (with
        under-BN conv biases) fold the bias into running_mean; this
        repo's own exports (zero biases) reduce to a strict load."""
        load_reference_state_dict(self, sd)

    def forward(self, x, hidden=None):
        if x.is_cuda and self.training and self._custom_train_ok():
            return self._forward_nhwc_train(x)
        h = F.relu_(self.conv0(x))
        for blk in self.blocks:
            h = F.relu_(h + blk(h))
        flat = h.flatten(2)                                   # (B, C, H*W)
        head_cell = (flat * x[:, :1].flatten(2)).sum(-1)      # feature at own head
        board_avg = flat.mean(-1)
        policy = self.head_p(head_cell)
        value = torch.tanh(self.head_v(torch.cat([head_cell, board_avg], 1)))
        return {'policy': policy, 'value': value}

    # ---- hand-written CDNA4 training path (NHWC bf16) --------------------
    def _custom_train_ok(self):
        import os
        if os.environ.get('HANDYRL_NO_FUSED') == '1':
            return False
        if self.conv0.conv.weight.shape[0] != 32 or \
                self.conv0.conv.weight.shape[1] > 32:
            return False                      # kernels assume 32 channels
        from .. import ops
        return ops.available()

    def _nhwc_consts(self, x):
        from .. import ops
        dev = x.device
        if not hasattr(self, '_nbr') or self._nbr.device != dev:
            self._nbr = ops.torus_neighbor_table(dev)
            self._zero_shift = torch.zeros(32, device=dev)
        return self._nbr, self._zero_shift

    def _forward_nhwc_train(self, x):
        """Training forward on the custom MFMA conv + NHWC BN kernels.
        x: (N, 17, 7, 11) float (0/1 observation planes)."""
        N = x.shape[0]
        nbr, zero_shift = self._nhwc_consts(x)
        h = x.permute(0, 2, 3, 1).reshape(N, 77, 17)
        h = F.pad(h, (0, 15)).to(torch.bfloat16).contiguous()   # ci pad to 32

        layers = [self.conv0] + list(self.blocks)
        for i, layer in enumerate(layers):
            if layer.bn.num_batches_tracked is not None:
                layer.bn.num_batches_tracked += 1
            momentum = layer.bn.momentum if layer.bn.momentum is not None else 0.1
            h = _TorusBlockFn.apply(
                h, layer.conv.weight, layer.bn.weight, layer.bn.bias,
                layer.bn.running_mean, layer.bn.running_var,
                nbr, zero_shift,
                momentum, layer.bn.eps, i > 0, True)

        hf = h.float()                                         # (N, 77, 32)
        plane = x[:, 0].reshape(N, 77)
        head_cell = (hf * plane.unsqueeze(-1)).sum(1)
        board_avg = hf.mean(1)
        policy = self.head_p(head_cell)
        value = torch.tanh(self.head_v(torch.cat([head_cell, board_avg], 1)))
        return {'policy': policy, 'value': value}


class GeeseFusedEval:
    """Hand-written CDNA4 inference path for GeeseNet (actor side).

    The whole 13-layer torus-conv tower runs as 13 fused MFMA kernels on
    NHWC bf16 activations (handyrl_amd/ops/src/ext.hip::torus_conv_fused):
    wrap-around gather + implicit-GEMM conv + folded BN affine + residual +
    ReLU in one kernel per layer, with the uint8 observation -> padded NHWC
    conversion as the entry kernel.  BN-folded weights live in persistent
    packed buffers; ``refresh()`` re-folds from the live training module
    (call after each optimizer step; actors then run <=1 step stale).
    """

    def __init__(self, net, device):
        from .. import ops
        self.ops = ops
        self.net = net
        self.device = device
        self.nbr = ops.torus_neighbor_table(device)
        self.layers = [net.conv0] + list(net.blocks)
        self.wfrag = [torch.zeros(9, 2, 4, 16, 8, dtype=torch.bfloat16,
                                  device=device) for _ in self.layers]
        self.shift = [torch.zeros(32, dtype=torch.float32, device=device)
                      for _ in self.layers]
        self.refresh()

    @torch.no_grad()
    def refresh(self):
        for i, layer in enumerate(self.layers):
            bn = layer.bn
            rstd = torch.rsqrt(bn.running_var + bn.eps)
            scale = bn.weight * rstd
            shift = bn.bias - bn.running_mean * scale
            self.wfrag[i].copy_(self.ops.pack_weights_hip(layer.conv.weight, scale))
            self.shift[i].copy_(shift)

    def forward_canonical(self, obs_u8):
        """CANONICAL obs_u8 (G,17,7,11) CUDA uint8 -> {'policy','value'}
        fp32 with 4 game-major seat rows per game (row g*4+k = seat k).
        The seat channel rotation happens inside obs_to_nhwc_rot — the
        host never materializes per-seat observations."""
        ops = self.ops
        h = ops.obs_to_nhwc_rot(obs_u8)                 # (4G, 77, 32)
        h = ops.torus_conv_fused(h, self.wfrag[0], self.shift[0], self.nbr,
                                 None, True)
        for i in range(1, len(self.layers)):
            h = ops.torus_conv_fused(h, self.wfrag[i], self.shift[i], self.nbr,
                                     h, True)
        hf = h.float()                                  # (4G, 77, 32)
        G = obs_u8.shape[0]
        # seat k's own-head plane is canonical channel k
        plane0 = obs_u8.reshape(G, 17, 77)[:, :4].float().reshape(G * 4, 77)
        head_cell = (hf * plane0.unsqueeze(-1)).sum(1)
        board_avg = hf.mean(1)
        policy = self.net.head_p(head_cell)
        value = torch.tanh(self.net.head_v(torch.cat([head_cell, board_avg], 1)))
        return {'policy': policy, 'value': value}

    def forward(self, obs_u8):
        """obs_u8 (M,17,7,11) CUDA uint8 -> {'policy','value'} fp32."""
        ops = self.ops
        h = ops.obs_to_nhwc(obs_u8)
        h = ops.torus_conv_fused(h, self.wfrag[0], self.shift[0], self.nbr,
                                 None, True)
        for i in range(1, len(self.layers)):
            h = ops.torus_conv_fused(h, self.wfrag[i], self.shift[i], self.nbr,
                                     h, True)
        hf = h.float()                                         # (M, 77, 32)
        plane0 = obs_u8.flatten(2, 3)[:, 0].float()            # own-head plane
        head_cell = (hf * plane0.unsqueeze(-1)).sum(1)
        board_avg = hf.mean(1)
        policy = self.net.head_p(head_cell)
        value = torch.tanh(self.net.head_v(torch.cat([head_cell, board_avg], 1)))
        return {'policy': policy, 'value': value}


def export_reference_state_dict(model):
    """State dict in the REFERENCE GeeseNet layout (the inverse direction
    of load_reference_state_dict): the reference net declares a redundant
    `conv.bias` under each BatchNorm (hungry_geese.py:26) that this model
    drops, so emit an explicit zero bias per TorusConv2d.  BN(z + 0) ==
    BN(z), so the exported checkpoint loads into the reference net (which
    uses strict=False) with identical outputs instead of silently keeping
    its random-init conv biases."""
    sd = {k: v.clone() for k, v in model.state_dict().items()}
    for name, mod in model.named_modules():
        if isinstance(mod, TorusConv2d) and mod.conv.bias is None:
            w = mod.conv.weight
            sd[name + '.conv.bias'] = torch.zeros(
                w.shape[0], dtype=w.dtype, device=w.device)
    return sd


def load_reference_state_dict(model, sd):
    """Load a REFERENCE GeeseNet checkpoint into this model EXACTLY.

    The reference's TorusConv2d (hungry_geese.py:23-35) keeps a redundant
    conv bias under BatchNorm; this model drops it.  The bias folds into
    the running mean with no numerical change: BN(z + b) == BN'(z) with
    running_mean' = running_mean - b.  All other keys map one to one
    (tests/test_checkpoint_compat.py proves output equivalence)."""
    sd = dict(sd)
    for name, mod in model.named_modules():
        if isinstance(mod, TorusConv2d):
            b = sd.pop(name + '.conv.bias', None)
            if b is not None:
                key = name + '.bn.running_mean'
                sd[key] = sd[key] - b
    model.load_state_dict(sd, strict=True)
