"""Locate where the custom MFMA/NHWC training path diverges from a
quantization-matched eager reference (the tight parity test measures
~2e-2 absolute on the final logits; accumulation-order noise alone
should be ~1e-5).

Compares one stage at a time on identical bf16-quantized inputs:
  conv   : torus_conv_fused (train form: no BN fold) vs fp32 circular conv
  bnfwd  : bn_nhwc_fwd vs eager batch_norm(+res+relu) on the same input
  block  : one full block fwd (conv+bn) vs emulation
  tower  : 3-block forward policy/value vs emulation (the failing test)

Prints max-abs / max-rel per stage.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F


def stats(name, a, b):
    d = (a.float() - b.float()).abs()
    scale = b.float().abs().mean().clamp(min=1e-8)
    print('%-8s max_abs %.3e  mean_abs %.3e  max_rel_to_mean %.3e'
          % (name, d.max(), d.mean(), d.max() / scale), flush=True)


def main():
    from handyrl_amd import ops
    from handyrl_amd.models.geese_net import GeeseNet
    torch.manual_seed(11)
    dev = torch.device('cuda', 0)
    net = GeeseNet(layers=3).to(dev).train()
    N = 256

    obs = (torch.rand(N, 17, 7, 11, device=dev) < 0.2).float()
    nbr = ops.torus_neighbor_table(dev)
    zero_shift = torch.zeros(32, device=dev)

    # ---- stage 1: conv only (layer 0) ----
    x_nhwc = F.pad(obs.permute(0, 2, 3, 1).reshape(N, 77, 17), (0, 15)) \
        .to(torch.bfloat16).contiguous()
    w0 = net.conv0.conv.weight.detach()
    wfrag = ops.pack_weights_hip(w0)
    conv_mine = ops.torus_conv_fused(x_nhwc, wfrag, zero_shift, nbr,
                                     None, False)
    w0q = w0.to(torch.bfloat16).float()
    conv_ref = F.conv2d(F.pad(obs, (1, 1, 1, 1), mode='circular'), w0q)
    conv_ref_nhwc = conv_ref.permute(0, 2, 3, 1).reshape(N, 77, 32)
    stats('conv', conv_mine, conv_ref_nhwc)

    # ---- stage 2: BN fwd on the SAME input ----
    conv_q = conv_ref_nhwc.to(torch.bfloat16)        # common input
    bn = net.conv0.bn
    rm_a = bn.running_mean.clone()
    rv_a = bn.running_var.clone()
    y_mine, mean_m, rstd_m = ops.bn_nhwc_fwd(
        conv_q.contiguous(), None, bn.weight, bn.bias, rm_a, rv_a,
        bn.momentum, bn.eps, True)
    rm_b = bn.running_mean.clone()
    rv_b = bn.running_var.clone()
    y_ref = F.batch_norm(
        conv_q.float().reshape(N, 77, 32).permute(0, 2, 1).reshape(N, 32, 7, 11),
        rm_b, rv_b, bn.weight, bn.bias, True, bn.momentum, bn.eps).relu()
    y_ref_nhwc = y_ref.permute(0, 2, 3, 1).reshape(N, 77, 32) \
        .to(torch.bfloat16)
    stats('bnfwd', y_mine, y_ref_nhwc)
    stats('bnmean', mean_m, conv_q.float().mean(dim=(0, 1)))
    var_ref = conv_q.float().var(dim=(0, 1), unbiased=False)
    stats('bnrstd', rstd_m, torch.rsqrt(var_ref + bn.eps))
    stats('rmean', rm_a, rm_b)
    stats('rvar', rv_a, rv_b)

    # ---- stage 4: full tower policy/value (the failing comparison) ----
    import copy
    net_ref = copy.deepcopy(net)
    out_mine = net(obs, None)

    class _Q(torch.autograd.Function):
        @staticmethod
        def forward(ctx, x):
            return x.to(torch.bfloat16).float()
        @staticmethod
        def backward(ctx, g):
            return g.to(torch.bfloat16).float()
    q = _Q.apply
    h = q(obs)
    layers = [net_ref.conv0] + list(net_ref.blocks)
    for i, layer in enumerate(layers):
        w = q(layer.conv.weight)
        conv_out = q(F.conv2d(F.pad(h, (1, 1, 1, 1), mode='circular'), w))
        b = layer.bn
        y = F.batch_norm(conv_out, b.running_mean, b.running_var,
                         b.weight, b.bias, True, b.momentum, b.eps)
        h = q(torch.relu(y + h if i > 0 else y.relu()))
    hf = h.flatten(2)
    head_cell = (hf * obs[:, :1].flatten(2)).sum(-1)
    board_avg = hf.mean(-1)
    policy = net_ref.head_p(head_cell)
    value = torch.tanh(net_ref.head_v(torch.cat([head_cell, board_avg], 1)))
    stats('policy', out_mine['policy'], policy)
    stats('value', out_mine['value'], value)
    # where do the towers diverge? compare running stats per layer
    for i, (lm, lr) in enumerate(zip([net.conv0] + list(net.blocks), layers)):
        stats('rm%d' % i, lm.bn.running_mean, lr.bn.running_mean)
        stats('rv%d' % i, lm.bn.running_var, lr.bn.running_var)


if __name__ == '__main__':
    main()
