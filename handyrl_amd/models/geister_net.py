"""GeisterNet — Conv-LSTM (DRC) recurrent policy-value-return network.

Architecture parity with the reference Geister net (reference
envs/geister.py:18-167): scalar features broadcast over the board and
concatenated with board planes, conv+BN stem, a Deep-Repeated Conv-LSTM
core (3 layers x 3 repeats, arXiv:1901.03559), a 144-way move head plus a
70-way layout head, and value/return scalar heads.

MI355X path: the ConvLSTM cell (concat -> 3x3 conv -> 4-gate epilogue) is
the fusion target for a single CDNA4 HIP kernel per cell evaluation
(implicit GEMM + sigmoid/tanh gate epilogue); the eager composition below
is the reference semantics and the CPU fallback.
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from .common import apply_bn


class ConvLSTMCell(nn.Module):
    def __init__(self, ch_in, ch_hidden, ksize=3):
        super().__init__()
        self.ch_hidden = ch_hidden
        self.conv = nn.Conv2d(ch_in + ch_hidden, 4 * ch_hidden, ksize,
                              padding=ksize // 2, bias=True)

    def init_hidden(self, shape, batch_size):
        zeros = lambda: torch.zeros(*batch_size, self.ch_hidden, *shape)
        return zeros(), zeros()

    def forward(self, x, state):
        h, c = state
        gates = self.conv(torch.cat([x, h], dim=-3))
        gi, gf, go, gg = torch.split(gates, self.ch_hidden, dim=-3)
        c_next = torch.sigmoid(gf) * c + torch.sigmoid(gi) * torch.tanh(gg)
        h_next = torch.sigmoid(go) * torch.tanh(c_next)
        return h_next, c_next


class DRC(nn.Module):
    """Deep Repeated Conv-LSTM: a stack of cells iterated several times per
    environment step for extra computation depth at constant parameters."""

    def __init__(self, num_layers, ch_in, ch_hidden, ksize=3):
        super().__init__()
        # attribute named 'blocks' for .pth state_dict compatibility with
        # reference geister.py's DRC (checkpoint layout is a contract)
        self.blocks = nn.ModuleList(
            ConvLSTMCell(ch_in, ch_hidden, ksize) for _ in range(num_layers))

    def init_hidden(self, shape, batch_size):
        hs, cs = [], []
        for cell in self.blocks:
            h, c = cell.init_hidden(shape, batch_size)
            hs.append(h)
            cs.append(c)
        return hs, cs

    def forward(self, x, hidden, num_repeats):
        if hidden is None:
            hidden = self.init_hidden(x.shape[-2:], x.shape[:-3])
        hs, cs = list(hidden[0]), list(hidden[1])
        for _ in range(num_repeats):
            for i, cell in enumerate(self.blocks):
                inp = x if i == 0 else hs[i - 1]
                hs[i], cs[i] = cell(inp, (hs[i], cs[i]))
        return hs[-1], (hs, cs)


class SpatialHead(nn.Module):
    """conv3x3+BN+ReLU -> 1x1 conv, flattened (reference Conv2dHead)."""

    def __init__(self, ch_in, filters, out_filters, hw):
        super().__init__()
        self.conv1 = nn.Conv2d(ch_in, filters, 3, padding=1, bias=False)
        self.bn = nn.BatchNorm2d(filters)
        self.conv2 = nn.Conv2d(filters, out_filters, 1, bias=False)

    def forward(self, x):
        h = F.relu(apply_bn(self.bn, self.conv1(x)))
        return self.conv2(h).flatten(1)


class ScalarHead(nn.Module):
    """1x1 conv+BN+ReLU -> FC scalar head (reference ScalarHead)."""

    def __init__(self, ch_in, filters, hw, outputs):
        super().__init__()
        self.conv = nn.Conv2d(ch_in, filters, 1, bias=False)
        self.bn = nn.BatchNorm2d(filters)
        self.fc = nn.Linear(filters * hw[0] * hw[1], outputs, bias=False)

    def forward(self, x):
        h = F.relu(apply_bn(self.bn, self.conv(x)))
        return self.fc(h.flatten(1))


class GeisterNet(nn.Module):
    def __init__(self):
        super().__init__()
        filters = 32
        ch_in = 7 + 18      # board planes + broadcast scalar features
        self.board_shape = (6, 6)

        self.conv1 = nn.Conv2d(ch_in, filters, 3, padding=1, bias=False)
        self.bn1 = nn.BatchNorm2d(filters)
        self.body = DRC(3, filters, filters)

        self.head_p_move = SpatialHead(filters, 8, 4, self.board_shape)
        self.head_p_set = nn.Linear(1, 70, bias=True)
        self.head_v = ScalarHead(filters, 2, self.board_shape, 1)
        self.head_r = ScalarHead(filters, 2, self.board_shape, 1)

    def init_hidden(self, batch_size=[]):
        return self.body.init_hidden(self.board_shape, batch_size)

    def forward(self, x, hidden):
        board, scalar = x['board'], x['scalar']
        s_planes = scalar.view(*scalar.size(), 1, 1).expand(*scalar.size(), *self.board_shape)
        h = torch.cat([s_planes, board], dim=-3)
        h = F.relu(apply_bn(self.bn1, self.conv1(h)))
        h, hidden = self.body(h, hidden, num_repeats=3)

        p_move = self.head_p_move(h)
        p_set = self.head_p_set(scalar[:, :1])
        policy = torch.cat([p_move, p_set], dim=-1)
        value = torch.tanh(self.head_v(h))
        ret = self.head_r(h)
        return {'policy': policy, 'value': value, 'return': ret, 'hidden': hidden}


def convlstm_cell_split(cell, x, state):
    """Concat-free reformulation of ConvLSTMCell.forward — the exact math
    the planned fused DRC kernel implements (docs/drc_kernel_plan.md):
    the (ch_in + ch_hidden) -> 4*hidden conv splits along input channels
    into an x-half and an h-half (conv(cat(x, h), W) == conv(x, Wx) +
    conv(h, Wh)), so the kernel K-orders the two weight halves instead of
    materializing a concat tensor.  CPU-verified equivalent in
    tests/test_models.py."""
    h, c = state
    ch_in = cell.conv.in_channels - cell.ch_hidden
    wx, wh = cell.conv.weight[:, :ch_in], cell.conv.weight[:, ch_in:]
    gates = (F.conv2d(x, wx, cell.conv.bias, padding=cell.conv.padding)
             + F.conv2d(h, wh, padding=cell.conv.padding))
    gi, gf, go, gg = torch.split(gates, cell.ch_hidden, dim=-3)
    c_next = torch.sigmoid(gf) * c + torch.sigmoid(gi) * torch.tanh(gg)
    h_next = torch.sigmoid(go) * torch.tanh(c_next)
    return h_next, c_next
