"""Motion encoders, ConvGRUs, flow head, update blocks.

Re-design of networks/model_utils.py:110-194 (NCHW, checkpoint-compatible
child names — see SURVEY.md §5.4).  The pointwise GRU state update goes
through raft_amd.ops.gru_gates (fused HIP kernel on GPU).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from raft_amd import ops
from raft_amd.models.layers import Conv2dTF


class BasicMotionEncoder(nn.Module):
    """corr->1x1(256)->3x3(192); flow->7x7(128)->3x3(64); concat->3x3(126);
    output re-concat with raw flow -> 128ch (model_utils.py:110-119)."""

    def __init__(self, corr_channels: int):
        super().__init__()
        self.convc1 = Conv2dTF(corr_channels, 256, 1)
        self.convc2 = Conv2dTF(256, 192, 3)
        self.convf1 = Conv2dTF(2, 128, 7)
        self.convf2 = Conv2dTF(128, 64, 3)
        self.conv = Conv2dTF(192 + 64, 128 - 2, 3)

    def forward(self, flow, corr):
        cor = F.relu(self.convc1(corr))
        cor = F.relu(self.convc2(cor))
        flo = F.relu(self.convf1(flow))
        flo = F.relu(self.convf2(flo))
        out = F.relu(self.conv(torch.cat([cor, flo], dim=1)))
        return torch.cat([out, flow], dim=1)


class SmallMotionEncoder(nn.Module):
    """corr->1x1(96); flow->7x7(64)->3x3(32); concat->3x3(80); +flow -> 82ch
    (model_utils.py:121-129)."""

    def __init__(self, corr_channels: int):
        super().__init__()
        self.convc1 = Conv2dTF(corr_channels, 96, 1)
        self.convf1 = Conv2dTF(2, 64, 7)
        self.convf2 = Conv2dTF(64, 32, 3)
        self.conv = Conv2dTF(96 + 32, 80, 3)

    def forward(self, flow, corr):
        cor = F.relu(self.convc1(corr))
        flo = F.relu(self.convf1(flow))
        flo = F.relu(self.convf2(flo))
        out = F.relu(self.conv(torch.cat([cor, flo], dim=1)))
        return torch.cat([out, flow], dim=1)


class FlowHead(nn.Module):
    """3x3(hidden)->relu->3x3(2) = delta-flow (model_utils.py:131-135)."""

    def __init__(self, input_dim: int, hidden_dim: int):
        super().__init__()
        self.conv1 = Conv2dTF(input_dim, hidden_dim, 3)
        self.conv2 = Conv2dTF(hidden_dim, 2, 3)

    def forward(self, x):
        return self.conv2(F.relu(self.conv1(x)))


class SepConvGRU(nn.Module):
    """Two sequential GRU passes: horizontal 1x5 then vertical 5x1
    (model_utils.py:138-156).  Gate math h' = (1-σ(z))h + σ(z)tanh(q) is
    fused via ops.gru_gates; the candidate conv consumes [r*h, x]."""

    def __init__(self, hidden_dim: int, input_dim: int):
        super().__init__()
        cat_dim = hidden_dim + input_dim
        self.convz1 = Conv2dTF(cat_dim, hidden_dim, (1, 5))
        self.convr1 = Conv2dTF(cat_dim, hidden_dim, (1, 5))
        self.convq1 = Conv2dTF(cat_dim, hidden_dim, (1, 5))
        self.convz2 = Conv2dTF(cat_dim, hidden_dim, (5, 1))
        self.convr2 = Conv2dTF(cat_dim, hidden_dim, (5, 1))
        self.convq2 = Conv2dTF(cat_dim, hidden_dim, (5, 1))

    @staticmethod
    def _pass(h, x, convz, convr, convq):
        hx = torch.cat([h, x], dim=1)
        z_act = convz(hx)
        r = torch.sigmoid(convr(hx))
        q_act = convq(torch.cat([r * h, x], dim=1))
        return ops.gru_gates(h, z_act, q_act)

    def forward(self, h, x):
        h = self._pass(h, x, self.convz1, self.convr1, self.convq1)
        h = self._pass(h, x, self.convz2, self.convr2, self.convq2)
        return h


class ConvGRU(nn.Module):
    """Single 3x3 GRU (model_utils.py:158-169)."""

    def __init__(self, hidden_dim: int, input_dim: int):
        super().__init__()
        cat_dim = hidden_dim + input_dim
        self.convz = Conv2dTF(cat_dim, hidden_dim, 3)
        self.convr = Conv2dTF(cat_dim, hidden_dim, 3)
        self.convq = Conv2dTF(cat_dim, hidden_dim, 3)

    def forward(self, h, x):
        hx = torch.cat([h, x], dim=1)
        z_act = self.convz(hx)
        r = torch.sigmoid(self.convr(hx))
        q_act = self.convq(torch.cat([r * h, x], dim=1))
        return ops.gru_gates(h, z_act, q_act)


class BasicUpdateBlock(nn.Module):
    """Motion enc -> concat context -> SepConvGRU(128) -> FlowHead(256) +
    mask head 3x3(256)->1x1(576), scaled x0.25 (model_utils.py:172-185)."""

    def __init__(self, corr_channels: int, hidden_dim: int = 128,
                 context_dim: int = 128):
        super().__init__()
        self.encoder = BasicMotionEncoder(corr_channels)
        self.gru = SepConvGRU(hidden_dim, input_dim=context_dim + 128)
        self.flow_head = FlowHead(hidden_dim, hidden_dim=256)
        # Sequential indices 0/2 give checkpoint keys mask.0 / mask.2
        # matching TF scopes update_block/mask/{0,2} (model_utils.py:181-182).
        self.mask = nn.Sequential(
            Conv2dTF(hidden_dim, 256, 3), nn.ReLU(inplace=True),
            Conv2dTF(256, 64 * 9, 1),
        )

    def forward(self, net, inp, corr, flow):
        motion = self.encoder(flow, corr)
        x = torch.cat([inp, motion], dim=1)
        net = self.gru(net, x)
        delta_flow = self.flow_head(net)
        mask = 0.25 * self.mask(net)   # x0.25 scale: model_utils.py:183
        return net, mask, delta_flow


class SmallUpdateBlock(nn.Module):
    """Motion enc -> ConvGRU(96) -> FlowHead(128); no mask head
    (model_utils.py:187-194)."""

    def __init__(self, corr_channels: int, hidden_dim: int = 96,
                 context_dim: int = 64):
        super().__init__()
        self.encoder = SmallMotionEncoder(corr_channels)
        self.gru = ConvGRU(hidden_dim, input_dim=context_dim + 82)
        self.flow_head = FlowHead(hidden_dim, hidden_dim=128)

    def forward(self, net, inp, corr, flow):
        motion = self.encoder(flow, corr)
        x = torch.cat([inp, motion], dim=1)
        net = self.gru(net, x)
        delta_flow = self.flow_head(net)
        return net, None, delta_flow
