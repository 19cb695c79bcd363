"""Feature / context encoders (BasicEncoder, SmallEncoder) and their blocks.

Re-design of networks/model_utils.py:19-105 as NCHW PyTorch modules.  Module
child names are chosen so that ``state_dict()`` keys map 1:1 onto the
reference's TF variable scopes (checkpoint contract):

    fnet/layer2/0/downsample.0/W  <->  fnet.layer2.0.downsample.0.weight

Numerics quirks of the reference that we replicate deliberately:
  * ReLU applied to the residual branch *before* the skip-add and again
    after (model_utils.py:28-31) — a deviation from official RAFT, but it is
    what the converted ``.npz`` weights were validated against.
  * Instance norm has no affine parameters (model_utils.py:13).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from raft_amd.models.layers import Conv2dTF, make_norm


class ResidualBlock(nn.Module):
    """3x3 conv-norm-relu x2 + skip (model_utils.py:19-35).

    Note the reference applies relu to the second conv's output *before*
    the residual add (model_utils.py:28) and relu again after (:31).
    """

    def __init__(self, in_planes: int, out_planes: int, norm_fn: str = "group",
                 stride: int = 1):
        super().__init__()
        self.conv1 = Conv2dTF(in_planes, out_planes, 3, stride=stride)
        self.norm1 = make_norm(norm_fn, out_planes)
        self.conv2 = Conv2dTF(out_planes, out_planes, 3)
        self.norm2 = make_norm(norm_fn, out_planes)
        self.stride = stride
        if stride != 1:
            # Sequential indices 0/1 give checkpoint keys downsample.0/.1
            # matching TF scopes 'downsample.0' / 'downsample.1'
            # (model_utils.py:33-34).
            self.downsample = nn.Sequential(
                Conv2dTF(in_planes, out_planes, 1, stride=stride),
                make_norm(norm_fn, out_planes),
            )
        else:
            self.downsample = None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = F.relu(self.norm1(self.conv1(x)))
        y = F.relu(self.norm2(self.conv2(y)))  # relu BEFORE add: ref quirk
        res = x if self.downsample is None else self.downsample(x)
        return F.relu(res + y)


class BottleneckBlock(nn.Module):
    """1x1(C/4) -> 3x3(C/4, stride) -> 1x1(C) bottleneck (model_utils.py:37-57).

    Same relu-before-add quirk as ResidualBlock (:50-53).
    """

    def __init__(self, in_planes: int, out_planes: int, norm_fn: str = "group",
                 stride: int = 1):
        super().__init__()
        self.conv1 = Conv2dTF(in_planes, out_planes // 4, 1)
        self.norm1 = make_norm(norm_fn, out_planes // 4)
        self.conv2 = Conv2dTF(out_planes // 4, out_planes // 4, 3, stride=stride)
        self.norm2 = make_norm(norm_fn, out_planes // 4)
        self.conv3 = Conv2dTF(out_planes // 4, out_planes, 1)
        self.norm3 = make_norm(norm_fn, out_planes)
        if stride != 1:
            self.downsample = nn.Sequential(
                Conv2dTF(in_planes, out_planes, 1, stride=stride),
                make_norm(norm_fn, out_planes),
            )
        else:
            self.downsample = None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = F.relu(self.norm1(self.conv1(x)))
        y = F.relu(self.norm2(self.conv2(y)))
        y = F.relu(self.norm3(self.conv3(y)))
        res = x if self.downsample is None else self.downsample(x)
        return F.relu(res + y)


class BasicEncoder(nn.Module):
    """Stem 7x7/2(64) -> layers 64/96/128 of 2 ResidualBlocks -> 1x1 proj.

    Output stride 8 (model_utils.py:61-83). The reference shares fnet weights
    across both frames via tf.AUTO_REUSE (:69); callers here batch both
    frames through one call instead.
    """

    def __init__(self, output_dim: int = 256, norm_fn: str = "instance",
                 dropout: float = 0.0):
        super().__init__()
        self.conv1 = Conv2dTF(3, 64, 7, stride=2)
        self.norm1 = make_norm(norm_fn, 64)
        self.layer1 = self._make_layer(64, 64, norm_fn, stride=1)
        self.layer2 = self._make_layer(64, 96, norm_fn, stride=2)
        self.layer3 = self._make_layer(96, 128, norm_fn, stride=2)
        self.conv2 = Conv2dTF(128, output_dim, 1)
        # Reference passes its drop prob to tensorpack keep_prob
        # (model_utils.py:81, a noted TODO-bug); with dropout=0.0 the layer is
        # never built. We implement standard dropout(p) and default to 0.
        self.dropout = nn.Dropout2d(p=dropout) if dropout > 0 else None

    @staticmethod
    def _make_layer(in_planes, dim, norm_fn, stride):
        return nn.Sequential(
            ResidualBlock(in_planes, dim, norm_fn, stride=stride),
            ResidualBlock(dim, dim, norm_fn, stride=1),
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = F.relu(self.norm1(self.conv1(x)))
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.conv2(x)
        if self.dropout is not None:
            x = self.dropout(x)
        return x


class SmallEncoder(nn.Module):
    """Stem 7x7/2(32) -> Bottleneck layers 32/64/96 -> 1x1 proj
    (model_utils.py:85-105)."""

    def __init__(self, output_dim: int = 128, norm_fn: str = "batch",
                 dropout: float = 0.0):
        super().__init__()
        self.conv1 = Conv2dTF(3, 32, 7, stride=2)
        self.norm1 = make_norm(norm_fn, 32)
        self.layer1 = self._make_layer(32, 32, norm_fn, stride=1)
        self.layer2 = self._make_layer(32, 64, norm_fn, stride=2)
        self.layer3 = self._make_layer(64, 96, norm_fn, stride=2)
        self.conv2 = Conv2dTF(96, output_dim, 1)
        self.dropout = nn.Dropout2d(p=dropout) if dropout > 0 else None

    @staticmethod
    def _make_layer(in_planes, dim, norm_fn, stride):
        return nn.Sequential(
            BottleneckBlock(in_planes, dim, norm_fn, stride=stride),
            BottleneckBlock(dim, dim, norm_fn, stride=1),
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = F.relu(self.norm1(self.conv1(x)))
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.conv2(x)
        if self.dropout is not None:
            x = self.dropout(x)
        return x
