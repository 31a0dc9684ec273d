"""NASNet-A-style cells for the CIFAR conv search space.

MI355X-native re-implementation of the structures the reference ports from
slim (research/improve_nas/trainer/nasnet.py:250-530 build_nasnet_cifar,
nasnet_utils.py: factorized_reduction :94, drop_path :137,
_stacked_separable_conv :182, NasNetABaseCell :318) — written from the
NASNet-A paper genotype, not translated: convolutions/pooling run through
torch (MIOpen on ROCm — library conv, the sanctioned path for non-headline
ops), batch-norm statistics compute in fp32 under bf16 activations, and
the classifier head + optimizers use the adanet_amd HIP kernels.
"""

from __future__ import annotations

import math
from typing import List, Optional

import torch
import torch.nn.functional as F
from torch import nn

from adanet_amd.ops.linear import HipLinear


class BNfp32(nn.Module):
    """BatchNorm with fp32 statistics/affine under bf16 activations —
    hand-written CDNA4 kernels on GPU (csrc/batchnorm.hip, K8), fp32
    F.batch_norm on CPU (inside HipBatchNorm2d)."""

    def __init__(self, c, momentum=0.1, eps=1e-3):
        super().__init__()
        from adanet_amd.ops.batchnorm import HipBatchNorm2d
        self.bn = HipBatchNorm2d(c, momentum=momentum, eps=eps)

    def forward(self, x):
        return self.bn(x)


def drop_path(x, keep_prob: float, training: bool):
    """Per-sample stochastic branch drop (reference nasnet_utils.py:137)."""
    if not training or keep_prob >= 1.0:
        return x
    # Fold 1/keep into the tiny [B,1,1,1] mask so the full-size tensor is
    # touched ONCE (x * mask / keep was two full-tensor kernels fwd and two
    # in backward — the broadcast mul soup was 4%+ of the NASNet step).
    mask = (torch.rand(x.shape[0], 1, 1, 1, device=x.device)
            < keep_prob).to(x.dtype)
    return x * (mask * (1.0 / keep_prob))


class SepConv(nn.Module):
    """Two stacked depthwise-separable convs, each ReLU->dw->pw->BN
    (reference _stacked_separable_conv, nasnet_utils.py:182-248)."""

    def __init__(self, c_in, c_out, kernel, stride):
        super().__init__()
        from adanet_amd.ops.conv import HipConv1x1, HipDepthwiseConv2d
        pad = kernel // 2
        # Fully native separable conv: depthwise on the direct kernel
        # (csrc/depthwise.hip, any channel count) + pointwise on the
        # batched MFMA GEMM when channels align (ops/conv.py).
        self.op = nn.Sequential(
            nn.ReLU(),
            HipDepthwiseConv2d(c_in, kernel, stride=stride, padding=pad),
            HipConv1x1(c_in, c_out, bias=False),
            BNfp32(c_out),
            nn.ReLU(),
            HipDepthwiseConv2d(c_out, kernel, stride=1, padding=pad),
            HipConv1x1(c_out, c_out, bias=False),
            BNfp32(c_out),
        )

    def forward(self, x):
        return self.op(x)


class ReluConvBN(nn.Module):

    def __init__(self, c_in, c_out, kernel=1, stride=1):
        super().__init__()
        from adanet_amd.ops.conv import HipConv1x1
        from adanet_amd.ops.conv import HipConvNxN
        conv = (HipConv1x1(c_in, c_out, bias=False)
                if kernel == 1 and stride == 1 else
                HipConvNxN(c_in, c_out, kernel, stride=stride,
                           padding=kernel // 2, bias=False))
        self.op = nn.Sequential(
            nn.ReLU(),
            conv,
            BNfp32(c_out),
        )

    def forward(self, x):
        return self.op(x)


class FactorizedReduction(nn.Module):
    """Stride-2 reduction without information loss: two offset 1x1 paths
    (reference nasnet_utils.py:94-135)."""

    def __init__(self, c_in, c_out):
        super().__init__()
        assert c_out % 2 == 0
        from adanet_amd.ops.conv import HipConvNxN
        self.relu = nn.ReLU()
        # stride-2 1x1 paths via unfold+MFMA GEMM (MIOpen's naive bf16
        # wrw kernel was 6.9 ms/call on these; ops/conv.py HipConvNxN).
        self.p1 = HipConvNxN(c_in, c_out // 2, 1, stride=2, bias=False)
        self.p2 = HipConvNxN(c_in, c_out - c_out // 2, 1, stride=2,
                             bias=False)
        self.bn = BNfp32(c_out)

    def forward(self, x):
        x = self.relu(x)
        a = self.p1(x)
        b = self.p2(F.pad(x, (0, 1, 0, 1))[:, :, 1:, 1:])
        return self.bn(torch.cat([a, b], dim=1))


class _Pool(nn.Module):

    def __init__(self, kind, stride):
        super().__init__()
        from adanet_amd.ops.conv import HipPool2d
        # Native 3x3 pooling (csrc/pool.hip) — no MIOpen on the cell path.
        self.pool = HipPool2d(kind, stride)

    def forward(self, x):
        return self.pool(x)


class _Identity(nn.Module):

    def __init__(self, c_in, c_out, stride):
        super().__init__()
        if stride == 1 and c_in == c_out:
            self.op = nn.Identity()
        else:
            self.op = FactorizedReduction(c_in, c_out) if stride == 2 else (
                ReluConvBN(c_in, c_out))

    def forward(self, x):
        return self.op(x)


def _make_op(name, c_in, c_out, stride):
    if name.startswith("sep"):
        k = int(name[3])
        return SepConv(c_in, c_out, k, stride)
    if name in ("avg", "max"):
        ops = [_Pool(name, stride)]
        if c_in != c_out:
            ops.append(ReluConvBN(c_in, c_out))
        return nn.Sequential(*ops)
    if name == "id":
        return _Identity(c_in, c_out, stride)
    raise ValueError(name)


# NASNet-A genotype: per block (op_left, input_left, op_right, input_right)
# where inputs 0 = h_{i-1} (prev-prev), 1 = h_i (prev), 2+ = earlier blocks.
NORMAL_GENOTYPE = [
    ("sep3", 1, "id", 1),
    ("sep3", 0, "sep5", 1),
    ("avg", 1, "id", 0),
    ("avg", 0, "avg", 0),
    ("sep5", 0, "sep3", 0),
]
REDUCTION_GENOTYPE = [
    ("sep7", 0, "sep5", 1),
    ("max", 1, "sep7", 0),
    ("avg", 1, "sep5", 0),
    ("max", 1, "sep3", 2),
    ("avg", 2, "id", 3),
]


class NasNetACell(nn.Module):
    """One NASNet-A cell (reference NasNetABaseCell.__call__,
    nasnet_utils.py:318+). Block outputs concatenate channel-wise."""

    def __init__(self, c_prev_prev, c_prev, filters, reduction: bool,
                 prev_reduction: bool, drop_path_keep: float = 1.0):
        super().__init__()
        self.reduction = reduction
        self.drop_path_keep = drop_path_keep
        genotype = REDUCTION_GENOTYPE if reduction else NORMAL_GENOTYPE
        # input adaptors: bring both inputs to `filters` channels (and
        # halve h_{i-1} spatially if the previous cell reduced).
        if prev_reduction:
            self.pre0 = FactorizedReduction(c_prev_prev, filters)
        else:
            self.pre0 = ReluConvBN(c_prev_prev, filters)
        self.pre1 = ReluConvBN(c_prev, filters)
        self.ops_left = nn.ModuleList()
        self.ops_right = nn.ModuleList()
        self.idx = []
        for (op_l, in_l, op_r, in_r) in genotype:
            stride_l = 2 if (reduction and in_l < 2) else 1
            stride_r = 2 if (reduction and in_r < 2) else 1
            self.ops_left.append(_make_op(op_l, filters, filters, stride_l))
            self.ops_right.append(_make_op(op_r, filters, filters, stride_r))
            self.idx.append((in_l, in_r))
        self.out_channels = filters * len(genotype)

    def forward(self, h_prev_prev, h_prev):
        states = [self.pre0(h_prev_prev), self.pre1(h_prev)]
        for i, (op_l, op_r) in enumerate(zip(self.ops_left, self.ops_right)):
            in_l, in_r = self.idx[i]
            left = op_l(states[in_l])
            right = op_r(states[in_r])
            left = drop_path(left, self.drop_path_keep, self.training)
            right = drop_path(right, self.drop_path_keep, self.training)
            states.append(left + right)
        return torch.cat(states[2:], dim=1)


class NasNetCIFAR(nn.Module):
    """build_nasnet_cifar analog (reference nasnet.py:300): stem 3x3 conv,
    num_cells normal cells with reduction cells at the 1/3 and 2/3 marks,
    global average pool, HipLinear classifier."""

    def __init__(self, num_cells: int = 3, num_conv_filters: int = 10,
                 num_classes: int = 10, in_channels: int = 3,
                 stem_multiplier: int = 3, drop_path_keep: float = 0.9):
        super().__init__()
        c_stem = stem_multiplier * num_conv_filters
        self.stem = nn.Sequential(
            __import__("adanet_amd.ops.conv", fromlist=["HipConvNxN"]
                       ).HipConvNxN(in_channels, c_stem, 3, padding=1,
                                    bias=False),
            BNfp32(c_stem))
        self.cells = nn.ModuleList()
        reduction_points = set()
        if num_cells >= 3:
            reduction_points = {num_cells // 3, 2 * num_cells // 3}
        filters = num_conv_filters
        c_pp, c_p = c_stem, c_stem
        prev_red = False
        total = num_cells + len(reduction_points)
        pos = 0
        for i in range(num_cells):
            if i in reduction_points:
                filters *= 2
                cell = NasNetACell(c_pp, c_p, filters, reduction=True,
                                   prev_reduction=prev_red,
                                   drop_path_keep=drop_path_keep)
                self.cells.append(cell)
                c_pp, c_p = c_p, cell.out_channels
                prev_red = True
                pos += 1
            cell = NasNetACell(c_pp, c_p, filters, reduction=False,
                               prev_reduction=prev_red,
                               drop_path_keep=drop_path_keep)
            self.cells.append(cell)
            c_pp, c_p = c_p, cell.out_channels
            prev_red = False
            pos += 1
        self.last_layer_dim = c_p
        self.classifier = HipLinear(c_p, num_classes)

    def forward(self, x):
        if x.dim() == 2:  # flattened CIFAR: [B, 3072] -> [B, 3, 32, 32]
            x = x.reshape(x.shape[0], 3, 32, 32)
        s = self.stem(x)
        h_pp, h_p = s, s
        for cell in self.cells:
            out = cell(h_pp, h_p)
            h_pp, h_p = h_p, out
        feat = F.relu(h_p).mean(dim=(2, 3))
        logits = self.classifier(feat)
        return feat, logits
