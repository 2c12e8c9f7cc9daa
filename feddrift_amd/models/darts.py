"""DARTS search space for FedNAS — the full 8-primitive cell search.

Re-expresses the reference search space
(fedml_api/model/cv/darts/genotypes.py:5-14 PRIMITIVES,
operations.py:4-13 OPS, model_search.py Cell/Network:26-125,258-300)
for the engine: 8 candidate ops per edge (none / max-pool / avg-pool /
skip / separable 3x3 / separable 5x5 / dilated 3x3 / dilated 5x5), cells
of 4 intermediate nodes with 14 mixed edges, normal + reduction alpha
matrices [14, 8], and the reference's genotype derivation (top-2 input
edges per node by the best non-'none' weight).
"""

from __future__ import annotations

from typing import List, Tuple

import torch
import torch.nn.functional as F
from torch import nn

PRIMITIVES = [
    "none",
    "max_pool_3x3",
    "avg_pool_3x3",
    "skip_connect",
    "sep_conv_3x3",
    "sep_conv_5x5",
    "dil_conv_3x3",
    "dil_conv_5x5",
]


class Zero(nn.Module):
    def __init__(self, stride: int):
        super().__init__()
        self.stride = stride

    def forward(self, x):
        if self.stride == 1:
            return x * 0.0
        return x[:, :, ::self.stride, ::self.stride] * 0.0


class Identity(nn.Module):
    def forward(self, x):
        return x


class FactorizedReduce(nn.Module):
    """Stride-2 channel-preserving reduce (two offset 1x1 convs)."""

    def __init__(self, c_in: int, c_out: int, affine: bool = True):
        super().__init__()
        assert c_out % 2 == 0
        self.relu = nn.ReLU(inplace=False)
        self.conv_1 = nn.Conv2d(c_in, c_out // 2, 1, stride=2, bias=False)
        self.conv_2 = nn.Conv2d(c_in, c_out // 2, 1, stride=2, bias=False)
        self.bn = nn.BatchNorm2d(c_out, affine=affine)

    def forward(self, x):
        x = self.relu(x)
        return self.bn(torch.cat(
            [self.conv_1(x), self.conv_2(x[:, :, 1:, 1:])], dim=1))


class ReLUConvBN(nn.Module):
    def __init__(self, c_in, c_out, k, stride, pad, affine=True):
        super().__init__()
        self.op = nn.Sequential(
            nn.ReLU(inplace=False),
            nn.Conv2d(c_in, c_out, k, stride=stride, padding=pad,
                      bias=False),
            nn.BatchNorm2d(c_out, affine=affine))

    def forward(self, x):
        return self.op(x)


def _sep_conv(c, k, stride, pad, affine):
    return nn.Sequential(
        nn.ReLU(inplace=False),
        nn.Conv2d(c, c, k, stride=stride, padding=pad, groups=c,
                  bias=False),
        nn.Conv2d(c, c, 1, bias=False),
        nn.BatchNorm2d(c, affine=affine),
        nn.ReLU(inplace=False),
        nn.Conv2d(c, c, k, stride=1, padding=pad, groups=c, bias=False),
        nn.Conv2d(c, c, 1, bias=False),
        nn.BatchNorm2d(c, affine=affine))


def _dil_conv(c, k, stride, pad, dil, affine):
    return nn.Sequential(
        nn.ReLU(inplace=False),
        nn.Conv2d(c, c, k, stride=stride, padding=pad, dilation=dil,
                  groups=c, bias=False),
        nn.Conv2d(c, c, 1, bias=False),
        nn.BatchNorm2d(c, affine=affine))


def make_op(name: str, c: int, stride: int, affine: bool = False):
    if name == "none":
        return Zero(stride)
    if name == "max_pool_3x3":
        return nn.Sequential(nn.MaxPool2d(3, stride=stride, padding=1),
                             nn.BatchNorm2d(c, affine=False))
    if name == "avg_pool_3x3":
        return nn.Sequential(
            nn.AvgPool2d(3, stride=stride, padding=1,
                         count_include_pad=False),
            nn.BatchNorm2d(c, affine=False))
    if name == "skip_connect":
        return Identity() if stride == 1 else \
            FactorizedReduce(c, c, affine=affine)
    if name == "sep_conv_3x3":
        return _sep_conv(c, 3, stride, 1, affine)
    if name == "sep_conv_5x5":
        return _sep_conv(c, 5, stride, 2, affine)
    if name == "dil_conv_3x3":
        return _dil_conv(c, 3, stride, 2, 2, affine)
    if name == "dil_conv_5x5":
        return _dil_conv(c, 5, stride, 4, 2, affine)
    raise NameError(name)


class MixedOp(nn.Module):
    """All 8 candidate ops on one edge, softmax-mixed by alpha."""

    def __init__(self, c: int, stride: int):
        super().__init__()
        self.ops = nn.ModuleList(
            [make_op(p, c, stride) for p in PRIMITIVES])

    def forward(self, x, weights):
        return sum(w * op(x) for w, op in zip(weights, self.ops))


class DartsCell(nn.Module):
    """4 intermediate nodes; node i mixes edges from the 2 cell inputs
    and every earlier node (2+3+4+5 = 14 mixed edges)."""

    def __init__(self, steps, multiplier, c_pp, c_p, c, reduction,
                 reduction_prev):
        super().__init__()
        self.reduction = reduction
        self.pre0 = FactorizedReduce(c_pp, c, affine=False) \
            if reduction_prev else ReLUConvBN(c_pp, c, 1, 1, 0,
                                              affine=False)
        self.pre1 = ReLUConvBN(c_p, c, 1, 1, 0, affine=False)
        self.steps = steps
        self.multiplier = multiplier
        self.ops = nn.ModuleList()
        for i in range(steps):
            for j in range(2 + i):
                stride = 2 if reduction and j < 2 else 1
                self.ops.append(MixedOp(c, stride))

    def forward(self, s0, s1, weights):
        s0 = self.pre0(s0)
        s1 = self.pre1(s1)
        states = [s0, s1]
        off = 0
        for _ in range(self.steps):
            s = sum(self.ops[off + j](h, weights[off + j])
                    for j, h in enumerate(states))
            off += len(states)
            states.append(s)
        return torch.cat(states[-self.multiplier:], dim=1)


class DartsNetwork(nn.Module):
    """Search supernet: stem -> cells (reduction at 1/3 and 2/3) ->
    classifier, with shared alphas_normal / alphas_reduce."""

    def __init__(self, c: int = 8, num_classes: int = 10, layers: int = 4,
                 steps: int = 4, multiplier: int = 4, in_ch: int = 3):
        super().__init__()
        self.steps = steps
        self.multiplier = multiplier
        c_curr = 3 * c
        self.stem = nn.Sequential(
            nn.Conv2d(in_ch, c_curr, 3, padding=1, bias=False),
            nn.BatchNorm2d(c_curr))
        c_pp, c_p, c_curr = c_curr, c_curr, c
        self.cells = nn.ModuleList()
        reduction_prev = False
        for i in range(layers):
            reduction = i in (layers // 3, 2 * layers // 3)
            if reduction:
                c_curr *= 2
            cell = DartsCell(steps, multiplier, c_pp, c_p, c_curr,
                             reduction, reduction_prev)
            self.cells.append(cell)
            reduction_prev = reduction
            c_pp, c_p = c_p, multiplier * c_curr
        self.global_pool = nn.AdaptiveAvgPool2d(1)
        self.classifier = nn.Linear(c_p, num_classes)
        k = sum(2 + i for i in range(steps))
        self.alphas_normal = nn.Parameter(1e-3 * torch.randn(
            k, len(PRIMITIVES)))
        self.alphas_reduce = nn.Parameter(1e-3 * torch.randn(
            k, len(PRIMITIVES)))

    def arch_parameters(self):
        return [self.alphas_normal, self.alphas_reduce]

    def weight_parameters(self):
        arch = {"alphas_normal", "alphas_reduce"}
        return [p for n, p in self.named_parameters() if n not in arch]

    def forward(self, x):
        s0 = s1 = self.stem(x)
        for cell in self.cells:
            w = F.softmax(
                self.alphas_reduce if cell.reduction
                else self.alphas_normal, dim=-1)
            s0, s1 = s1, cell(s0, s1, w)
        out = self.global_pool(s1).flatten(1)
        return self.classifier(out)

    def genotype(self) -> Tuple[List[Tuple[str, int]],
                                List[Tuple[str, int]]]:
        """Reference derivation (model_search.py:258-300): per node keep
        the 2 input edges with the highest best-non-'none' weight; each
        kept edge contributes its argmax non-'none' primitive."""
        none_idx = PRIMITIVES.index("none")

        def parse(alpha):
            w = torch.softmax(alpha, dim=-1).detach().cpu().numpy()
            gene = []
            start, n = 0, 2
            for _ in range(self.steps):
                rows = w[start:start + n]
                edges = sorted(
                    range(n),
                    key=lambda j: -max(rows[j][k]
                                       for k in range(len(PRIMITIVES))
                                       if k != none_idx))[:2]
                for j in sorted(edges):
                    k_best = max(
                        (k for k in range(len(PRIMITIVES))
                         if k != none_idx),
                        key=lambda k: rows[j][k])
                    gene.append((PRIMITIVES[k_best], j))
                start += n
                n += 1
            return gene

        return parse(self.alphas_normal), parse(self.alphas_reduce)
