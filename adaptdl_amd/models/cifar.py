"""CIFAR-10 model zoo (counterpart of the reference's
examples/pytorch-cifar/models/* — 15 standard architectures, written
from the published papers' architectures for 32x32 inputs).

3x3 convolutions use :class:`FusedConv2d` so shapes that qualify run
the MFMA weight-gradient / stride-2 backward-data kernels and the rest
fall back to MIOpen transparently; BatchNorm+ReLU pairs use
:class:`FusedBatchNormAct2d` on the main trunks.  Everything works in
fp32/CPU too (the fused paths self-gate on device/dtype/layout).
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from adaptdl_amd.torch.layers import FusedBatchNormAct2d, FusedConv2d

__all__ = [
    "LeNet", "VGG", "VGG11", "VGG13", "VGG16", "VGG19",
    "PreActResNet18", "PreActResNet34", "GoogLeNet", "DenseNet121",
    "ResNeXt29_2x64d", "MobileNet", "MobileNetV2", "DPN26", "DPN92",
    "ShuffleNetG2", "ShuffleNetV2", "SENet18", "PNASNetA", "PNASNetB",
    "CIFAR_MODELS",
]


def _conv3x3(cin, cout, stride=1):
    return FusedConv2d(cin, cout, 3, stride=stride, padding=1, bias=False)


def _bn(c, relu=False):
    return FusedBatchNormAct2d(c, relu=relu)


# ---------------------------------------------------------------- LeNet

class LeNet(nn.Module):
    def __init__(self, num_classes=10):
        super().__init__()
        self.conv1 = nn.Conv2d(3, 6, 5)
        self.conv2 = nn.Conv2d(6, 16, 5)
        self.fc1 = nn.Linear(16 * 5 * 5, 120)
        self.fc2 = nn.Linear(120, 84)
        self.fc3 = nn.Linear(84, num_classes)

    def forward(self, x):
        x = F.max_pool2d(F.relu(self.conv1(x)), 2)
        x = F.max_pool2d(F.relu(self.conv2(x)), 2)
        x = x.flatten(1)
        return self.fc3(F.relu(self.fc2(F.relu(self.fc1(x)))))


# ----------------------------------------------------------------- VGG

_VGG_CFG = {
    11: [64, "M", 128, "M", 256, 256, "M", 512, 512, "M", 512, 512, "M"],
    13: [64, 64, "M", 128, 128, "M", 256, 256, "M", 512, 512, "M",
         512, 512, "M"],
    16: [64, 64, "M", 128, 128, "M", 256, 256, 256, "M", 512, 512, 512,
         "M", 512, 512, 512, "M"],
    19: [64, 64, "M", 128, 128, "M", 256, 256, 256, 256, "M", 512, 512,
         512, 512, "M", 512, 512, 512, 512, "M"],
}


class VGG(nn.Module):
    def __init__(self, depth=16, num_classes=10):
        super().__init__()
        layers, cin = [], 3
        for v in _VGG_CFG[depth]:
            if v == "M":
                layers.append(nn.MaxPool2d(2))
            else:
                layers += [_conv3x3(cin, v), _bn(v, relu=True)]
                cin = v
        self.features = nn.Sequential(*layers)
        self.classifier = nn.Linear(512, num_classes)

    def forward(self, x):
        return self.classifier(self.features(x).flatten(1))


def VGG11(**kw):
    return VGG(11, **kw)


def VGG13(**kw):
    return VGG(13, **kw)


def VGG16(**kw):
    return VGG(16, **kw)


def VGG19(**kw):
    return VGG(19, **kw)


# -------------------------------------------------------- PreActResNet

class _PreActBlock(nn.Module):
    def __init__(self, cin, cout, stride=1):
        super().__init__()
        self.bn1 = nn.BatchNorm2d(cin)
        self.conv1 = _conv3x3(cin, cout, stride)
        self.bn2 = nn.BatchNorm2d(cout)
        self.conv2 = _conv3x3(cout, cout)
        self.shortcut = None
        if stride != 1 or cin != cout:
            self.shortcut = nn.Conv2d(cin, cout, 1, stride=stride,
                                      bias=False)

    def forward(self, x):
        out = F.relu(self.bn1(x))
        sc = self.shortcut(out) if self.shortcut is not None else x
        out = self.conv1(out)
        out = self.conv2(F.relu(self.bn2(out)))
        return out + sc


class _PreActResNet(nn.Module):
    def __init__(self, blocks, num_classes=10):
        super().__init__()
        self.conv1 = _conv3x3(3, 64)
        layers, cin = [], 64
        for cout, n, stride in zip((64, 128, 256, 512), blocks,
                                   (1, 2, 2, 2)):
            for s in [stride] + [1] * (n - 1):
                layers.append(_PreActBlock(cin, cout, s))
                cin = cout
        self.layers = nn.Sequential(*layers)
        self.bn = nn.BatchNorm2d(512)
        self.linear = nn.Linear(512, num_classes)

    def forward(self, x):
        out = self.layers(self.conv1(x))
        out = F.adaptive_avg_pool2d(F.relu(self.bn(out)), 1)
        return self.linear(out.flatten(1))


def PreActResNet18(**kw):
    return _PreActResNet((2, 2, 2, 2), **kw)


def PreActResNet34(**kw):
    return _PreActResNet((3, 4, 6, 3), **kw)


# ------------------------------------------------------------ GoogLeNet

class _Inception(nn.Module):
    def __init__(self, cin, c1, c3r, c3, c5r, c5, pp):
        super().__init__()
        self.b1 = nn.Sequential(nn.Conv2d(cin, c1, 1, bias=False),
                                _bn(c1, relu=True))
        self.b2 = nn.Sequential(nn.Conv2d(cin, c3r, 1, bias=False),
                                _bn(c3r, relu=True),
                                _conv3x3(c3r, c3), _bn(c3, relu=True))
        self.b3 = nn.Sequential(nn.Conv2d(cin, c5r, 1, bias=False),
                                _bn(c5r, relu=True),
                                _conv3x3(c5r, c5), _bn(c5, relu=True),
                                _conv3x3(c5, c5), _bn(c5, relu=True))
        self.b4 = nn.Sequential(nn.MaxPool2d(3, 1, 1),
                                nn.Conv2d(cin, pp, 1, bias=False),
                                _bn(pp, relu=True))

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.b3(x),
                          self.b4(x)], 1)


class GoogLeNet(nn.Module):
    def __init__(self, num_classes=10):
        super().__init__()
        self.stem = nn.Sequential(_conv3x3(3, 192), _bn(192, relu=True))
        self.a3 = _Inception(192, 64, 96, 128, 16, 32, 32)
        self.b3 = _Inception(256, 128, 128, 192, 32, 96, 64)
        self.a4 = _Inception(480, 192, 96, 208, 16, 48, 64)
        self.b4 = _Inception(512, 160, 112, 224, 24, 64, 64)
        self.c4 = _Inception(512, 128, 128, 256, 24, 64, 64)
        self.d4 = _Inception(512, 112, 144, 288, 32, 64, 64)
        self.e4 = _Inception(528, 256, 160, 320, 32, 128, 128)
        self.a5 = _Inception(832, 256, 160, 320, 32, 128, 128)
        self.b5 = _Inception(832, 384, 192, 384, 48, 128, 128)
        self.pool = nn.MaxPool2d(3, 2, 1)
        self.linear = nn.Linear(1024, num_classes)

    def forward(self, x):
        out = self.b3(self.a3(self.stem(x)))
        out = self.pool(out)
        out = self.e4(self.d4(self.c4(self.b4(self.a4(out)))))
        out = self.pool(out)
        out = self.b5(self.a5(out))
        out = F.adaptive_avg_pool2d(out, 1)
        return self.linear(out.flatten(1))


# ------------------------------------------------------------- DenseNet

class _DenseLayer(nn.Module):
    def __init__(self, cin, growth):
        super().__init__()
        self.bn1 = nn.BatchNorm2d(cin)
        self.conv1 = nn.Conv2d(cin, 4 * growth, 1, bias=False)
        self.bn2 = nn.BatchNorm2d(4 * growth)
        self.conv2 = _conv3x3(4 * growth, growth)

    def forward(self, x):
        out = self.conv1(F.relu(self.bn1(x)))
        out = self.conv2(F.relu(self.bn2(out)))
        return torch.cat([x, out], 1)


class DenseNet121(nn.Module):
    def __init__(self, growth=32, num_classes=10):
        super().__init__()
        cfg = (6, 12, 24, 16)
        c = 2 * growth
        self.stem = _conv3x3(3, c)
        blocks = []
        for i, n in enumerate(cfg):
            for _ in range(n):
                blocks.append(_DenseLayer(c, growth))
                c += growth
            if i < len(cfg) - 1:
                cout = c // 2
                blocks += [nn.BatchNorm2d(c), nn.ReLU(inplace=True),
                           nn.Conv2d(c, cout, 1, bias=False),
                           nn.AvgPool2d(2)]
                c = cout
        self.blocks = nn.Sequential(*blocks)
        self.bn = nn.BatchNorm2d(c)
        self.linear = nn.Linear(c, num_classes)

    def forward(self, x):
        out = self.blocks(self.stem(x))
        out = F.adaptive_avg_pool2d(F.relu(self.bn(out)), 1)
        return self.linear(out.flatten(1))


# -------------------------------------------------------------- ResNeXt

class _ResNeXtBlock(nn.Module):
    expansion = 2

    def __init__(self, cin, cardinality, bwidth, stride):
        super().__init__()
        group = cardinality * bwidth
        cout = self.expansion * group
        self.conv1 = nn.Conv2d(cin, group, 1, bias=False)
        self.bn1 = _bn(group, relu=True)
        self.conv2 = nn.Conv2d(group, group, 3, stride=stride, padding=1,
                               groups=cardinality, bias=False)
        self.bn2 = _bn(group, relu=True)
        self.conv3 = nn.Conv2d(group, cout, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(cout)
        self.shortcut = None
        if stride != 1 or cin != cout:
            self.shortcut = nn.Sequential(
                nn.Conv2d(cin, cout, 1, stride=stride, bias=False),
                nn.BatchNorm2d(cout))

    def forward(self, x):
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        out = self.bn3(self.conv3(out))
        sc = self.shortcut(x) if self.shortcut is not None else x
        return F.relu(out + sc)


class ResNeXt29_2x64d(nn.Module):
    def __init__(self, num_classes=10):
        super().__init__()
        cardinality, bwidth = 2, 64
        self.stem = nn.Sequential(_conv3x3(3, 64), _bn(64, relu=True))
        layers, cin = [], 64
        for stride in (1, 2, 2):
            for s in [stride, 1, 1]:
                layers.append(_ResNeXtBlock(cin, cardinality, bwidth, s))
                cin = _ResNeXtBlock.expansion * cardinality * bwidth
            bwidth *= 2
        self.layers = nn.Sequential(*layers)
        self.linear = nn.Linear(cin, num_classes)

    def forward(self, x):
        out = self.layers(self.stem(x))
        return self.linear(F.adaptive_avg_pool2d(out, 1).flatten(1))


# ------------------------------------------------------------ MobileNet

class _DWSep(nn.Module):
    def __init__(self, cin, cout, stride=1):
        super().__init__()
        self.dw = nn.Conv2d(cin, cin, 3, stride=stride, padding=1,
                            groups=cin, bias=False)
        self.bn1 = _bn(cin, relu=True)
        self.pw = nn.Conv2d(cin, cout, 1, bias=False)
        self.bn2 = _bn(cout, relu=True)

    def forward(self, x):
        return self.bn2(self.pw(self.bn1(self.dw(x))))


class MobileNet(nn.Module):
    _cfg = [64, (128, 2), 128, (256, 2), 256, (512, 2),
            512, 512, 512, 512, 512, (1024, 2), 1024]

    def __init__(self, num_classes=10):
        super().__init__()
        self.stem = nn.Sequential(_conv3x3(3, 32), _bn(32, relu=True))
        layers, cin = [], 32
        for v in self._cfg:
            cout, stride = (v, 1) if isinstance(v, int) else v
            layers.append(_DWSep(cin, cout, stride))
            cin = cout
        self.layers = nn.Sequential(*layers)
        self.linear = nn.Linear(1024, num_classes)

    def forward(self, x):
        out = self.layers(self.stem(x))
        return self.linear(F.adaptive_avg_pool2d(out, 1).flatten(1))


class _InvertedResidual(nn.Module):
    def __init__(self, cin, cout, expand, stride):
        super().__init__()
        mid = cin * expand
        self.use_res = stride == 1 and cin == cout
        self.conv1 = nn.Conv2d(cin, mid, 1, bias=False)
        self.bn1 = _bn(mid, relu=True)
        self.conv2 = nn.Conv2d(mid, mid, 3, stride=stride, padding=1,
                               groups=mid, bias=False)
        self.bn2 = _bn(mid, relu=True)
        self.conv3 = nn.Conv2d(mid, cout, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(cout)

    def forward(self, x):
        out = self.bn3(self.conv3(self.bn2(self.conv2(
            self.bn1(self.conv1(x))))))
        return x + out if self.use_res else out


class MobileNetV2(nn.Module):
    # (expand, cout, repeats, stride) — CIFAR strides (first stage s=1)
    _cfg = [(1, 16, 1, 1), (6, 24, 2, 1), (6, 32, 3, 2), (6, 64, 4, 2),
            (6, 96, 3, 1), (6, 160, 3, 2), (6, 320, 1, 1)]

    def __init__(self, num_classes=10):
        super().__init__()
        self.stem = nn.Sequential(_conv3x3(3, 32), _bn(32, relu=True))
        layers, cin = [], 32
        for expand, cout, n, stride in self._cfg:
            for s in [stride] + [1] * (n - 1):
                layers.append(_InvertedResidual(cin, cout, expand, s))
                cin = cout
        self.layers = nn.Sequential(*layers)
        self.head = nn.Sequential(nn.Conv2d(320, 1280, 1, bias=False),
                                  _bn(1280, relu=True))
        self.linear = nn.Linear(1280, num_classes)

    def forward(self, x):
        out = self.head(self.layers(self.stem(x)))
        return self.linear(F.adaptive_avg_pool2d(out, 1).flatten(1))


# ------------------------------------------------------------------ DPN

class _DPNBlock(nn.Module):
    """Dual-path: residual channels + densely concatenated channels."""

    def __init__(self, last_c, mid, cout, dense, stride, first):
        super().__init__()
        self.cout = cout
        self.conv1 = nn.Conv2d(last_c, mid, 1, bias=False)
        self.bn1 = _bn(mid, relu=True)
        self.conv2 = nn.Conv2d(mid, mid, 3, stride=stride, padding=1,
                               groups=32, bias=False)
        self.bn2 = _bn(mid, relu=True)
        self.conv3 = nn.Conv2d(mid, cout + dense, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(cout + dense)
        self.shortcut = None
        if first:
            self.shortcut = nn.Sequential(
                nn.Conv2d(last_c, cout + dense, 1, stride=stride,
                          bias=False),
                nn.BatchNorm2d(cout + dense))

    def forward(self, x):
        out = self.bn3(self.conv3(self.bn2(self.conv2(
            self.bn1(self.conv1(x))))))
        sc = self.shortcut(x) if self.shortcut is not None else x
        d = self.cout
        res = sc[:, :d] + out[:, :d]
        dense = torch.cat([sc[:, d:], out[:, d:]], 1)
        return F.relu(torch.cat([res, dense], 1))


class _DPN(nn.Module):
    def __init__(self, mids, couts, denses, blocks, num_classes=10):
        super().__init__()
        self.stem = nn.Sequential(_conv3x3(3, 64), _bn(64, relu=True))
        layers, last_c = [], 64
        for i, (mid, cout, dense, n) in enumerate(
                zip(mids, couts, denses, blocks)):
            stride = 1 if i == 0 else 2
            for j in range(n):
                layers.append(_DPNBlock(last_c, mid, cout, dense,
                                        stride if j == 0 else 1, j == 0))
                last_c = cout + (j + 2) * dense
        self.layers = nn.Sequential(*layers)
        self.linear = nn.Linear(last_c, num_classes)

    def forward(self, x):
        out = self.layers(self.stem(x))
        return self.linear(F.adaptive_avg_pool2d(out, 1).flatten(1))


def DPN26(**kw):
    return _DPN((96, 192, 384, 768), (256, 512, 1024, 2048),
                (16, 32, 24, 128), (2, 2, 2, 2), **kw)


def DPN92(**kw):
    return _DPN((96, 192, 384, 768), (256, 512, 1024, 2048),
                (16, 32, 24, 128), (3, 4, 20, 3), **kw)


# ----------------------------------------------------------- ShuffleNet

def _channel_shuffle(x, groups):
    n, c, h, w = x.shape
    return x.view(n, groups, c // groups, h, w) \
        .transpose(1, 2).reshape(n, c, h, w)


class _ShuffleBlock(nn.Module):
    def __init__(self, cin, cout, stride, groups):
        super().__init__()
        self.stride = stride
        mid = cout // 4
        g = 1 if cin == 24 else groups
        self.groups = groups
        cadd = cout - cin if stride == 2 else cout
        self.conv1 = nn.Conv2d(cin, mid, 1, groups=g, bias=False)
        self.bn1 = _bn(mid, relu=True)
        self.conv2 = nn.Conv2d(mid, mid, 3, stride=stride, padding=1,
                               groups=mid, bias=False)
        self.bn2 = nn.BatchNorm2d(mid)
        self.conv3 = nn.Conv2d(mid, cadd, 1, groups=groups, bias=False)
        self.bn3 = nn.BatchNorm2d(cadd)

    def forward(self, x):
        out = self.bn1(self.conv1(x))
        out = _channel_shuffle(out, self.groups)
        out = self.bn3(self.conv3(self.bn2(self.conv2(out))))
        if self.stride == 2:
            return F.relu(torch.cat([F.avg_pool2d(x, 2), out], 1))
        return F.relu(x + out)


class ShuffleNetG2(nn.Module):
    def __init__(self, num_classes=10):
        super().__init__()
        couts, blocks, groups = (200, 400, 800), (4, 8, 4), 2
        self.stem = nn.Sequential(_conv3x3(3, 24), _bn(24, relu=True))
        layers, cin = [], 24
        for cout, n in zip(couts, blocks):
            for j in range(n):
                layers.append(_ShuffleBlock(cin, cout, 2 if j == 0 else 1,
                                            groups))
                cin = cout
        self.layers = nn.Sequential(*layers)
        self.linear = nn.Linear(800, num_classes)

    def forward(self, x):
        out = self.layers(self.stem(x))
        return self.linear(F.adaptive_avg_pool2d(out, 1).flatten(1))


class _ShuffleV2Block(nn.Module):
    def __init__(self, cin, cout, stride):
        super().__init__()
        self.stride = stride
        mid = cout // 2
        branch_in = cin if stride == 2 else cin // 2
        self.b2 = nn.Sequential(
            nn.Conv2d(branch_in, mid, 1, bias=False), _bn(mid, relu=True),
            nn.Conv2d(mid, mid, 3, stride=stride, padding=1, groups=mid,
                      bias=False),
            nn.BatchNorm2d(mid),
            nn.Conv2d(mid, mid, 1, bias=False), _bn(mid, relu=True))
        self.b1 = None
        if stride == 2:
            self.b1 = nn.Sequential(
                nn.Conv2d(cin, cin, 3, stride=2, padding=1, groups=cin,
                          bias=False),
                nn.BatchNorm2d(cin),
                nn.Conv2d(cin, mid, 1, bias=False), _bn(mid, relu=True))

    def forward(self, x):
        if self.stride == 2:
            out = torch.cat([self.b1(x), self.b2(x)], 1)
        else:
            x1, x2 = x.chunk(2, dim=1)
            out = torch.cat([x1, self.b2(x2)], 1)
        return _channel_shuffle(out, 2)


class ShuffleNetV2(nn.Module):
    def __init__(self, num_classes=10):
        super().__init__()
        couts, blocks = (116, 232, 464), (4, 8, 4)   # 1.0x width
        self.stem = nn.Sequential(_conv3x3(3, 24), _bn(24, relu=True))
        layers, cin = [], 24
        for cout, n in zip(couts, blocks):
            for j in range(n):
                layers.append(_ShuffleV2Block(cin, cout,
                                              2 if j == 0 else 1))
                cin = cout
        self.layers = nn.Sequential(*layers)
        self.head = nn.Sequential(nn.Conv2d(464, 1024, 1, bias=False),
                                  _bn(1024, relu=True))
        self.linear = nn.Linear(1024, num_classes)

    def forward(self, x):
        out = self.head(self.layers(self.stem(x)))
        return self.linear(F.adaptive_avg_pool2d(out, 1).flatten(1))


# ---------------------------------------------------------------- SENet

class _SEBlock(nn.Module):
    def __init__(self, cin, cout, stride=1, reduction=16):
        super().__init__()
        self.bn1 = nn.BatchNorm2d(cin)
        self.conv1 = _conv3x3(cin, cout, stride)
        self.bn2 = nn.BatchNorm2d(cout)
        self.conv2 = _conv3x3(cout, cout)
        self.fc1 = nn.Conv2d(cout, cout // reduction, 1)
        self.fc2 = nn.Conv2d(cout // reduction, cout, 1)
        self.shortcut = None
        if stride != 1 or cin != cout:
            self.shortcut = nn.Conv2d(cin, cout, 1, stride=stride,
                                      bias=False)

    def forward(self, x):
        out = F.relu(self.bn1(x))
        sc = self.shortcut(out) if self.shortcut is not None else x
        out = self.conv1(out)
        out = self.conv2(F.relu(self.bn2(out)))
        w = F.adaptive_avg_pool2d(out, 1)
        w = torch.sigmoid(self.fc2(F.relu(self.fc1(w))))
        return out * w + sc


class SENet18(nn.Module):
    def __init__(self, num_classes=10):
        super().__init__()
        self.conv1 = _conv3x3(3, 64)
        layers, cin = [], 64
        for cout, n, stride in zip((64, 128, 256, 512), (2, 2, 2, 2),
                                   (1, 2, 2, 2)):
            for s in [stride] + [1] * (n - 1):
                layers.append(_SEBlock(cin, cout, s))
                cin = cout
        self.layers = nn.Sequential(*layers)
        self.bn = nn.BatchNorm2d(512)
        self.linear = nn.Linear(512, num_classes)

    def forward(self, x):
        out = self.layers(self.conv1(x))
        out = F.adaptive_avg_pool2d(F.relu(self.bn(out)), 1)
        return self.linear(out.flatten(1))


# -------------------------------------------------------------- PNASNet

class _SepConv(nn.Module):
    def __init__(self, cin, cout, k, stride):
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, k, stride=stride,
                              padding=(k - 1) // 2, groups=cin
                              if cin == cout else 1, bias=False)
        self.bn = nn.BatchNorm2d(cout)

    def forward(self, x):
        return self.bn(self.conv(x))


class _PNASCellA(nn.Module):
    def __init__(self, cin, cout, stride):
        super().__init__()
        self.sep = _SepConv(cin, cout, 7, stride)
        self.stride = stride
        self.proj = None
        if stride == 2 or cin != cout:
            self.proj = nn.Sequential(
                nn.Conv2d(cin, cout, 1, stride=stride, bias=False),
                nn.BatchNorm2d(cout))

    def forward(self, x):
        y1 = self.sep(x)
        y2 = self.proj(x) if self.proj is not None else x
        if self.proj is None and self.stride == 2:
            y2 = F.max_pool2d(x, 3, 2, 1)
        return F.relu(y1 + y2)


class _PNASCellB(nn.Module):
    def __init__(self, cin, cout, stride):
        super().__init__()
        self.sep1 = _SepConv(cin, cout, 7, stride)
        self.sep2 = _SepConv(cin, cout, 3, stride)
        self.proj = nn.Sequential(
            nn.Conv2d(cin, cout, 1, stride=stride, bias=False),
            nn.BatchNorm2d(cout))
        self.bn = nn.BatchNorm2d(cout)

    def forward(self, x):
        y1 = self.sep1(x) + self.sep2(x)
        y2 = self.proj(x)
        return F.relu(self.bn(F.relu(y1)) + y2)


class _PNASNet(nn.Module):
    def __init__(self, cell, planes, num_classes=10):
        super().__init__()
        self.stem = nn.Sequential(_conv3x3(3, planes),
                                  _bn(planes, relu=True))
        layers, cin = [], planes
        for stage in range(3):
            if stage > 0:
                layers.append(cell(cin, cin * 2, 2))
                cin *= 2
            for _ in range(6):
                layers.append(cell(cin, cin, 1))
        self.layers = nn.Sequential(*layers)
        self.linear = nn.Linear(cin, num_classes)

    def forward(self, x):
        out = self.layers(self.stem(x))
        return self.linear(F.adaptive_avg_pool2d(out, 1).flatten(1))


def PNASNetA(**kw):
    return _PNASNet(_PNASCellA, 44, **kw)


def PNASNetB(**kw):
    return _PNASNet(_PNASCellB, 32, **kw)


CIFAR_MODELS = {
    "lenet": LeNet, "vgg11": VGG11, "vgg13": VGG13, "vgg16": VGG16,
    "vgg19": VGG19, "preact_resnet18": PreActResNet18,
    "preact_resnet34": PreActResNet34, "googlenet": GoogLeNet,
    "densenet121": DenseNet121, "resnext29": ResNeXt29_2x64d,
    "mobilenet": MobileNet, "mobilenetv2": MobileNetV2, "dpn26": DPN26,
    "dpn92": DPN92, "shufflenet": ShuffleNetG2,
    "shufflenetv2": ShuffleNetV2, "senet18": SENet18,
    "pnasnet_a": PNASNetA, "pnasnet_b": PNASNetB,
}
