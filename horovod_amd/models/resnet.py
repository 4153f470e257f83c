"""ResNet family for the synthetic ImageNet benchmark.

Own implementation of the standard bottleneck ResNet (He et al., 2015) —
the benchmark model family named by BASELINE.md (ResNet-50/101 synthetic,
tf_cnn_benchmarks-style).  torchvision is not available in this image.

MI355X notes: run under channels_last so MIOpen picks NHWC kernels, and in
bf16 autocast (CDNA4's native matrix dtype).
"""
import torch
import torch.nn as nn


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_planes, planes, stride=1, downsample=None,
                 norm_layer=nn.BatchNorm2d, fused_bn=False):
        super().__init__()
        self.fused_bn = fused_bn
        self.conv1 = nn.Conv2d(in_planes, planes, 1, bias=False)
        self.conv2 = nn.Conv2d(planes, planes, 3, stride=stride, padding=1,
                               bias=False)
        self.conv3 = nn.Conv2d(planes, planes * self.expansion, 1, bias=False)
        if fused_bn:
            from horovod_amd.ops import FusedBNAddReLU, FusedBNReLU
            self.bn1 = FusedBNReLU(planes)
            self.bn2 = FusedBNReLU(planes)
            self.bn3 = FusedBNAddReLU(planes * self.expansion)
        else:
            self.bn1 = norm_layer(planes)
            self.bn2 = norm_layer(planes)
            self.bn3 = norm_layer(planes * self.expansion)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        identity = x
        if self.downsample is not None:
            identity = self.downsample(x)
        if self.fused_bn:
            out = self.bn1(self.conv1(x))
            out = self.bn2(self.conv2(out))
            return self.bn3(self.conv3(out), identity)
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        out += identity
        return self.relu(out)


class ResNet(nn.Module):
    def __init__(self, layers, num_classes=1000, norm_layer=nn.BatchNorm2d,
                 fused_bn=False):
        super().__init__()
        self._norm_layer = norm_layer
        self._fused_bn = fused_bn
        self.in_planes = 64
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = norm_layer(64)
        self.relu = nn.ReLU(inplace=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(64, layers[0])
        self.layer2 = self._make_layer(128, layers[1], stride=2)
        self.layer3 = self._make_layer(256, layers[2], stride=2)
        self.layer4 = self._make_layer(512, layers[3], stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d((1, 1))
        self.fc = nn.Linear(512 * Bottleneck.expansion, num_classes)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.constant_(m.weight, 1)
                nn.init.constant_(m.bias, 0)
        # zero-init last BN in each block (standard trick; improves early
        # training and matches tf_cnn_benchmarks defaults)
        for m in self.modules():
            if isinstance(m, Bottleneck):
                nn.init.constant_(m.bn3.weight, 0)

    def _make_layer(self, planes, blocks, stride=1):
        norm_layer = self._norm_layer
        downsample = None
        if stride != 1 or self.in_planes != planes * Bottleneck.expansion:
            downsample = nn.Sequential(
                nn.Conv2d(self.in_planes, planes * Bottleneck.expansion, 1,
                          stride=stride, bias=False),
                norm_layer(planes * Bottleneck.expansion),
            )
        layers = [Bottleneck(self.in_planes, planes, stride, downsample,
                             norm_layer, fused_bn=self._fused_bn)]
        self.in_planes = planes * Bottleneck.expansion
        for _ in range(1, blocks):
            layers.append(Bottleneck(self.in_planes, planes,
                                     norm_layer=norm_layer,
                                     fused_bn=self._fused_bn))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(self.relu(self.bn1(self.conv1(x))))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


def resnet50(num_classes=1000, **kw):
    return ResNet([3, 4, 6, 3], num_classes, **kw)


def resnet101(num_classes=1000, **kw):
    return ResNet([3, 4, 23, 3], num_classes, **kw)


def resnet152(num_classes=1000, **kw):
    return ResNet([3, 8, 36, 3], num_classes, **kw)
