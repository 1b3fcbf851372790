"""AlexNet / caffe-style CIFAR CNN / DenseNet / ResNeXt for CIFAR, plus the
2x1500 PTB LSTM — rounding out the reference model zoo
(VGG/models/{alexnet,caffe_cifar,densenet,resnext,lstm}.py; registry at
VGG/models/__init__.py:16-27)."""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


class AlexNet(nn.Module):
    """CIFAR-sized AlexNet."""

    def __init__(self, num_classes: int = 10):
        super().__init__()
        self.features = nn.Sequential(
            nn.Conv2d(3, 64, 3, stride=2, padding=1), nn.ReLU(inplace=True),
            nn.MaxPool2d(2),
            nn.Conv2d(64, 192, 3, padding=1), nn.ReLU(inplace=True),
            nn.MaxPool2d(2),
            nn.Conv2d(192, 384, 3, padding=1), nn.ReLU(inplace=True),
            nn.Conv2d(384, 256, 3, padding=1), nn.ReLU(inplace=True),
            nn.Conv2d(256, 256, 3, padding=1), nn.ReLU(inplace=True),
            nn.MaxPool2d(2),
        )
        self.classifier = nn.Linear(256 * 2 * 2, num_classes)

    def forward(self, x):
        return self.classifier(self.features(x).flatten(1))


class CaffeCifar(nn.Module):
    """The classic caffe cifar10_quick net (reference models/caffe_cifar.py)."""

    def __init__(self, num_classes: int = 10):
        super().__init__()
        self.conv1 = nn.Conv2d(3, 32, 5, padding=2)
        self.conv2 = nn.Conv2d(32, 32, 5, padding=2)
        self.conv3 = nn.Conv2d(32, 64, 5, padding=2)
        self.fc1 = nn.Linear(64 * 4 * 4, 64)
        self.fc2 = nn.Linear(64, num_classes)

    def forward(self, x):
        x = F.max_pool2d(F.relu(self.conv1(x)), 2)
        x = F.max_pool2d(F.relu(self.conv2(x)), 2)
        x = F.max_pool2d(F.relu(self.conv3(x)), 2)
        return self.fc2(F.relu(self.fc1(x.flatten(1))))


class _DenseLayer(nn.Module):
    def __init__(self, cin, growth):
        super().__init__()
        self.bn = nn.BatchNorm2d(cin)
        self.conv = nn.Conv2d(cin, growth, 3, padding=1, bias=False)

    def forward(self, x):
        out = self.conv(F.relu(self.bn(x), inplace=True))
        return torch.cat([x, out], 1)


class DenseNetCifar(nn.Module):
    """DenseNet-BC-style (depth=40, growth=12 default) for CIFAR."""

    def __init__(self, depth: int = 40, growth: int = 12, num_classes: int = 10):
        super().__init__()
        n = (depth - 4) // 3
        ch = 16
        self.conv1 = nn.Conv2d(3, ch, 3, padding=1, bias=False)
        blocks = []
        for b in range(3):
            for _ in range(n):
                blocks.append(_DenseLayer(ch, growth))
                ch += growth
            if b < 2:
                blocks.append(nn.BatchNorm2d(ch))
                blocks.append(nn.ReLU(inplace=True))
                blocks.append(nn.Conv2d(ch, ch // 2, 1, bias=False))
                blocks.append(nn.AvgPool2d(2))
                ch //= 2
        self.features = nn.Sequential(*blocks)
        self.bn = nn.BatchNorm2d(ch)
        self.fc = nn.Linear(ch, num_classes)

    def forward(self, x):
        x = self.features(self.conv1(x))
        x = F.relu(self.bn(x), inplace=True)
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(x)


class _ResNeXtBlock(nn.Module):
    def __init__(self, cin, cout, cardinality=8, base_width=64, stride=1):
        super().__init__()
        width = cout // 2
        self.conv1 = nn.Conv2d(cin, width, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(width)
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1,
                               groups=cardinality, bias=False)
        self.bn2 = nn.BatchNorm2d(width)
        self.conv3 = nn.Conv2d(width, cout, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(cout)
        self.shortcut = None
        if stride != 1 or cin != cout:
            self.shortcut = nn.Sequential(
                nn.Conv2d(cin, cout, 1, stride=stride, bias=False), nn.BatchNorm2d(cout)
            )

    def forward(self, x):
        identity = x if self.shortcut is None else self.shortcut(x)
        out = F.relu(self.bn1(self.conv1(x)), inplace=True)
        out = F.relu(self.bn2(self.conv2(out)), inplace=True)
        out = self.bn3(self.conv3(out))
        return F.relu(out + identity, inplace=True)


class ResNeXtCifar(nn.Module):
    def __init__(self, depth: int = 29, cardinality: int = 8, num_classes: int = 10):
        super().__init__()
        n = (depth - 2) // 9
        self.conv1 = nn.Conv2d(3, 64, 3, padding=1, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        stages = []
        cin = 64
        for i, cout in enumerate((256, 512, 1024)):
            for j in range(n):
                stages.append(
                    _ResNeXtBlock(cin, cout, cardinality, stride=2 if (i > 0 and j == 0) else 1)
                )
                cin = cout
        self.stages = nn.Sequential(*stages)
        self.fc = nn.Linear(1024, num_classes)

    def forward(self, x):
        x = F.relu(self.bn1(self.conv1(x)), inplace=True)
        x = self.stages(x)
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(x)


class PTBLSTM(nn.Module):
    """2-layer 1500-unit word LM (reference models/lstm.py:5: emb 1500,
    2x1500 LSTM, dropout 0.65, tied-capable)."""

    def __init__(self, vocab_size: int = 10000, emb: int = 1500, hidden: int = 1500,
                 layers: int = 2, dropout: float = 0.65):
        super().__init__()
        self.drop = nn.Dropout(dropout)
        self.encoder = nn.Embedding(vocab_size, emb)
        self.rnn = nn.LSTM(emb, hidden, num_layers=layers, dropout=dropout)
        self.decoder = nn.Linear(hidden, vocab_size)
        self.hidden_size = hidden
        self.nlayers = layers

    def forward(self, x, hidden=None):
        emb = self.drop(self.encoder(x))
        out, hidden = self.rnn(emb, hidden)
        out = self.drop(out)
        return self.decoder(out), hidden

    def init_hidden(self, bsz, device=None):
        w = next(self.parameters())
        shape = (self.nlayers, bsz, self.hidden_size)
        return (w.new_zeros(shape), w.new_zeros(shape))


class MnistNet(nn.Module):
    """The 2-conv MNIST net the reference trains for its smallest recipe
    (reference VGG/dl_trainer.py:59-76): conv5x5(1->10) -> pool -> conv5x5
    (10->20)+dropout2d -> pool -> fc320->50 -> dropout -> fc50->10."""

    def __init__(self, num_classes: int = 10):
        super().__init__()
        self.conv1 = nn.Conv2d(1, 10, kernel_size=5)
        self.conv2 = nn.Conv2d(10, 20, kernel_size=5)
        self.conv2_drop = nn.Dropout2d()
        self.fc1 = nn.Linear(320, 50)
        self.fc2 = nn.Linear(50, num_classes)

    def forward(self, x):
        x = torch.relu(torch.max_pool2d(self.conv1(x), 2))
        x = torch.relu(torch.max_pool2d(self.conv2_drop(self.conv2(x)), 2))
        x = x.flatten(1)
        x = torch.relu(self.fc1(x))
        x = nn.functional.dropout(x, training=self.training)
        return self.fc2(x)
