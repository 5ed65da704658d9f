"""Small MLP — BASELINE.json config 1 (MNIST-shaped synthetic, CPU mode)."""
import torch.nn as nn


class MLP(nn.Module):
    def __init__(self, in_features=784, hidden=256, num_classes=10,
                 dropout=0.0, num_layers=2):
        super().__init__()
        layers = []
        d = in_features
        for _ in range(num_layers - 1):
            layers += [nn.Linear(d, hidden), nn.ReLU(inplace=True)]
            if dropout > 0:
                layers.append(nn.Dropout(dropout))
            d = hidden
        layers.append(nn.Linear(d, num_classes))
        self.net = nn.Sequential(*layers)

    def forward(self, x):
        return self.net(x.flatten(1))
