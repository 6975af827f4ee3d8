"""MNIST distill nets (parity: reference
example/distill/mnist_distill/train_with_fleet.py:55-95 — three student
architectures selectable by nn_type, trained against teacher soft labels
with the same KD soft-label CE the resnet example uses).

Torch-native equivalents of the fluid graphs:
  softmax_regression: flatten -> fc 10
  multilayer_perceptron: flatten -> fc 200 tanh -> fc 200 tanh -> fc 10
  convolutional_neural_network: conv5x5(20) -> maxpool2 -> relu -> BN ->
                                conv5x5(50) -> maxpool2 -> relu -> fc 10
Logits are returned raw (the loss applies softmax/CE)."""
import torch
import torch.nn as nn
import torch.nn.functional as F


class MnistSoftmaxRegression(nn.Module):
    def __init__(self, num_classes=10):
        super().__init__()
        self.fc = nn.Linear(28 * 28, num_classes)

    def forward(self, x):
        return self.fc(x.flatten(1))


class MnistMLP(nn.Module):
    def __init__(self, num_classes=10, hidden=200):
        super().__init__()
        self.fc1 = nn.Linear(28 * 28, hidden)
        self.fc2 = nn.Linear(hidden, hidden)
        self.fc3 = nn.Linear(hidden, num_classes)

    def forward(self, x):
        h = torch.tanh(self.fc1(x.flatten(1)))
        h = torch.tanh(self.fc2(h))
        return self.fc3(h)


class MnistCNN(nn.Module):
    """simple_img_conv_pool x2 + batch_norm after the first block
    (reference :79-95)."""

    def __init__(self, num_classes=10):
        super().__init__()
        self.conv1 = nn.Conv2d(1, 20, 5)
        self.bn1 = nn.BatchNorm2d(20)
        self.conv2 = nn.Conv2d(20, 50, 5)
        self.fc = nn.Linear(50 * 4 * 4, num_classes)

    def forward(self, x):
        h = F.relu(F.max_pool2d(self.conv1(x), 2))
        h = self.bn1(h)
        h = F.relu(F.max_pool2d(self.conv2(h), 2))
        return self.fc(h.flatten(1))
