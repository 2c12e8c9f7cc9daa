"""Model zoo for the drift path.

Matches the reference architectures exactly (fedml_api/model/...):
  * LogisticRegression (linear/lr.py:4-11): linear + in-graph sigmoid, CE on
    the sigmoid output — a reference quirk kept for metric parity.
  * FeedForwardNN (fnn/fnn.py:4-15): D -> 2D -> O with ReLU (SEA: 3->6->2).
  * CNN_DropOut (cv/cnn.py:71-135): 784 -> 28x28, conv3x3(32) -> conv3x3(64)
    -> maxpool -> dropout(.25) -> fc128 -> dropout(.5) -> fc10 with an
    IN-GRAPH nn.Softmax before CrossEntropyLoss (double-softmax quirk,
    cv/cnn.py:134) — preserved, not "fixed".
  * resnet18 (torchvision in the reference, main_fedavg.py:219-222):
    torchvision is unavailable offline, so models/resnet.py provides an
    equivalent random-init ResNet-18.

reinitialize() re-seeds torch with the module-level torch_seed and resets
every child layer, so all K ensemble slots start identical
(fedml_api/model/utils.py:7-24).
"""

from __future__ import annotations

import torch
from torch import nn

torch_seed = 42


def set_torch_seed(seed: int) -> None:
    global torch_seed
    torch_seed = seed


class LogisticRegression(nn.Module):
    def __init__(self, input_dim: int, output_dim: int):
        super().__init__()
        self.linear = nn.Linear(input_dim, output_dim)

    def forward(self, x):
        return torch.sigmoid(self.linear(x))


class FeedForwardNN(nn.Module):
    def __init__(self, input_dim: int, output_dim: int, hidden_dim: int):
        super().__init__()
        self.fc1 = nn.Linear(input_dim, hidden_dim)
        self.relu = nn.ReLU()
        self.fc2 = nn.Linear(hidden_dim, output_dim)

    def forward(self, x):
        return self.fc2(self.relu(self.fc1(x)))


class CNN_DropOut(nn.Module):
    def __init__(self, only_digits: bool = True):
        super().__init__()
        self.conv2d_1 = nn.Conv2d(1, 32, kernel_size=3)
        self.max_pooling = nn.MaxPool2d(2, stride=2)
        self.conv2d_2 = nn.Conv2d(32, 64, kernel_size=3)
        self.dropout_1 = nn.Dropout(0.25)
        self.flatten = nn.Flatten()
        self.linear_1 = nn.Linear(9216, 128)
        self.dropout_2 = nn.Dropout(0.5)
        self.linear_2 = nn.Linear(128, 10 if only_digits else 62)
        self.relu = nn.ReLU()
        self.softmax = nn.Softmax(dim=1)

    def forward(self, x):
        # reference forward (cv/cnn.py:126-135): NO activation after either
        # conv — conv1 -> conv2 -> maxpool; the relu member is only used on
        # linear_1. Quirk preserved exactly.
        x = torch.unsqueeze(x.reshape(-1, 28, 28), 1)
        x = self.conv2d_1(x)
        x = self.conv2d_2(x)
        x = self.max_pooling(x)
        x = self.dropout_1(x)
        x = self.flatten(x)
        x = self.relu(self.linear_1(x))
        x = self.dropout_2(x)
        x = self.linear_2(x)
        return self.softmax(x)


def reinitialize(model: nn.Module) -> None:
    torch.manual_seed(torch_seed)
    from .cv_extra import CifarResNet, DenseNet, MobileNet
    from .resnet import FlatImageModel, ResNet
    if isinstance(model, (ResNet, FlatImageModel, CifarResNet, DenseNet,
                          MobileNet)):
        # deep models: recurse (the reference instead reloads pretrained
        # torchvision weights here, utils.py:10-18 — unavailable offline)
        for layer in model.modules():
            if hasattr(layer, "reset_parameters"):
                layer.reset_parameters()
        return
    for layer in model.children():
        if hasattr(layer, "reset_parameters"):
            layer.reset_parameters()


_IMAGE_SHAPES = {3072: (3, 32, 32), 150528: (3, 224, 224)}


def create_model(model_name: str, output_dim: int, feature_dim: int) -> nn.Module:
    if model_name == "lr":
        model = LogisticRegression(feature_dim, output_dim)
    elif model_name == "fnn":
        model = FeedForwardNN(feature_dim, output_dim, feature_dim * 2)
    elif model_name == "cnn":
        model = CNN_DropOut(only_digits=(output_dim <= 10))
    elif model_name in ("resnet", "resnet56", "resnet110", "resnet_gn",
                        "mobilenet", "densenet"):
        from .cv_extra import densenet121, MobileNet, resnet56, resnet110
        from .resnet import FlatImageModel, resnet18
        shape = _IMAGE_SHAPES.get(feature_dim)
        if shape is None:
            raise ValueError(f"no image shape for {feature_dim} features")
        backbones = {
            "resnet": lambda: resnet18(num_classes=output_dim),
            "resnet56": lambda: resnet56(num_classes=output_dim),
            "resnet110": lambda: resnet110(num_classes=output_dim),
            "resnet_gn": lambda: resnet56(num_classes=output_dim,
                                          group_norm=True),
            "mobilenet": lambda: MobileNet(num_classes=output_dim),
            "densenet": lambda: densenet121(num_classes=output_dim),
        }
        model = FlatImageModel(backbones[model_name](), shape)
    elif model_name == "rnn":
        from .rnn import CharLSTM
        # vocab = the dataset's class count (synthetic text drift uses a
        # small alphabet; LEAF shakespeare passes 90)
        model = CharLSTM(vocab_size=max(output_dim, 2),
                         embedding_dim=8,
                         hidden_size=64 if output_dim < 90 else 256)
    else:
        raise NameError(model_name)
    reinitialize(model)
    return model
