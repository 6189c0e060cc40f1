"""Frame/state encoders.

Golden eager definitions (PyTorch modules).  On a GPU with the HIP extension
loaded, the compute engine (r2d2_amd.ops) runs the same math through
hand-written gfx950 kernels; these modules own the canonical parameters.

- NatureCNN: the DQN/Nature encoder the reference uses (reference:
  /root/reference/model.py:39-49) with a parameterized input channel count
  (the reference hardcodes 1 grayscale channel; BASELINE.json's benchmark
  convention is 4).
- MLPEncoder: CartPole plumbing config (BASELINE.json configs[0]).
- ImpalaCNN: the IMPALA deep ResNet encoder (BASELINE.json configs[4]).
"""

import torch
import torch.nn as nn
import torch.nn.functional as F


class NatureCNN(nn.Module):
    """Conv(8x8 s4) -> Conv(4x4 s2) -> Conv(3x3 s1) -> Linear(3136->hidden).

    Input: (N, C, 84, 84) float in [0,1].  Output: (N, hidden)."""

    def __init__(self, in_channels: int = 4, hidden_dim: int = 512):
        super().__init__()
        self.conv1 = nn.Conv2d(in_channels, 32, 8, 4)
        self.conv2 = nn.Conv2d(32, 64, 4, 2)
        self.conv3 = nn.Conv2d(64, 64, 3, 1)
        self.fc = nn.Linear(64 * 7 * 7, hidden_dim)
        self.out_dim = hidden_dim

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = F.relu(self.conv1(x), inplace=True)
        x = F.relu(self.conv2(x), inplace=True)
        x = F.relu(self.conv3(x), inplace=True)
        x = x.flatten(1)
        return F.relu(self.fc(x), inplace=True)


class MLPEncoder(nn.Module):
    """Two-layer MLP for low-dimensional observations (CartPole)."""

    def __init__(self, obs_dim: int = 4, hidden_dim: int = 128):
        super().__init__()
        self.fc1 = nn.Linear(obs_dim, hidden_dim)
        self.fc2 = nn.Linear(hidden_dim, hidden_dim)
        self.out_dim = hidden_dim

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = F.relu(self.fc1(x), inplace=True)
        return F.relu(self.fc2(x), inplace=True)


class _ResidualBlock(nn.Module):
    def __init__(self, ch: int):
        super().__init__()
        self.conv1 = nn.Conv2d(ch, ch, 3, 1, padding=1)
        self.conv2 = nn.Conv2d(ch, ch, 3, 1, padding=1)

    def forward(self, x):
        y = self.conv1(F.relu(x))
        y = self.conv2(F.relu(y))
        return x + y


class _ImpalaStage(nn.Module):
    def __init__(self, in_ch: int, out_ch: int):
        super().__init__()
        self.conv = nn.Conv2d(in_ch, out_ch, 3, 1, padding=1)
        self.res1 = _ResidualBlock(out_ch)
        self.res2 = _ResidualBlock(out_ch)

    def forward(self, x):
        x = self.conv(x)
        x = F.max_pool2d(x, 3, stride=2, padding=1)
        x = self.res1(x)
        return self.res2(x)


class ImpalaCNN(nn.Module):
    """IMPALA-deep ResNet encoder (Espeholt et al. 2018): 3 stages of
    conv + maxpool + 2 residual blocks, channels (16, 32, 32).
    84x84 input -> 32 x 11 x 11 -> Linear -> hidden."""

    def __init__(self, in_channels: int = 4, hidden_dim: int = 512):
        super().__init__()
        chans = (16, 32, 32)
        stages = []
        c = in_channels
        for oc in chans:
            stages.append(_ImpalaStage(c, oc))
            c = oc
        self.stages = nn.Sequential(*stages)
        self.fc = nn.Linear(32 * 11 * 11, hidden_dim)
        self.out_dim = hidden_dim

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.stages(x)
        x = F.relu(x)
        x = x.flatten(1)
        return F.relu(self.fc(x), inplace=True)


def make_encoder(kind: str, obs_shape, hidden_dim: int, mlp_hidden: int = 128):
    if kind == "mlp":
        return MLPEncoder(obs_shape[0], mlp_hidden)
    if kind == "nature":
        return NatureCNN(obs_shape[0], hidden_dim)
    if kind == "impala":
        return ImpalaCNN(obs_shape[0], hidden_dim)
    raise ValueError(f"unknown encoder {kind!r}")
