"""Actor & critic networks.

State-dict-key parity with the reference (/root/reference/models.py:15-88) is
the checkpoint compatibility contract: both modules expose exactly
``fc1.{weight,bias}, fc2.*, fc2_2.*, fc3.*`` with the same shapes, so ``.pth``
files interchange between the two frameworks.

Architectural quirks of the reference that are part of that contract and are
kept deliberately:
  * the actor has NO activation between fc2 and fc2_2
    (/root/reference/models.py:36-37) — fc2∘fc2_2 is effectively one linear
    composite, but the two-layer shape is what checkpoints carry;
  * the critic injects the action at the second layer by concatenation
    (fc2 input is 256+act_dim wide, /root/reference/models.py:57);
  * init is Normal(0, 1/sqrt(fan_in)) per hidden layer and Normal(0, 3e-3)
    on the output layer (/root/reference/models.py:6-13, 26-30 semantics).

The hidden width is configurable (default 256; BASELINE.json config 5 uses a
1024-wide critic for the MFMA-throughput benchmark).
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F


def fanin_init(tensor: torch.Tensor, fanin: int | None = None) -> torch.Tensor:
    """Normal(0, 1/sqrt(fan_in)) init (reference models.py:6-13 semantics)."""
    fanin = fanin or tensor.size(1)
    std = 1.0 / math.sqrt(fanin)
    with torch.no_grad():
        return tensor.normal_(0.0, std)


class actor(nn.Module):
    """Deterministic policy MLP: obs -> h -> h -> h -> tanh(act).

    Forward graph (reference models.py:32-41): relu(fc1) -> fc2 (no
    activation) -> relu(fc2_2) -> tanh(fc3).  Output is in (-1, 1); the env
    layer's NormalizeAction affine-rescales to the action space.
    """

    def __init__(self, input_size: int, output_size: int, hidden: int = 256):
        super().__init__()
        self.input_size = input_size
        self.output_size = output_size
        self.hidden = hidden
        self.fc1 = nn.Linear(input_size, hidden)
        self.fc2 = nn.Linear(hidden, hidden)
        self.fc2_2 = nn.Linear(hidden, hidden)
        self.fc3 = nn.Linear(hidden, output_size)
        self.init_weights()

    def init_weights(self) -> None:
        fanin_init(self.fc1.weight)
        fanin_init(self.fc2.weight)
        fanin_init(self.fc2_2.weight)
        with torch.no_grad():
            self.fc3.weight.normal_(0.0, 3e-3)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        h = F.relu(self.fc1(x))
        h = self.fc2(h)            # deliberate: no activation (contract quirk)
        h = F.relu(self.fc2_2(h))
        return torch.tanh(self.fc3(h))


class critic(nn.Module):
    """Distributional (C51) critic: (s, a) -> probability vector over atoms.

    Forward graph (reference models.py:76-88): h1 = relu(fc1(s));
    h2 = relu(fc2(cat(h1, a))); h3 = relu(fc2_2(h2)); q = softmax(fc3(h3)).
    ``dist_info`` carries {'type': 'categorical', 'v_min', 'v_max', 'n_atoms'}
    (the 'mixture_of_gaussian' head the reference declares but never
    implements, models.py:63-65, is rejected here with a clear error instead
    of silently passing).
    """

    def __init__(self, state_size: int, action_size: int, dist_info: dict,
                 hidden: int = 256):
        super().__init__()
        if dist_info.get("type", "categorical") != "categorical":
            raise NotImplementedError(
                "only the categorical (C51) critic head exists; "
                "'%s' is not implemented" % dist_info.get("type"))
        self.state_size = state_size
        self.action_size = action_size
        self.hidden = hidden
        self.dist_info = dict(dist_info)
        self.n_atoms = int(dist_info["n_atoms"])
        self.fc1 = nn.Linear(state_size, hidden)
        self.fc2 = nn.Linear(hidden + action_size, hidden)
        self.fc2_2 = nn.Linear(hidden, hidden)
        self.fc3 = nn.Linear(hidden, self.n_atoms)
        self.init_weights()

    def init_weights(self) -> None:
        fanin_init(self.fc1.weight)
        fanin_init(self.fc2.weight)
        fanin_init(self.fc2_2.weight)
        with torch.no_grad():
            self.fc3.weight.normal_(0.0, 3e-3)

    def forward(self, state: torch.Tensor, action: torch.Tensor,
                log: bool = False) -> torch.Tensor:
        h = F.relu(self.fc1(state))
        h = F.relu(self.fc2(torch.cat([h, action], dim=-1)))
        h = F.relu(self.fc2_2(h))
        logits = self.fc3(h)
        if log:
            return F.log_softmax(logits, dim=-1)
        return F.softmax(logits, dim=-1)

    def logits(self, state: torch.Tensor, action: torch.Tensor) -> torch.Tensor:
        h = F.relu(self.fc1(state))
        h = F.relu(self.fc2(torch.cat([h, action], dim=-1)))
        h = F.relu(self.fc2_2(h))
        return self.fc3(h)


def bin_centers(v_min: float, v_max: float, n_atoms: int,
                device=None, dtype=torch.float32) -> torch.Tensor:
    """The C51 atom support z_i = v_min + i*Δ (reference ddpg.py:42-47)."""
    return torch.linspace(v_min, v_max, n_atoms, device=device, dtype=dtype)
