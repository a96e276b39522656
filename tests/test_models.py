"""Model tests: state-dict key/shape parity (the checkpoint contract,
reference models.py:18-23, 56-62) and forward-graph properties."""

import numpy as np
import pytest
import torch

from d4pg_amd.models import actor, critic, bin_centers, fanin_init

DIST = {"type": "categorical", "v_min": -300.0, "v_max": 0.0, "n_atoms": 51}


def test_actor_state_dict_keys():
    a = actor(3, 1)
    assert sorted(a.state_dict().keys()) == sorted([
        "fc1.weight", "fc1.bias", "fc2.weight", "fc2.bias",
        "fc2_2.weight", "fc2_2.bias", "fc3.weight", "fc3.bias"])
    assert a.state_dict()["fc1.weight"].shape == (256, 3)
    assert a.state_dict()["fc3.weight"].shape == (1, 256)


def test_critic_state_dict_keys_and_concat_layer():
    c = critic(3, 1, DIST)
    sd = c.state_dict()
    assert sorted(sd.keys()) == sorted([
        "fc1.weight", "fc1.bias", "fc2.weight", "fc2.bias",
        "fc2_2.weight", "fc2_2.bias", "fc3.weight", "fc3.bias"])
    # action injected at layer 2: fc2 input is hidden+act_dim (ref models.py:57)
    assert sd["fc2.weight"].shape == (256, 257)
    assert sd["fc3.weight"].shape == (51, 256)


def test_actor_output_bounded():
    a = actor(3, 2)
    x = torch.randn(16, 3) * 10
    y = a(x)
    assert y.shape == (16, 2)
    assert torch.all(y > -1) and torch.all(y < 1)


def test_critic_outputs_probabilities():
    c = critic(3, 1, DIST)
    q = c(torch.randn(8, 3), torch.randn(8, 1))
    assert q.shape == (8, 51)
    assert torch.all(q >= 0)
    assert torch.allclose(q.sum(dim=1), torch.ones(8), atol=1e-5)


def test_critic_log_forward_matches():
    c = critic(3, 1, DIST)
    s, a = torch.randn(4, 3), torch.randn(4, 1)
    assert torch.allclose(c(s, a, log=True), torch.log(c(s, a)), atol=1e-5)


def test_actor_no_activation_between_fc2_fc2_2():
    """The contract quirk (ref models.py:36-37): fc2 output feeds fc2_2
    WITHOUT a ReLU, so forcing fc2's output negative must still propagate
    information (a ReLU in between would zero it)."""
    a = actor(3, 1)
    with torch.no_grad():
        a.fc2.weight.zero_()
        a.fc2.bias.fill_(-5.0)   # fc2 output = -5 everywhere
        a.fc2_2.weight.fill_(-0.1)  # makes fc2_2 input sign matter
    x = torch.randn(4, 3)
    y1 = a(x)
    with torch.no_grad():
        a.fc2.bias.fill_(-10.0)
    y2 = a(x)
    assert not torch.allclose(y1, y2), \
        "fc2 bias change must reach the output (no ReLU in between)"


def test_fanin_init_std():
    t = torch.empty(4000, 256)
    fanin_init(t)
    assert abs(t.std().item() - 1.0 / np.sqrt(256)) < 0.01


def test_mixture_of_gaussian_rejected():
    with pytest.raises(NotImplementedError):
        critic(3, 1, {"type": "mixture_of_gaussian", "n_atoms": 51})


def test_bin_centers():
    z = bin_centers(-300.0, 0.0, 51)
    assert z.shape == (51,)
    assert z[0] == -300.0 and z[-1] == 0.0
    assert torch.allclose(z[1] - z[0], torch.tensor(6.0))


def test_checkpoint_roundtrip(tmp_path):
    a1, a2 = actor(3, 1), actor(3, 1)
    p = tmp_path / "actor.pth"
    torch.save(a1.state_dict(), p)
    a2.load_state_dict(torch.load(p, weights_only=True))
    x = torch.randn(5, 3)
    assert torch.allclose(a1(x), a2(x))


def test_wide_critic():
    """BASELINE config 5 uses a 1024-wide critic."""
    c = critic(17, 6, DIST, hidden=1024)
    assert c.state_dict()["fc2.weight"].shape == (1024, 1030)
    q = c(torch.randn(4, 17), torch.randn(4, 6))
    assert q.shape == (4, 51)
