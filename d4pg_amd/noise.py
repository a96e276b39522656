"""Exploration noise processes.

Capability parity with /root/reference/random_process.py:4-45:
``GaussianNoise`` (the active process, ddpg.py:75) and
``OrnsteinUhlenbeckProcess``.  Both expose .sample() / .reset().

Deviations (documented per SURVEY.md §7 "quirk-vs-correctness"):
  * the reference's Gaussian epsilon decay is dead code (``self.iter`` is
    never incremented, random_process.py:16-21); here decay is functional but
    defaults off (decay_period=None) so default behavior matches.
  * each process takes an optional numpy Generator so parallel actors get
    decorrelated streams (the reference shared the global numpy RNG across
    fork which correlates workers).

The GPU-side batched philox variant used by vectorized actor ranks lives in
ops/ (K11 in SURVEY.md §2c); this module is the host/CPU path.
"""

from __future__ import annotations

import numpy as np


class GaussianNoise:
    """action += eps * N(mu, sigma), eps fixed at 0.3 by default
    (reference random_process.py:4-21)."""

    def __init__(self, action_dim: int, mu: float = 0.0, sigma: float = 1.0,
                 eps: float = 0.3, decay_period: int | None = None,
                 rng: np.random.Generator | None = None):
        self.action_dim = action_dim
        self.mu = mu
        self.sigma = sigma
        self.eps0 = eps
        self.eps = eps
        self.decay_period = decay_period
        self.iter = 0
        self.rng = rng or np.random.default_rng()

    def sample(self) -> np.ndarray:
        return self.eps * self.rng.normal(self.mu, self.sigma, self.action_dim)

    def reset(self) -> None:
        if self.decay_period:
            self.iter += 1
            frac = min(1.0, self.iter / self.decay_period)
            self.eps = self.eps0 * (1.0 - frac)


class OrnsteinUhlenbeckProcess:
    """dx = theta*(mu - x)*dt + sigma*sqrt(dt)*N(0,1)
    (reference random_process.py:23-45), with epsilon decay applied on
    reset() as in the reference."""

    def __init__(self, action_dim: int, mu: float = 0.0, theta: float = 0.15,
                 sigma: float = 0.2, dt: float = 1e-2, eps: float = 1.0,
                 decay_period: int = 100000,
                 rng: np.random.Generator | None = None):
        self.action_dim = action_dim
        self.mu = mu
        self.theta = theta
        self.sigma = sigma
        self.dt = dt
        self.eps0 = eps
        self.eps = eps
        self.decay_period = decay_period
        self.n_resets = 0
        self.rng = rng or np.random.default_rng()
        self.x = np.ones(action_dim) * mu

    def sample(self) -> np.ndarray:
        dx = (self.theta * (self.mu - self.x) * self.dt
              + self.sigma * np.sqrt(self.dt)
              * self.rng.standard_normal(self.action_dim))
        self.x = self.x + dx
        return self.eps * self.x

    def reset(self) -> None:
        self.x = np.ones(self.action_dim) * self.mu
        self.n_resets += 1
        frac = min(1.0, self.n_resets / self.decay_period)
        self.eps = self.eps0 * (1.0 - frac)
