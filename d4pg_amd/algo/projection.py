"""Categorical (C51) projection of the Bellman-shifted target distribution.

Semantics follow the reference's two implementations
(/root/reference/ddpg.py:122-140 ``reproj_categorical_dist`` and
:142-185 ``reproject2``): shift each atom z_j by the n-step Bellman backup
``Tz_j = clip(r + gamma_n * (1-done) * z_j, v_min, v_max)``, then split its
probability mass onto the two neighboring atoms of the fixed support.
Terminal rows collapse to a delta at clip(r) — expressed here uniformly via
the (1-done) factor, which distributes every atom's mass onto the same bins
(identical result since the source distribution sums to 1).

Resolved deviation (SURVEY.md §7 hard-part 3): ``reproject2`` (the active
path) discounts by gamma, ignoring the n-step horizon (ddpg.py:155), while
``reproj_categorical_dist`` uses gamma**n_steps (ddpg.py:129).  This build
uses **gamma**n_steps** — the correct D4PG backup — everywhere.

Equal-bin handling: when Tz lands exactly on an atom (floor==ceil), the mass
split (u-b, b-l) would be (0,0); the index-adjustment trick of
reproj_categorical_dist (ddpg.py:133-134) shifts one bound so the weights
become (0,1)/(1,0), preserving total mass — this is also what the HIP kernel
(ops/hip/projection part of the fused step) implements per-row in LDS.
"""

from __future__ import annotations

import torch


def categorical_projection(next_dist: torch.Tensor, rewards: torch.Tensor,
                           dones: torch.Tensor, v_min: float, v_max: float,
                           gamma_n: float) -> torch.Tensor:
    """Project ``next_dist`` [B, K] through the Bellman backup.

    Args:
      next_dist: target-critic probabilities, [B, K].
      rewards:   n-step returns, [B] or [B, 1].
      dones:     terminal flags (0/1 float), [B] or [B, 1].
      gamma_n:   gamma ** n_steps.
    Returns the projected distribution m, [B, K], rows summing to 1.
    """
    B, K = next_dist.shape
    dtype, device = next_dist.dtype, next_dist.device
    r = rewards.reshape(B, 1).to(dtype)
    d = dones.reshape(B, 1).to(dtype)
    delta = (v_max - v_min) / (K - 1)
    z = torch.arange(K, device=device, dtype=dtype) * delta + v_min   # [K]

    tz = (r + gamma_n * (1.0 - d) * z).clamp_(v_min, v_max)          # [B, K]
    b = (tz - v_min) / delta
    l = b.floor().long()
    u = b.ceil().long()
    # equal-bin adjustment (see module docstring): exactly one bound moves.
    eq = l == u
    l = torch.where(eq & (u > 0), l - 1, l)
    u = torch.where(eq & (l == u), u + 1, u)   # only fires when l was NOT moved

    w_l = u.to(dtype) - b
    w_u = b - l.to(dtype)
    m = torch.zeros(B, K, dtype=dtype, device=device)
    m.scatter_add_(1, l, next_dist * w_l)
    m.scatter_add_(1, u, next_dist * w_u)
    return m
