"""Tensor glue helpers (API parity with /root/reference/utils.py:4-9,
modernized past the pre-0.4 Variable/volatile autograd API)."""

from __future__ import annotations

import numpy as np
import torch


def to_numpy(x: torch.Tensor) -> np.ndarray:
    return x.detach().cpu().numpy()


def to_tensor(x, volatile: bool = False, requires_grad: bool = False,
              device=None, dtype=torch.float32) -> torch.Tensor:
    """numpy -> FloatTensor.  ``volatile`` kept for signature parity; it maps
    to requires_grad=False (the modern equivalent is wrapping the call site
    in torch.no_grad(), which the algo core does)."""
    t = torch.as_tensor(np.asarray(x), dtype=dtype, device=device)
    if requires_grad and not volatile:
        t = t.clone().requires_grad_(True)
    return t
