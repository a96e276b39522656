"""d4pg_amd — MI355X-native Distributed Distributional DDPG (D4PG) framework.

A from-scratch CDNA4/gfx950 re-design of the capabilities of
ajgupta93/d4pg-pytorch: N parallel actors run gym-style
environments and feed prioritized replay while a learner trains a categorical
(C51) distributional critic + deterministic actor with n-step returns and
target-network soft updates.

Architecture (MI355X-first, not a port):
  * central GPU learner owns params, Adam state, target nets and the entire
    prioritized replay in HBM3E (reference instead used HogWild shared-memory
    CPU workers, /root/reference/main.py:188-368);
  * learner hot path runs as hand-written HIP kernels (fused MLP fwd/bwd on
    MFMA, C51 projection, fused Adam + soft-update, on-HBM sum-tree PER);
  * multi-GPU scaling is one process per GPU over torch.distributed
    (RCCL on ROCm, gloo on CPU) with a flat bucketed gradient all-reduce,
    replacing the reference's `param_global._grad = param_local.grad`
    shared-memory aliasing (/root/reference/ddpg.py:104-108).

Public surface mirrors the reference where users touch it:
  * `DDPG` class with the same constructor/methods (algo/d4pg.py),
  * actor/critic modules with identical state_dict keys fc1/fc2/fc2_2/fc3
    (models.py — the `.pth` checkpoint compatibility contract),
  * the 19 CLI flags of /root/reference/main.py:31-56 (config.py).
"""

__version__ = "0.1.0"

from .models import actor, critic, fanin_init  # noqa: F401
from .algo.d4pg import DDPG  # noqa: F401
from .config import make_parser, configure_env_params  # noqa: F401
