"""Prioritized experience replay (PER).

Capability parity with /root/reference/prioritized_replay_memory.py:33-335
(OpenAI-baselines lineage: proportional sampling p_i^alpha via a sum segment
tree, IS weights (N*p)^-beta normalized by the max weight via a min tree,
priorities written back as |td|+eps, new transitions at max_priority^alpha).

New design, not a translation: the trees are flat numpy arrays updated in
*batched level-order* passes (one vectorized np.add.reduceat-style sweep per
tree level instead of per-element Python recursion) and sampling is a
*batched prefix descent* — all B probes walk the tree together, one
vectorized level per iteration.  This is the exact algorithm the on-HBM HIP
tree (k_per_sample / k_per_update / k_per_leaves in ops/hip/engine.hip, K12
in SURVEY.md §2c) implements with one wave per probe, so the CPU path
doubles as its parity oracle.

Deviation (documented): the reference samples mass in
``random() * sum(0, len-1)`` whose exclusive ``end`` drops the newest element
(baselines off-by-one); this build samples over the full occupied range.
"""

from __future__ import annotations

import numpy as np

from .uniform import SoAStore


def _next_pow2(n: int) -> int:
    c = 1
    while c < n:
        c *= 2
    return c


class _SegmentTree:
    """Flat array-backed binary tree over ``capacity`` (pow-2) leaves.

    Layout: node 1 is the root; node i's children are 2i, 2i+1; leaves live
    at [capacity, 2*capacity).  Same invariants as the reference
    (prioritized_replay_memory.py:34-112) with vectorized batch updates.
    """

    def __init__(self, capacity: int, op, neutral: float):
        assert capacity > 0 and (capacity & (capacity - 1)) == 0, \
            "capacity must be positive and a power of 2"
        self.capacity = capacity
        self.op = op
        self.neutral = neutral
        self.tree = np.full(2 * capacity, neutral, dtype=np.float64)

    # -- batched core --
    def set_batch(self, idx: np.ndarray, val: np.ndarray) -> None:
        """Write leaves then repair ancestors level-by-level (vectorized)."""
        idx = np.asarray(idx, dtype=np.int64) + self.capacity
        self.tree[idx] = val
        parents = np.unique(idx >> 1)
        while parents.size and parents[0] >= 1:
            self.tree[parents] = self.op(self.tree[2 * parents],
                                         self.tree[2 * parents + 1])
            parents = np.unique(parents >> 1)
            if parents[0] == 0:
                break

    # -- reference-compatible element access --
    def __setitem__(self, idx, val) -> None:
        self.set_batch(np.atleast_1d(idx), np.atleast_1d(val))

    def __getitem__(self, idx):
        out = self.tree[np.asarray(idx, dtype=np.int64) + self.capacity]
        return out if isinstance(idx, (np.ndarray, list)) else float(out)

    def reduce(self, start: int = 0, end: int | None = None) -> float:
        """Reduce over [start, end) (reference semantics: end exclusive,
        None = whole range, negative end wraps)."""
        if end is None:
            end = self.capacity
        if end < 0:
            end += self.capacity
        res = self.neutral
        lo, hi = start + self.capacity, end - 1 + self.capacity
        while lo <= hi:
            if lo & 1:
                res = self.op(res, self.tree[lo])
                lo += 1
            if not hi & 1:
                res = self.op(res, self.tree[hi])
                hi -= 1
            lo >>= 1
            hi >>= 1
        return float(res)


class SumSegmentTree(_SegmentTree):
    def __init__(self, capacity: int):
        super().__init__(capacity, np.add, 0.0)

    def sum(self, start: int = 0, end: int | None = None) -> float:
        return self.reduce(start, end)

    def find_prefixsum_idx(self, prefixsum):
        """Batched root-to-leaf prefix-sum descent.

        Scalar in, scalar out (reference
        prioritized_replay_memory.py:126-149 parity); array in, array out
        (the vectorized form the GPU kernel mirrors).
        """
        scalar = np.isscalar(prefixsum)
        mass = np.atleast_1d(np.asarray(prefixsum, dtype=np.float64)).copy()
        node = np.ones(mass.shape, dtype=np.int64)
        while node[0] < self.capacity:          # all probes at equal depth
            left = 2 * node
            lsum = self.tree[left]
            go_right = mass > lsum
            mass = np.where(go_right, mass - lsum, mass)
            node = np.where(go_right, left + 1, left)
        leaf = node - self.capacity
        return int(leaf[0]) if scalar else leaf


class MinSegmentTree(_SegmentTree):
    def __init__(self, capacity: int):
        super().__init__(capacity, np.minimum, float("inf"))

    def min(self, start: int = 0, end: int | None = None) -> float:
        return self.reduce(start, end)


class ReplayBuffer:
    """Uniform buffer with the baselines-style API
    (reference prioritized_replay_memory.py:165-222), SoA-backed."""

    def __init__(self, size: int, rng: np.random.Generator | None = None):
        self._store = SoAStore(size)
        self._maxsize = size
        self.rng = rng or np.random.default_rng()

    def __len__(self) -> int:
        return len(self._store)

    @property
    def _next_idx(self) -> int:
        return self._store.pos

    def add(self, obs_t, action, reward, obs_tp1, done) -> int:
        return self._store.add(obs_t, action, reward, obs_tp1, done)

    def _encode_sample(self, idxes):
        s, a, r, s2, d = self._store.gather(idxes)
        return s, a, r.reshape(-1), s2, d.reshape(-1)

    def sample(self, batch_size: int):
        idxes = self.rng.integers(0, len(self), size=batch_size)
        return self._encode_sample(idxes)


class PrioritizedReplayBuffer(ReplayBuffer):
    def __init__(self, size: int, alpha: float,
                 rng: np.random.Generator | None = None):
        super().__init__(size, rng=rng)
        assert alpha >= 0
        self._alpha = alpha
        cap = _next_pow2(size)
        self._it_sum = SumSegmentTree(cap)
        self._it_min = MinSegmentTree(cap)
        self._max_priority = 1.0

    def add(self, *args, **kwargs) -> int:
        idx = self._next_idx
        super().add(*args, **kwargs)
        p = self._max_priority ** self._alpha
        self._it_sum.set_batch(np.array([idx]), np.array([p]))
        self._it_min.set_batch(np.array([idx]), np.array([p]))
        return idx

    def _sample_proportional(self, batch_size: int) -> np.ndarray:
        total = self._it_sum.sum(0, len(self))
        mass = self.rng.random(batch_size) * total
        idx = self._it_sum.find_prefixsum_idx(mass)
        # guard: an unoccupied leaf can only be hit by fp round-off at the
        # very top of the range; clamp into the occupied region.
        return np.minimum(idx, len(self) - 1)

    def sample(self, batch_size: int, beta: float):
        assert beta > 0
        n = len(self)
        idxes = self._sample_proportional(batch_size)
        total = self._it_sum.sum()
        p_min = self._it_min.min() / total
        max_weight = (p_min * n) ** (-beta)
        p_sample = self._it_sum[idxes] / total
        weights = ((p_sample * n) ** (-beta) / max_weight).astype(np.float32)
        s, a, r, s2, d = self._encode_sample(idxes)
        return s, a, r, s2, d, weights, idxes

    def update_priorities(self, idxes, priorities) -> None:
        idxes = np.asarray(idxes, dtype=np.int64)
        priorities = np.asarray(priorities, dtype=np.float64)
        assert idxes.shape == priorities.shape
        assert np.all(priorities > 0)
        assert np.all((0 <= idxes) & (idxes < len(self)))
        # reference semantics: last write wins on duplicate indices;
        # np fancy-assign in set_batch already does that.
        p = priorities ** self._alpha
        self._it_sum.set_batch(idxes, p)
        self._it_min.set_batch(idxes, p)
        self._max_priority = max(self._max_priority, float(priorities.max()))

    def state_dict(self) -> dict:
        st = self._store
        return {
            "size": st.size, "pos": st.pos, "max_priority": self._max_priority,
            "alpha": self._alpha,
            "leaves": self._it_sum.tree[self._it_sum.capacity:
                                        self._it_sum.capacity + st.size].copy(),
            "arrays": None if not st._alloc_done else {
                "states": st.states[:st.size].copy(),
                "actions": st.actions[:st.size].copy(),
                "rewards": st.rewards[:st.size].copy(),
                "next_states": st.next_states[:st.size].copy(),
                "dones": st.dones[:st.size].copy(),
            },
        }

    def load_state_dict(self, state: dict) -> None:
        self._max_priority = state["max_priority"]
        arrays = state["arrays"]
        if arrays is None:
            return
        n = state["size"]
        st = self._store
        st._alloc(arrays["states"][0], arrays["actions"][0])
        st.states[:n] = arrays["states"]
        st.actions[:n] = arrays["actions"]
        st.rewards[:n] = arrays["rewards"]
        st.next_states[:n] = arrays["next_states"]
        st.dones[:n] = arrays["dones"]
        st.size, st.pos = n, state["pos"]
        idx = np.arange(n)
        self._it_sum.set_batch(idx, state["leaves"])
        self._it_min.set_batch(idx, state["leaves"])
