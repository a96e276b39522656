"""n-step return folding (K15 in SURVEY.md §2c).

Semantics of the reference's sliding n-step window
(/root/reference/replay_memory.py:38-45 and main.py:224-234): once ``n``
transitions are buffered, each env step emits one stored tuple
``(s_{t-n+1}, a_{t-n+1}, sum_{k=0..n-1} gamma^k r_{t-n+1+k}, s_{t+1}, done)``.
On episode end the window resets WITHOUT flushing partial tails (reference
behavior — the last emitted tuple carries done=True).

The cumulative reward is maintained incrementally in O(1) per step
(subtract the expiring head, divide by gamma, add the new tail) instead of
the reference's O(n) re-summation; an exact re-sum runs every ``resync``
steps to stop float drift.
"""

from __future__ import annotations

from collections import deque


class NStepFolder:
    def __init__(self, n_steps: int, gamma: float, resync: int = 256):
        self.n = max(1, int(n_steps))
        self.gamma = float(gamma)
        self.resync = resync
        self._count = 0
        self.reset()

    def reset(self) -> None:
        self.states = deque(maxlen=self.n)
        self.actions = deque(maxlen=self.n)
        self.rewards = deque(maxlen=self.n)
        self._cum = 0.0       # sum_{k} gamma^k r_k over the current window
        self._gpow = self.gamma ** (self.n - 1)

    def push(self, state, action, reward, next_state, done):
        """Feed one transition; yield 0 or 1 matured n-step tuples."""
        full = len(self.rewards) == self.n
        if full:
            head = self.rewards[0]
            self._cum = (self._cum - head) / self.gamma
        self.states.append(state)
        self.actions.append(action)
        self.rewards.append(reward)
        self._cum += (self._gpow if len(self.rewards) == self.n else
                      self.gamma ** (len(self.rewards) - 1)) * reward

        out = []
        if len(self.rewards) == self.n:
            self._count += 1
            if self._count % self.resync == 0:
                self._cum = sum(self.gamma ** k * r
                                for k, r in enumerate(self.rewards))
            out.append((self.states[0], self.actions[0], self._cum,
                        next_state, done))
        return out
