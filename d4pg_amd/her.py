"""Hindsight Experience Replay (HER) relabeling + episode collection.

Capability parity with /root/reference/main.py:137-185
(``addExperienceToBuffer``): roll one episode with exploration noise, then
write transitions with goal-concatenated observations; with HER enabled,
each timestep additionally (with probability ``her_ratio``) stores a
hindsight copy relabeled to a future achieved goal, reward recomputed via
``env.compute_reward``, done set when the recomputed reward is 0.

Fixed deviations (SURVEY.md §7 quirk list):
  * the reference's hindsight transition stores the loop-final ``action``
    variable instead of the timestep's action (main.py:184, known bug) —
    fixed here: the relabeled tuple carries ``episode[t]``'s action;
  * the reference only writes to the buffer at all when HER is on
    (``if args.her and not done`` guards the entire add-loop, main.py:154,
    so ``--her 0`` never trains) — here non-HER episodes are stored too,
    which is what the reference's commented-out code intended.

n-step note: the reference applies n-step folding only in the separate
Worker warmup path (main.py:224-234), not in HER adds; here both paths fold
through the same NStepFolder for consistency.
"""

from __future__ import annotations

import numpy as np

from .replay.nstep import NStepFolder


def flat_obs(o):
    """Goal-concat for dict observations (reference main.py:144 semantics),
    identity for flat ones."""
    if isinstance(o, dict):
        return np.concatenate([np.asarray(o["observation"], np.float32).ravel(),
                               np.asarray(o["desired_goal"], np.float32).ravel()])
    return np.asarray(o, np.float32).ravel()


def rollout_episode(policy, env, noise=True, max_steps=None):
    """Collect one episode.  Returns (episode, total_reward, success) where
    episode is a list of (obs, action, reward, next_obs, done, info) with
    raw (possibly dict) observations."""
    episode = []
    obs = env.reset()
    total_r = 0.0
    success = False
    steps = max_steps or env._max_episode_steps
    for _ in range(steps):
        a = policy.select_action(flat_obs(obs), explore=noise)
        next_obs, r, done, info = env.step(a)
        if isinstance(info, dict) and "is_success" in info:
            done = bool(info["is_success"]) or done
            success = success or bool(info["is_success"])
        episode.append((obs, a, r, next_obs, done, info))
        total_r += r
        obs = next_obs
        if done:
            break
    return episode, total_r, success


def add_experience(replay_buffer, env, episode, her: bool = False,
                   her_ratio: float = 0.8, n_steps: int = 1,
                   gamma: float = 0.99,
                   rng: np.random.Generator | None = None) -> int:
    """Store an episode's transitions (+ optional HER relabels).
    Returns the number of tuples written."""
    rng = rng or np.random.default_rng()
    written = 0

    folder = NStepFolder(n_steps, gamma)
    for (obs, a, r, next_obs, done, info) in episode:
        for tr in folder.push(flat_obs(obs), a, r, flat_obs(next_obs), done):
            replay_buffer.add(*tr)
            written += 1

    if her and episode and isinstance(episode[0][0], dict):
        T = len(episode)
        her_folder = NStepFolder(1, gamma)  # relabeled tuples are 1-step
        for t, (obs, a, r, next_obs, done, info) in enumerate(episode):
            if rng.random() >= her_ratio:
                continue
            # "future" strategy: substitute an achieved goal from t..T-1
            fut = int(rng.integers(t, T))
            new_goal = np.asarray(episode[fut][3]["achieved_goal"])
            achieved = np.asarray(next_obs["achieved_goal"])
            new_r = float(np.asarray(env.compute_reward(achieved, new_goal)))
            new_done = new_r == 0.0
            s = np.concatenate([np.asarray(obs["observation"],
                                           np.float32).ravel(),
                                new_goal.astype(np.float32).ravel()])
            s2 = np.concatenate([np.asarray(next_obs["observation"],
                                            np.float32).ravel(),
                                 new_goal.astype(np.float32).ravel()])
            replay_buffer.add(s, a, new_r, s2, new_done)   # a = this step's
            written += 1
        del her_folder
    return written
