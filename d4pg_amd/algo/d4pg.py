"""D4PG algorithm core (the learner).

Capability/API parity with the reference's ``DDPG`` class
(/root/reference/ddpg.py:15-255): same constructor signature, same public
methods (train, hard_update, share_memory, assign_global_optimizer,
copy_gradients, update_target_parameters, sync_local_global, sample,
reproj_categorical_dist), same hyperparameter defaults (PER alpha=0.6,
beta 0.4->1.0 over 1e5 stateful calls, eps=1e-6), and the same training
semantics per step:

  1. sample a (possibly prioritized) minibatch,
  2. target critic distribution at (s', actor_target(s')),
  3. C51 projection of the Bellman-shifted target (algo/projection.py),
  4. critic CE loss -sum(m * log(q + 1e-10)) averaged over the batch,
  5. actor loss -E[Q] = -(q_dist @ bin_centers).mean(),
  6. Adam steps, target soft updates (tau-lerp),
  7. PER priority write-back |td|+eps where td = -sum(m*q) (the reference's
     proxy, ddpg.py:220-222 — kept for parity, see ``true_td_priorities``).

Kept quirks: IS weights returned by PER.sample are NOT applied
to the loss (reference ddpg.py:217 ignores them) unless ``is_weighting=True``;
priorities use the -sum(m*q) proxy unless ``true_td_priorities=True``.
Fixed deviations: the projection discounts by gamma**n_steps (not the
reference reproject2's gamma, ddpg.py:155).

Backends:
  * ``eager``  — plain torch ops, runs on CPU or GPU; the test oracle.
  * ``hip``    — the hand-written CDNA4 fused-step path (ops/), GPU only:
    the whole train step (forwards, projection, backwards, Adam, soft
    update, priority refresh) runs as fused HIP kernels over a flat
    parameter slab, replayed without per-op dispatch overhead.
"""

from __future__ import annotations

import numpy as np
import torch
import torch.nn as nn

from ..models import actor as Actor, critic as Critic
from ..noise import GaussianNoise
from ..replay.uniform import Replay
from ..replay.per import PrioritizedReplayBuffer
from ..replay.schedules import LinearSchedule
from .projection import categorical_projection


class DDPG:
    def __init__(self, obs_dim, act_dim, env=None, memory_size=50000,
                 batch_size=64, lr_critic=1e-4, lr_actor=1e-4, gamma=0.99,
                 tau=0.001, prioritized_replay=True, critic_dist_info=None,
                 n_steps=1, *, device="cpu", backend="eager", hidden=256,
                 is_weighting=False, true_td_priorities=False, seed=None,
                 noise="gaussian", noise_eps=0.3, ou_theta=0.15,
                 ou_sigma=0.2, ou_mu=0.0):
        self.gamma = gamma
        self.n_steps = n_steps
        self.n_step_gamma = gamma ** n_steps
        self.batch_size = batch_size
        self.obs_dim = obs_dim
        self.act_dim = act_dim
        self.memory_size = memory_size
        self.tau = tau
        self.env = env
        self.device = torch.device(device)
        self.backend = backend
        self.is_weighting = is_weighting
        self.true_td_priorities = true_td_priorities

        if critic_dist_info is None:
            critic_dist_info = {"type": "categorical", "v_min": -100.0,
                                "v_max": 100.0, "n_atoms": 51}
        self.dist_type = critic_dist_info["type"]
        if self.dist_type != "categorical":
            raise NotImplementedError(
                "unsupported critic distribution type %r" % self.dist_type)
        self.v_min = float(critic_dist_info["v_min"])
        self.v_max = float(critic_dist_info["v_max"])
        self.n_atoms = int(critic_dist_info["n_atoms"])
        self.delta = (self.v_max - self.v_min) / float(self.n_atoms - 1)
        # column vector like the reference (ddpg.py:47) for matmul parity
        self.bin_centers = np.arange(self.n_atoms, dtype=np.float64) \
            .reshape(-1, 1) * self.delta + self.v_min
        self._z = torch.linspace(self.v_min, self.v_max, self.n_atoms,
                                 device=self.device)

        rng = np.random.default_rng(seed)
        self.rng = rng
        if seed is not None:
            torch.manual_seed(seed)

        self.actor = Actor(obs_dim, act_dim, hidden=hidden).to(self.device)
        self.actor_target = Actor(obs_dim, act_dim, hidden=hidden).to(self.device)
        self.actor_target.load_state_dict(self.actor.state_dict())
        self.critic = Critic(obs_dim, act_dim, critic_dist_info,
                             hidden=hidden).to(self.device)
        self.critic_target = Critic(obs_dim, act_dim, critic_dist_info,
                                    hidden=hidden).to(self.device)
        self.critic_target.load_state_dict(self.critic.state_dict())

        self.optimizer_actor = torch.optim.Adam(self.actor.parameters(),
                                                lr=lr_actor)
        self.optimizer_critic = torch.optim.Adam(self.critic.parameters(),
                                                 lr=lr_critic)
        self.optimizer_global_actor = None
        self.optimizer_global_critic = None

        # --noise selector: the reference declares OU flags but only ever
        # constructs GaussianNoise (ddpg.py:74-75); here 'ou' actually
        # wires --ou_theta/sigma/mu through
        if noise == "ou":
            from ..noise import OrnsteinUhlenbeckProcess
            self.noise = OrnsteinUhlenbeckProcess(
                act_dim, mu=ou_mu, theta=ou_theta, sigma=ou_sigma, rng=rng)
        else:
            self.noise = GaussianNoise(act_dim, eps=noise_eps, rng=rng)

        self.prioritized_replay = prioritized_replay
        if prioritized_replay:
            self.replayBuffer = PrioritizedReplayBuffer(memory_size,
                                                        alpha=0.6, rng=rng)
            self.beta_schedule = LinearSchedule(100000, initial_p=0.4,
                                                final_p=1.0)
            self.prioritized_replay_eps = 1e-6
        else:
            self.replayBuffer = Replay(memory_size, env, n_steps=n_steps,
                                       gamma=gamma, rng=rng)

        self._fused = None   # lazily-built HIP fused-step engine bridge
        self.train_steps_done = 0
        # learner-DP hook (parallel/dp.py): called with the module whose
        # .grad tensors are about to be consumed by the optimizer, right
        # after backward — the DP wrapper all-reduce-averages them there,
        # preserving the reference's update order (critic Adam BEFORE the
        # policy forward, ddpg.py:232/236).
        self.grad_sync = None

    @property
    def engine(self):
        """The fused HIP engine bridge (backend='hip' after the first
        train step), else None."""
        return self._fused

    # ------------------------------------------------------------------
    # parameter plumbing (reference ddpg.py:92-120 parity)
    # ------------------------------------------------------------------
    def hard_update(self) -> None:
        self.actor_target.load_state_dict(self.actor.state_dict())
        self.critic_target.load_state_dict(self.critic.state_dict())

    def share_memory(self) -> None:
        self.actor.share_memory()
        self.critic.share_memory()

    def assign_global_optimizer(self, optimizer_global_actor,
                                optimizer_global_critic) -> None:
        self.optimizer_global_actor = optimizer_global_actor
        self.optimizer_global_critic = optimizer_global_critic

    def copy_gradients(self, model_local: nn.Module,
                       model_global: nn.Module) -> None:
        """One-shot grad aliasing local->global (reference ddpg.py:104-108
        semantics: after the first call, local backward writes land directly
        in the global grads)."""
        for p_local, p_global in zip(model_local.parameters(),
                                     model_global.parameters()):
            if p_global.grad is not None:
                return
            p_global._grad = p_local.grad

    def update_target_parameters(self) -> None:
        with torch.no_grad():
            for tp, sp in zip(self.actor_target.parameters(),
                              self.actor.parameters()):
                tp.lerp_(sp, self.tau)
            for tp, sp in zip(self.critic_target.parameters(),
                              self.critic.parameters()):
                tp.lerp_(sp, self.tau)

    def sync_local_global(self, global_model: "DDPG") -> None:
        self.actor.load_state_dict(global_model.actor.state_dict())
        self.critic.load_state_dict(global_model.critic.state_dict())

    # ------------------------------------------------------------------
    # acting
    # ------------------------------------------------------------------
    @torch.no_grad()
    def select_action(self, obs: np.ndarray, explore: bool = True) -> np.ndarray:
        """B=1 policy inference for env stepping (+clipped Gaussian noise,
        reference main.py:145-146 semantics)."""
        if self._fused is not None:
            self._fused.sync_params_if_dirty()
        x = torch.as_tensor(np.asarray(obs, dtype=np.float32),
                            device=self.device).reshape(1, -1)
        a = self.actor(x).cpu().numpy().reshape(-1)
        if explore:
            a = a + self.noise.sample()
        return np.clip(a, -1.0, 1.0)

    # ------------------------------------------------------------------
    # projection (numpy API parity; torch path uses algo/projection.py)
    # ------------------------------------------------------------------
    def reproj_categorical_dist(self, target_z_dist, rewards, terminates):
        """Numpy-facing projection with reference signature
        (ddpg.py:122-140); discounts by gamma**n_steps."""
        m = categorical_projection(
            torch.as_tensor(np.asarray(target_z_dist, dtype=np.float32)),
            torch.as_tensor(np.asarray(rewards, dtype=np.float32)),
            torch.as_tensor(np.asarray(terminates, dtype=np.float32)),
            self.v_min, self.v_max, self.n_step_gamma)
        return m.numpy()

    # kept name from the reference's active path (ddpg.py:142); same
    # vectorized implementation here (the gamma-vs-gamma**n discrepancy is
    # resolved to gamma**n, see algo/projection.py docstring).
    reproject2 = reproj_categorical_dist

    # ------------------------------------------------------------------
    # sampling
    # ------------------------------------------------------------------
    def sample(self, batch_size=None):
        batch_size = batch_size or self.batch_size
        if self.prioritized_replay:
            beta = self.beta_schedule.value()
            return self.replayBuffer.sample(batch_size, beta=beta)
        s, a, r, s2, d = self.replayBuffer.sample(batch_size)
        return s, a, r, s2, d, None, None

    # ------------------------------------------------------------------
    # training
    # ------------------------------------------------------------------
    def train(self, global_model: "DDPG | None" = None):
        """One gradient step.  With ``global_model`` (HogWild-parity mode)
        gradients flow into the shared global parameters and the shared
        optimizers step (reference ddpg.py:200-255); without it, the local
        optimizers step (the MI355X learner topology, where distribution is
        handled by RCCL all-reduce in parallel/learner.py instead)."""
        if self.backend == "hip":
            return self._train_step_hip()
        batch = self.sample(self.batch_size)
        return self._train_step_eager(batch, global_model)

    def _train_step_eager(self, batch, global_model=None):
        states, actions, rewards, next_states, terminates, weights, idxes = batch
        dev = self.device
        s = torch.as_tensor(np.asarray(states, dtype=np.float32), device=dev)
        a = torch.as_tensor(np.asarray(actions, dtype=np.float32), device=dev)
        r = torch.as_tensor(np.asarray(rewards, dtype=np.float32), device=dev)
        s2 = torch.as_tensor(np.asarray(next_states, dtype=np.float32), device=dev)
        d = torch.as_tensor(np.asarray(terminates, dtype=np.float32), device=dev)

        # -- critic update --
        with torch.no_grad():
            a2 = self.actor_target(s2)
            target_dist = self.critic_target(s2, a2)
            m = categorical_projection(target_dist, r, d, self.v_min,
                                       self.v_max, self.n_step_gamma)
        q_dist = self.critic(s, a)
        per_sample_ce = -(m * torch.log(q_dist + 1e-10)).sum(dim=1)
        if self.is_weighting and weights is not None:
            w = torch.as_tensor(np.asarray(weights, dtype=np.float32),
                                device=dev)
            qdist_loss = (w * per_sample_ce).mean()
        else:
            qdist_loss = per_sample_ce.mean()

        td_errors = None
        if self.prioritized_replay:
            if self.true_td_priorities:
                exp_q = (q_dist.detach() * self._z).sum(dim=1)
                exp_m = (m * self._z).sum(dim=1)
                td_errors = exp_m - exp_q
            else:
                # reference proxy (ddpg.py:220-222): -sum(m * q)
                td_errors = -(m * q_dist.detach()).sum(dim=1)

        self.critic.zero_grad()
        if global_model is not None:
            global_model.critic.zero_grad()
        qdist_loss.backward()
        if global_model is not None:
            self.copy_gradients(self.critic, global_model.critic)
            self.optimizer_global_critic.step()
        else:
            if self.grad_sync is not None:
                self.grad_sync(self.critic)
            self.optimizer_critic.step()

        # -- actor update --
        z = self._z.reshape(-1, 1)
        policy_q = self.critic(s, self.actor(s))
        policy_loss = -(policy_q.matmul(z)).mean()
        self.actor.zero_grad()
        if global_model is not None:
            global_model.actor.zero_grad()
        policy_loss.backward()
        if global_model is not None:
            self.copy_gradients(self.actor, global_model.actor)
            self.optimizer_global_actor.step()
            self.sync_local_global(global_model)
        else:
            if self.grad_sync is not None:
                self.grad_sync(self.actor)
            self.optimizer_actor.step()

        self.update_target_parameters()

        if self.prioritized_replay and idxes is not None:
            new_p = np.abs(td_errors.cpu().numpy()) + self.prioritized_replay_eps
            self.replayBuffer.update_priorities(idxes, new_p)

        self.train_steps_done += 1
        return float(qdist_loss.detach()), float(policy_loss.detach())

    def _train_step_hip(self):
        if self._fused is None:
            from ..ops import build_fused_engine
            self._fused = build_fused_engine(self)
        return self._fused.step()

    # ------------------------------------------------------------------
    # checkpointing
    # ------------------------------------------------------------------
    def save(self, run_dir: str) -> None:
        """Reference-format checkpoint: actor.pth / critic.pth state_dicts
        with fc1/fc2/fc2_2/fc3 keys (main.py:367-368)."""
        import os
        if self._fused is not None:
            self._fused.sync_params_if_dirty()
        torch.save(self.actor.state_dict(),
                   os.path.join(run_dir, "actor.pth"))
        torch.save(self.critic.state_dict(),
                   os.path.join(run_dir, "critic.pth"))

    def state_dict(self) -> dict:
        """Full-resume checkpoint (capability the reference lacks —
        SURVEY.md §5 checkpoint row)."""
        if self._fused is not None:
            self._fused.sync_params_if_dirty()
        st = {
            "actor": self.actor.state_dict(),
            "critic": self.critic.state_dict(),
            "actor_target": self.actor_target.state_dict(),
            "critic_target": self.critic_target.state_dict(),
            "optimizer_actor": self.optimizer_actor.state_dict(),
            "optimizer_critic": self.optimizer_critic.state_dict(),
            "train_steps_done": self.train_steps_done,
            "torch_rng": torch.get_rng_state(),
        }
        if self.prioritized_replay and hasattr(self.replayBuffer,
                                               "state_dict"):
            st["beta_schedule"] = self.beta_schedule.state_dict()
            st["replay"] = self.replayBuffer.state_dict()
        if self._fused is not None:
            # device-side optimizer state (Adam moments live in engine
            # slabs, not in the torch optimizers) + schedule counters
            eng = self._fused.engine
            st["engine"] = {
                "m_actor": eng.store_slab("m_actor"),
                "v_actor": eng.store_slab("v_actor"),
                "m_critic": eng.store_slab("m_critic"),
                "v_critic": eng.store_slab("v_critic"),
                "counters": eng.counters(),
                "seed": eng.info()["seed"],
            }
        # nn.Module.state_dict() returns REFERENCES to the live tensors —
        # continued training would silently mutate the checkpoint.  A
        # checkpoint must be a snapshot: deep-copy everything.
        import copy as _copy
        return _copy.deepcopy(st)

    def load_state_dict(self, st: dict, load_replay: bool = True) -> None:
        # A fresh backend='hip' agent restoring a GPU-format checkpoint must
        # build the fused bridge FIRST: the on-HBM replay blob and engine
        # slabs in `st` have nowhere to land otherwise, and the CPU replay
        # buffer cannot parse the GPU-format replay dict.
        if self.backend == "hip" and "engine" in st and self._fused is None:
            from ..ops import build_fused_engine
            self._fused = build_fused_engine(self)
        if "replay" in st and load_replay:
            gpu_format = "sum_tree" in st["replay"]
            have_gpu_buf = hasattr(self.replayBuffer, "engine")
            if gpu_format != have_gpu_buf:
                raise ValueError(
                    "checkpoint replay format (%s) does not match this "
                    "agent's replay buffer (%s) — load with backend=%r or "
                    "pass load_replay=False"
                    % ("on-HBM" if gpu_format else "CPU",
                       "on-HBM" if have_gpu_buf else "CPU",
                       "hip" if gpu_format else "eager"))
        self.actor.load_state_dict(st["actor"])
        self.critic.load_state_dict(st["critic"])
        self.actor_target.load_state_dict(st["actor_target"])
        self.critic_target.load_state_dict(st["critic_target"])
        self.optimizer_actor.load_state_dict(st["optimizer_actor"])
        self.optimizer_critic.load_state_dict(st["optimizer_critic"])
        self.train_steps_done = st["train_steps_done"]
        torch.set_rng_state(st["torch_rng"])
        if self.prioritized_replay and "beta_schedule" in st:
            self.beta_schedule.load_state_dict(st["beta_schedule"])
            if (load_replay and "replay" in st
                    and hasattr(self.replayBuffer, "load_state_dict")):
                self.replayBuffer.load_state_dict(st["replay"])
        if self._fused is not None and "engine" in st:
            eng = self._fused.engine
            eng.load_from_modules(self.actor, self.actor_target,
                                  self.critic, self.critic_target)
            for k in ("m_actor", "v_actor", "m_critic", "v_critic"):
                eng.load_slab(k, st["engine"][k])
            cnt = st["engine"]["counters"]
            if "seed" in st["engine"]:
                eng.set_seed(int(st["engine"]["seed"]))
            eng.set_schedule(cnt["adam_t_actor"], cnt["max_priority"])
            self._fused._params_dirty = False
