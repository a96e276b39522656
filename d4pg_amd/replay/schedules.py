"""Annealing schedules.

``LinearSchedule`` keeps the reference's *stateful* semantics
(/root/reference/prioritized_replay_memory.py:25-29 + ddpg.py:192): every
``value()`` call advances the internal step counter, so the PER beta anneal
progresses once per learner sample call.  This statefulness is part of the
behavioral contract (SURVEY.md §7 quirk list: "β-schedule advancing per call
(keep)") — a ``value_at(t)`` pure accessor is provided for code that wants
the stateless form.
"""

from __future__ import annotations


class LinearSchedule:
    def __init__(self, schedule_timesteps: int, final_p: float,
                 initial_p: float = 1.0):
        self.schedule_timesteps = schedule_timesteps
        self.initial_p = initial_p
        self.final_p = final_p
        self.t = 0

    def value_at(self, t: int) -> float:
        frac = min(float(t) / self.schedule_timesteps, 1.0)
        return self.initial_p + frac * (self.final_p - self.initial_p)

    def value(self) -> float:
        v = self.value_at(self.t)
        self.t += 1
        return v

    def state_dict(self) -> dict:
        return {"t": self.t}

    def load_state_dict(self, state: dict) -> None:
        self.t = int(state["t"])
