"""Elastic actor-learner tests (parallel/elastic.py): kill an actor
mid-run -> the learner keeps training; start a replacement -> it rejoins
on the current param version (SURVEY §5 failure-recovery row, VERDICT r1
next-steps #6).  Also checks the staleness tags (#8)."""

import os
import time

import numpy as np
import torch.multiprocessing as mp

PORT = 29651


def _mk_agent():
    from d4pg_amd.algo.d4pg import DDPG
    return DDPG(3, 1, memory_size=50000, batch_size=32,
                prioritized_replay=False, n_steps=1,
                critic_dist_info={"type": "categorical", "v_min": -300.0,
                                  "v_max": 0.0, "n_atoms": 51},
                device="cpu", backend="eager", seed=0)


def _collect(agent, n=40, seed=None):
    import torch
    rng = np.random.default_rng(seed)
    s = rng.standard_normal((n, 3)).astype(np.float32)
    with torch.no_grad():
        a = agent.actor(torch.from_numpy(s)).numpy()
    r = -rng.random(n).astype(np.float32)
    s2 = rng.standard_normal((n, 3)).astype(np.float32)
    d = np.zeros(n, np.float32)
    return s, a, r, s2, d


def _actor_main(aid, rounds, delay):
    from d4pg_amd.parallel.elastic import ElasticActor, make_store
    store = make_store(port=PORT, is_master=False)
    agent = _mk_agent()
    actor = ElasticActor(aid, agent, store, 3, 1,
                         lambda ag: _collect(ag, seed=aid * 1000))
    for _ in range(rounds):
        if not actor.round():
            break
        time.sleep(delay)


def test_kill_and_rejoin():
    from d4pg_amd.parallel.elastic import ElasticLearner, make_store
    ctx = mp.get_context("spawn")
    store = make_store(port=PORT, is_master=True)
    agent = _mk_agent()
    learner = ElasticLearner(agent, store, 3, 1, dead_after_s=1.5)

    a0 = ctx.Process(target=_actor_main, args=(0, 400, 0.05))
    a1 = ctx.Process(target=_actor_main, args=(1, 400, 0.05))
    a0.start()
    a1.start()

    # let both actors deliver, training along the way
    deadline = time.time() + 30
    while time.time() < deadline and \
            not (0 in learner.drained and 1 in learner.drained
                 and learner.ingested_total >= 160):
        learner.drain_mail()
        learner.train(2)
        learner.publish_params()
        time.sleep(0.02)
    assert 0 in learner.drained and 1 in learner.drained
    steps_before = agent.train_steps_done
    assert steps_before > 0

    # SIGKILL actor 1 mid-run: the learner must keep going and flag it
    a1.kill()
    a1.join(timeout=10)
    t_dead = None
    deadline = time.time() + 30
    while time.time() < deadline:
        learner.drain_mail()
        learner.train(2)
        learner.publish_params()
        if 1 in learner.dead_actors():
            t_dead = time.time()
            break
        time.sleep(0.02)
    assert t_dead is not None, "dead actor never detected"
    assert agent.train_steps_done > steps_before, \
        "learner stalled after actor death"

    # a NEW actor (fresh id) rejoins on the current params
    ver_at_rejoin = learner.version
    a2 = ctx.Process(target=_actor_main, args=(2, 400, 0.05))
    a2.start()
    seen2 = 0
    deadline = time.time() + 30
    while time.time() < deadline and seen2 == 0:
        learner.drain_mail()
        learner.publish_params()
        seen2 = learner.drained.get(2, 0)
        time.sleep(0.02)
    assert seen2 > 0, "replacement actor's mail never ingested"
    # staleness tags: rejoined actor acted on a CURRENT version (>= the
    # version live when it started), never on pre-death params
    recent = learner.staleness[-seen2:]
    assert max(recent) <= learner.version - ver_at_rejoin + 1

    learner.stop()
    for p in (a0, a2):
        p.join(timeout=20)
    a0.terminate() if a0.is_alive() else None
    a2.terminate() if a2.is_alive() else None
    # overall staleness stays bounded (actors always pull before collect)
    assert np.mean(learner.staleness) < 5


def test_staleness_tag_is_version_pulled():
    """An actor's mail is tagged with exactly the version it pulled."""
    from d4pg_amd.parallel.elastic import (ElasticActor, ElasticLearner,
                                           make_store)
    master = make_store(port=PORT + 2, is_master=True)
    agent = _mk_agent()
    learner = ElasticLearner(agent, master, 3, 1)
    actor = ElasticActor(7, _mk_agent(), master, 3, 1,
                         lambda ag: _collect(ag, n=5, seed=1))
    actor.round()
    learner.drain_mail()
    assert learner.staleness[-1] == 0          # acted on latest
    learner.publish_params()
    learner.publish_params()                    # actor now 2 behind
    # actor collects WITHOUT pulling (simulate slow pull by stubbing)
    actor.pull_params = lambda: False
    actor.round()
    learner.drain_mail()
    assert learner.staleness[-1] == 2


def test_learner_handles_empty_and_out_of_order_actors():
    """Mail from several actors with interleaved publish order; drain
    ingests everything exactly once."""
    from d4pg_amd.parallel.elastic import (ElasticActor, ElasticLearner,
                                           make_store)
    master = make_store(port=PORT + 4, is_master=True)
    agent = _mk_agent()
    learner = ElasticLearner(agent, master, 3, 1)
    actors = [ElasticActor(i, _mk_agent(), master, 3, 1,
                           lambda ag, i=i: _collect(ag, n=7, seed=i))
              for i in (3, 9, 17)]
    for _ in range(3):
        for a in actors:
            a.round()
    got = learner.drain_mail()
    assert got == 3 * 3 * 7
    assert sorted(learner.drained) == [3, 9, 17]
    assert all(learner.drained[a.aid] == 3 for a in actors)
    # second drain with no new mail is a no-op
    assert learner.drain_mail() == 0
