"""IMPALA shared-memory rollout-ring mode (round-1 VERDICT next #1):
actors write fixed-shape segments, the learner drains + batch-updates.
CPU tests here; the GPU pipeline is covered by bench.py --e2e and
tests/test_rollout_ring.py gpu cases."""
import numpy as np
import torch as t
import torch.nn as nn

from util_run_multi import run_multi


def _make_episode(length, seed=0):
    g = t.Generator().manual_seed(seed)
    return [
        {
            "state": {"state": t.rand(1, 4, generator=g)},
            "action": {"action": t.randint(0, 2, (1, 1), generator=g)},
            "next_state": {"state": t.rand(1, 4, generator=g)},
            "reward": float(t.rand(1, generator=g)),
            "terminal": i == length - 1,
            "action_log_prob": float(-t.rand(1, generator=g)),
        }
        for i in range(length)
    ]


def _models():
    class Actor(nn.Module):
        def __init__(self):
            super().__init__()
            self.fc = nn.Linear(4, 2)

        def forward(self, state, action=None):
            logits = self.fc(state)
            dist = t.distributions.Categorical(logits=logits)
            if action is None:
                action = dist.sample().view(-1, 1)
            return (
                action,
                dist.log_prob(action.view(-1)).view(-1, 1),
                dist.entropy().view(-1, 1),
            )

    class Critic(nn.Module):
        def __init__(self):
            super().__init__()
            self.fc = nn.Linear(4, 1)

        def forward(self, state):
            return self.fc(state)

    return Actor, Critic


class TestRingMatchesRpcMode:
    def test_losses_identical(self):
        """Ring-mode update must compute the SAME losses as the
        RPC-episode mode for the same episodes and weights."""
        def fn(rank, world):
            import copy

            from machin_amd.frame.algorithms.impala import IMPALA
            from machin_amd.frame.helpers.servers import (
                model_server_helper,
            )
            from machin_amd.parallel.rollout_ring import (
                make_episode_ring,
            )

            Actor, Critic = _models()
            servers = model_server_helper(model_num=1)
            g1 = world.create_rpc_group("ig1", ["0"])
            g2 = world.create_rpc_group("ig2", ["0"])
            t.manual_seed(0)
            a1, c1 = Actor(), Critic()
            a2, c2 = copy.deepcopy(a1), copy.deepcopy(c1)
            f_rpc = IMPALA(a1, c1, t.optim.SGD, nn.MSELoss(),
                           g1, servers, batch_size=3)
            f_ring = IMPALA(a2, c2, t.optim.SGD, nn.MSELoss(),
                            g2, servers, batch_size=3)
            episodes = [
                _make_episode(5, seed=1),
                _make_episode(3, seed=2),
                _make_episode(5, seed=3),
            ]
            ring, _ = make_episode_ring(episodes[0], unroll=5, slots=8)
            f_ring.use_rollout_ring(ring)
            for ep in episodes:
                f_rpc.store_episode(ep)
                f_ring.store_episode(ep)
            al1, vl1 = f_rpc.update(update_target=False)
            al2, vl2 = f_ring.update(update_target=False)
            assert abs(al1 - al2) < 1e-5, (al1, al2)
            assert abs(vl1 - vl2) < 1e-5, (vl1, vl2)
            # and the resulting parameters match
            for p, q in zip(a1.parameters(), a2.parameters()):
                assert t.allclose(p, q, atol=1e-6)
            for p, q in zip(c1.parameters(), c2.parameters()):
                assert t.allclose(p, q, atol=1e-6)
            return True

        assert all(run_multi(fn, world_size=1, timeout=240))

    def test_long_episode_is_split(self):
        """Episodes longer than the unroll become several segments."""
        from machin_amd.parallel.rollout_ring import make_episode_ring

        ep = _make_episode(12, seed=7)
        ring, codec = make_episode_ring(ep, unroll=5, slots=8)
        codec.write_episode(ring, ep)
        idx = ring.drain(8, timeout=2.0)
        while len(idx) < 3:
            idx += ring.drain(8, timeout=2.0)
        assert len(idx) == 3  # 5 + 5 + 2
        lengths = sorted(int(ring.slot(i)["length"]) for i in idx)
        assert lengths == [2, 5, 5]
        # the tail segment's padding is terminal-masked
        tail = next(i for i in idx if int(ring.slot(i)["length"]) == 2)
        assert (ring.slot(tail)["terminal"][2:] == 1.0).all()


class TestRingAcrossProcesses:
    def test_actors_write_learner_updates(self):
        sample = _make_episode(4, seed=0)
        from machin_amd.parallel.rollout_ring import make_episode_ring

        ring, _ = make_episode_ring(sample, unroll=6, slots=16)

        def fn(rank, world, ring):
            from machin_amd.frame.algorithms.impala import IMPALA
            from machin_amd.frame.helpers.servers import (
                model_server_helper,
            )

            Actor, Critic = _models()
            servers = model_server_helper(model_num=1)
            group = world.create_rpc_group("impala", ["0", "1", "2"])
            frame = IMPALA(
                Actor(), Critic(), t.optim.Adam, nn.MSELoss(),
                group, servers, batch_size=8,
            )
            frame.use_rollout_ring(ring)
            group.barrier()
            if rank in (1, 2):
                for e in range(3):
                    frame.store_episode(
                        _make_episode(4, seed=rank * 10 + e)
                    )
            group.barrier()
            result = None
            if rank == 0:
                result = frame.update()
                assert all(np.isfinite(v) for v in result)
            group.barrier()
            return result

        results = run_multi(fn, world_size=3, args=(ring,), timeout=240)
        assert results[0] is not None
