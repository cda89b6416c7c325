"""3-process tests for A3C, APEX, IMPALA, ARS (reference analog:
test/frame/algorithms/test_{a3c,apex,impala,ars}.py)."""
import numpy as np
import pytest
import torch as t
import torch.nn as nn

from util_run_multi import run_multi


class TestA3C:
    def test_act_and_update(self):
        def fn(rank, world):
            from machin_amd.frame.algorithms import A3C
            from machin_amd.frame.helpers.servers import grad_server_helper

            import torch.nn as nn

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

            servers = grad_server_helper(
                [Actor, Critic], learning_rate=5e-3, reduce_batch_size=2
            )
            a3c = A3C(Actor(), Critic(), nn.MSELoss(reduction="sum"),
                      servers)
            group = world.groups["grad_server_group"]
            group.barrier()
            # act pulls from server
            action = a3c.act({"state": t.zeros(1, 4)})
            assert action[0].shape == (1, 1)
            # store + update pushes grads
            episode = [
                {
                    "state": {"state": t.rand(1, 4)},
                    "action": {"action": t.randint(0, 2, (1, 1))},
                    "next_state": {"state": t.rand(1, 4)},
                    "reward": 1.0,
                    "terminal": i == 4,
                }
                for i in range(5)
            ]
            a3c.store_episode(episode)
            a3c.update()
            group.barrier()
            return True

        assert all(run_multi(fn, timeout=180))


class TestDQNApex:
    def test_samplers_and_learner(self):
        def fn(rank, world):
            import time

            from machin_amd.frame.algorithms import DQNApex
            from machin_amd.frame.helpers.servers import model_server_helper

            import torch.nn as nn

            class QNet(nn.Module):
                def __init__(self):
                    super().__init__()
                    self.fc = nn.Linear(4, 2)

                def forward(self, state):
                    return self.fc(state)

            servers = model_server_helper(model_num=1)
            group = world.create_rpc_group("apex", ["0", "1", "2"])
            apex = DQNApex(
                QNet(), QNet(), t.optim.Adam, nn.MSELoss(reduction="sum"),
                group, servers, batch_size=8, replay_size=100,
            )
            group.barrier()
            if rank in (0, 1):
                # samplers: act (pulls model), store episodes
                for _ in range(3):
                    episode = [
                        {
                            "state": {"state": t.rand(1, 4)},
                            "action": {
                                "action": apex.act_discrete_with_noise(
                                    {"state": t.rand(1, 4)}
                                )
                            },
                            "next_state": {"state": t.rand(1, 4)},
                            "reward": 1.0,
                            "terminal": i == 4,
                        }
                        for i in range(5)
                    ]
                    apex.store_episode(episode)
            group.barrier()
            loss = None
            if rank == 2:
                loss = apex.update()
                assert isinstance(loss, float)
            group.barrier()
            return True

        assert all(run_multi(fn, timeout=180))


class TestIMPALA:
    def test_store_and_update(self):
        def fn(rank, world):
            from machin_amd.frame.algorithms import IMPALA
            from machin_amd.frame.helpers.servers import model_server_helper

            import torch.nn as nn

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

            servers = model_server_helper(model_num=1)
            group = world.create_rpc_group("impala", ["0", "1", "2"])
            impala = IMPALA(
                Actor(), Critic(), t.optim.Adam,
                nn.MSELoss(reduction="sum"), group, servers,
                batch_size=4,
            )
            group.barrier()
            if rank in (0, 1):
                length = 3 if rank == 0 else 2
                impala.store_episode(
                    [
                        {
                            "state": {"state": t.rand(1, 4)},
                            "action": {"action": t.zeros(1, 1, dtype=t.long)},
                            "next_state": {"state": t.rand(1, 4)},
                            "reward": 0.5,
                            "action_log_prob": -0.1,
                            "terminal": i == length - 1,
                        }
                        for i in range(length)
                    ]
                )
            group.barrier()
            if rank == 2:
                import time

                deadline = time.monotonic() + 10
                while (
                    impala.replay_buffer.all_size() < 2
                    and time.monotonic() < deadline
                ):
                    time.sleep(0.05)
                pl, vl = impala.update()
                assert pl == pl and vl == vl  # no NaN
            group.barrier()
            # everyone can act and sees the pushed model
            action = impala.act({"state": t.zeros(1, 4)})
            assert action[0].shape == (1, 1)
            group.barrier()
            return True

        assert all(run_multi(fn, timeout=180))

    def test_missing_log_prob_raises(self):
        def fn(rank, world):
            from machin_amd.frame.algorithms import IMPALA
            from machin_amd.frame.helpers.servers import model_server_helper

            import torch.nn as nn

            class Actor(nn.Module):
                def __init__(self):
                    super().__init__()
                    self.fc = nn.Linear(4, 2)

                def forward(self, state):
                    return self.fc(state), t.zeros(1, 1)

            class Critic(nn.Module):
                def __init__(self):
                    super().__init__()
                    self.fc = nn.Linear(4, 1)

                def forward(self, state):
                    return self.fc(state)

            servers = model_server_helper(model_num=1)
            group = world.create_rpc_group("impala", ["0", "1", "2"])
            impala = IMPALA(
                Actor(), Critic(), t.optim.Adam,
                nn.MSELoss(reduction="sum"), group, servers,
            )
            group.barrier()
            raised = False
            try:
                impala.store_episode(
                    [
                        {
                            "state": {"state": t.rand(1, 4)},
                            "action": {"action": t.zeros(1, 1)},
                            "next_state": {"state": t.rand(1, 4)},
                            "reward": 0.0,
                            "terminal": True,
                        }
                    ]
                )
            except ValueError:
                raised = True
            group.barrier()
            return raised

        assert all(run_multi(fn, timeout=180))


class TestARS:
    def test_act_store_update(self):
        def fn(rank, world):
            from machin_amd.frame.algorithms import ARS
            from machin_amd.frame.helpers.servers import model_server_helper

            import torch.nn as nn

            class Actor(nn.Module):
                def __init__(self):
                    super().__init__()
                    self.fc = nn.Linear(4, 2, bias=False)

                def forward(self, state):
                    return t.argmax(self.fc(state), dim=1)

            servers = model_server_helper(model_num=1)
            group = world.create_rpc_group("ars", ["0", "1", "2"])
            ars = ARS(
                Actor(), t.optim.SGD, group, servers,
                noise_std_dev=0.1, learning_rate=0.1,
                noise_size=100000, rollout_num=6, used_rollout_num=6,
                normalize_state=True,
            )
            group.barrier()
            types = ars.get_actor_types()
            assert types[0] == "original"
            assert len(types) == 1 + 2 * len(ars._my_rollouts)
            state = {"state": t.zeros(1, 4)}
            ars.act(state, "original")
            for at in types:
                ars.act(state, at)
                if at != "original":
                    ars.store_reward(
                        1.0 if at.startswith("pos") else 0.0, at
                    )
            raised = False
            try:
                ars.act(state, "bogus_type")
            except ValueError:
                raised = True
            group.barrier()
            ars.update()
            group.barrier()
            # all members hold identical params after update
            w = next(iter(ars.actor.parameters())).detach()
            group.pair(f"w_{world.name}", w.clone())
            group.barrier()
            w0 = group.get_paired("w_0").to_here()
            same = t.allclose(w, w0, atol=1e-6)
            group.barrier()
            return raised and same

        assert all(run_multi(fn, timeout=180))
