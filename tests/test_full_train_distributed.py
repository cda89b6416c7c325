"""Distributed full-train gates: A3C, APEX-DQN, IMPALA, ARS solve
CartPole across 3 processes (reference CI table in BASELINE.md)."""
import pytest

from util_run_multi import run_multi

pytestmark = pytest.mark.slow


def _run_cartpole_worker(frame, act_fn, store_fn, update_fn, group,
                         max_episodes=1000, target=150.0, wins_needed=5):
    """Per-process training loop; returns best smoothed reward. Any
    process pairing 'solved' stops everyone."""
    import torch as t

    from machin_amd.env.envs.classic_control import CartPoleEnv

    env = CartPoleEnv(seed=None)
    smoothed, wins, best = 0.0, 0, 0.0
    for episode in range(max_episodes):
        if group.is_paired("solved"):
            return True
        obs = t.tensor(env.reset(), dtype=t.float32).view(1, 4)
        total_reward = 0.0
        transitions = []
        done = False
        while not done:
            with t.no_grad():
                action = act_fn(obs)
            obs_next, reward, done, _ = env.step(int(action))
            obs_next = t.tensor(obs_next, dtype=t.float32).view(1, 4)
            total_reward += reward
            transitions.append(
                {
                    "state": {"state": obs},
                    "action": {"action": t.tensor([[int(action)]])},
                    "next_state": {"state": obs_next},
                    "reward": reward,
                    "terminal": done and env.steps < env.max_episode_steps,
                }
            )
            obs = obs_next
        store_fn(transitions)
        update_fn(transitions)
        smoothed = smoothed * 0.9 + total_reward * 0.1
        best = max(best, smoothed)
        if smoothed > target:
            wins += 1
            if wins >= wins_needed:
                try:
                    group.pair("solved", True)
                except RuntimeError:
                    pass
                return True
        else:
            wins = 0
    return group.is_paired("solved")


class TestA3CFullTrain:
    def test_full_train(self):
        def fn(rank, world):
            import torch as t
            import torch.nn as nn

            from machin_amd.frame.algorithms import A3C
            from machin_amd.frame.helpers.servers import grad_server_helper

            class Actor(nn.Module):
                def __init__(self):
                    super().__init__()
                    self.fc1 = nn.Linear(4, 16)
                    self.fc2 = nn.Linear(16, 2)

                def forward(self, state, action=None):
                    logits = self.fc2(t.relu(self.fc1(state)))
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
                    self.fc1 = nn.Linear(4, 16)
                    self.fc2 = nn.Linear(16, 1)

                def forward(self, state):
                    return self.fc2(t.relu(self.fc1(state)))

            t.manual_seed(rank)
            servers = grad_server_helper(
                [Actor, Critic], learning_rate=5e-3, reduce_batch_size=3,
                reduce_method="mean",
            )
            a3c = A3C(
                Actor(), Critic(), nn.MSELoss(reduction="sum"), servers,
                entropy_weight=0.01, gae_lambda=0.97,
                actor_update_times=2, critic_update_times=4,
            )
            group = world.groups["grad_server_group"]
            group.barrier()
            solved = _run_cartpole_worker(
                a3c,
                lambda s: a3c.act({"state": s})[0].item(),
                a3c.store_episode,
                lambda _: a3c.update(),
                group,
                max_episodes=3000,
            )
            group.barrier()
            return solved

        # stochastic gate (reference: "weakly reproducible",
        # README.md:131-135): one retry before declaring failure
        for _attempt in range(2):
            results = run_multi(fn, timeout=900)
            if any(results):
                break
        assert any(results), "A3C did not solve CartPole on any process"


class TestApexFullTrain:
    def test_full_train(self):
        def fn(rank, world):
            import time

            import torch as t
            import torch.nn as nn

            from machin_amd.frame.algorithms import DQNApex
            from machin_amd.frame.helpers.servers import model_server_helper

            class QNet(nn.Module):
                def __init__(self):
                    super().__init__()
                    self.fc1 = nn.Linear(4, 16)
                    self.fc2 = nn.Linear(16, 16)
                    self.fc3 = nn.Linear(16, 2)

                def forward(self, state):
                    a = t.relu(self.fc1(state))
                    return self.fc3(t.relu(self.fc2(a)))

            t.manual_seed(rank)
            servers = model_server_helper(model_num=1)
            group = world.create_rpc_group("apex", ["0", "1", "2"])
            apex = DQNApex(
                QNet(), QNet(), t.optim.Adam,
                nn.MSELoss(reduction="sum"), group, servers,
                batch_size=64, learning_rate=1e-3, epsilon_decay=0.9997,
                update_rate=0.005, replay_size=10000,
            )
            group.barrier()
            if rank == 2:
                # learner: update until samplers declare solved
                apex.set_sync(False)
                deadline = time.monotonic() + 560
                while (
                    not group.is_paired("solved")
                    and time.monotonic() < deadline
                ):
                    if apex.replay_buffer.all_size() > 500:
                        apex.update()
                    else:
                        time.sleep(0.05)
                solved = group.is_paired("solved")
            else:
                # reference behavior: sync the model once per episode,
                # not once per act (apex.py:123-139). Exploration uses
                # a per-episode epsilon with a floor so the fast
                # samplers cannot burn out exploration before the
                # learner catches up.
                apex.set_sync(False)
                episode_counter = [0]

                def store(transitions):
                    apex.store_episode(transitions)
                    apex.manual_sync()
                    episode_counter[0] += 1
                    apex.epsilon = max(0.08, 0.995 ** episode_counter[0])

                solved = _run_cartpole_worker(
                    apex,
                    lambda s: apex.act_discrete_with_noise(
                        {"state": s}, decay_epsilon=False
                    ).item(),
                    store,
                    lambda _: None,
                    group,
                    max_episodes=6000,
                )
            group.barrier()
            return solved

        # the reference calls these gates "weakly reproducible"
        # (README.md:131-135); APEX's sampler/learner race makes
        # this the jitteriest one (~1 in 7 cold runs misses the
        # window), so allow one retry before declaring failure
        for attempt in range(2):
            results = run_multi(fn, timeout=700)
            if any(results[:2]):
                break
        assert any(results[:2]), "APEX samplers never reached the target"


class TestImpalaFullTrain:
    def test_full_train(self):
        def fn(rank, world):
            import time

            import torch as t
            import torch.nn as nn

            from machin_amd.frame.algorithms import IMPALA
            from machin_amd.frame.helpers.servers import model_server_helper

            class Actor(nn.Module):
                def __init__(self):
                    super().__init__()
                    self.fc1 = nn.Linear(4, 16)
                    self.fc2 = nn.Linear(16, 2)

                def forward(self, state, action=None):
                    logits = self.fc2(t.relu(self.fc1(state)))
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
                    self.fc1 = nn.Linear(4, 16)
                    self.fc2 = nn.Linear(16, 1)

                def forward(self, state):
                    return self.fc2(t.relu(self.fc1(state)))

            t.manual_seed(rank)
            servers = model_server_helper(model_num=1)
            group = world.create_rpc_group("impala", ["0", "1", "2"])
            impala = IMPALA(
                Actor(), Critic(), t.optim.Adam,
                nn.MSELoss(reduction="sum"), group, servers,
                batch_size=4, learning_rate=5e-3, entropy_weight=0.01,
            )
            group.barrier()
            if rank == 2:
                impala.set_sync(False)
                deadline = time.monotonic() + 480
                while (
                    not group.is_paired("solved")
                    and time.monotonic() < deadline
                ):
                    if impala.replay_buffer.all_size() >= 2:
                        impala.update()
                    else:
                        time.sleep(0.02)
                solved = group.is_paired("solved")
            else:
                def store(transitions):
                    for tr in transitions:
                        with t.no_grad():
                            lp = impala._eval_act(
                                tr["state"], tr["action"]
                            )[1]
                        tr["action_log_prob"] = float(lp.item())
                    impala.store_episode(transitions)

                solved = _run_cartpole_worker(
                    impala,
                    lambda s: impala.act({"state": s})[0].item(),
                    store,
                    lambda _: None,
                    group,
                    max_episodes=4000,
                )
            group.barrier()
            return solved

        # stochastic gate (reference: "weakly reproducible",
        # README.md:131-135): one retry before declaring failure
        for _attempt in range(2):
            results = run_multi(fn, timeout=600)
            if any(results[:2]):
                break
        assert any(results[:2]), "IMPALA samplers never reached the target"


class TestARSFullTrain:
    def test_full_train(self):
        def fn(rank, world):
            import torch as t
            import torch.nn as nn

            from machin_amd.env.envs.classic_control import CartPoleEnv
            from machin_amd.frame.algorithms import ARS
            from machin_amd.frame.helpers.servers import model_server_helper

            class Actor(nn.Module):
                def __init__(self):
                    super().__init__()
                    self.fc = nn.Linear(4, 2, bias=False)

                def forward(self, state):
                    return t.argmax(self.fc(state), dim=1)

            t.manual_seed(7)  # identical init everywhere
            servers = model_server_helper(model_num=1)
            group = world.create_rpc_group("ars", ["0", "1", "2"])
            ars = ARS(
                Actor(), t.optim.SGD, group, servers,
                noise_std_dev=0.1, learning_rate=0.1, noise_size=100000,
                rollout_num=6, used_rollout_num=6, normalize_state=True,
            )
            group.barrier()
            env = CartPoleEnv(seed=rank)

            def run_episode(actor_type):
                obs = t.tensor(env.reset(), dtype=t.float32).view(1, 4)
                total = 0.0
                done = False
                while not done:
                    with t.no_grad():
                        a = ars.act({"state": obs}, actor_type)
                    obs_np, r, done, _ = env.step(int(a.item()))
                    obs = t.tensor(obs_np, dtype=t.float32).view(1, 4)
                    total += r
                return total

            me = world.name
            smoothed, wins = 0.0, 0
            solved = False
            for it in range(200):
                for at in ars.get_actor_types():
                    reward = run_episode(at)
                    if at == "original":
                        smoothed = smoothed * 0.8 + reward * 0.2
                    else:
                        ars.store_reward(reward, at)
                ars.update()
                if smoothed > 150:
                    wins += 1
                else:
                    wins = 0
                # collective agreement on stopping (every member runs
                # the same pair/barrier sequence each iteration)
                group.pair(f"vote_{me}_{it}", wins >= 5)
                group.barrier()
                votes = [
                    group.get_paired(f"vote_{m}_{it}").to_here()
                    for m in group.get_group_members()
                ]
                group.barrier()
                group.unpair(f"vote_{me}_{it}")
                if any(votes):
                    solved = True
                    break
            group.barrier()
            return solved

        # stochastic gate (reference: "weakly reproducible",
        # README.md:131-135): one retry before declaring failure
        for _attempt in range(2):
            results = run_multi(fn, timeout=900)
            if any(results):
                break
        assert any(results), "ARS did not solve CartPole"


class TestDDPGApexFullTrain:
    def test_full_train(self):
        """APEX-DDPG solves Pendulum (reference gate:
        test/frame/algorithms/test_apex.py:349-358, smoothed > -400)."""
        def fn(rank, world):
            import time

            import torch as t
            import torch.nn as nn

            from machin_amd.env.envs.classic_control import PendulumEnv
            from machin_amd.frame.algorithms import DDPGApex
            from machin_amd.frame.helpers.servers import model_server_helper

            class Actor(nn.Module):
                def __init__(self):
                    super().__init__()
                    self.fc1 = nn.Linear(3, 16)
                    self.fc2 = nn.Linear(16, 16)
                    self.fc3 = nn.Linear(16, 1)

                def forward(self, state):
                    a = t.relu(self.fc1(state))
                    a = t.relu(self.fc2(a))
                    return t.tanh(self.fc3(a)) * 2.0

            class Critic(nn.Module):
                def __init__(self):
                    super().__init__()
                    self.fc1 = nn.Linear(4, 16)
                    self.fc2 = nn.Linear(16, 16)
                    self.fc3 = nn.Linear(16, 1)

                def forward(self, state, action):
                    x = t.cat([state, action], dim=1)
                    return self.fc3(t.relu(self.fc2(t.relu(self.fc1(x)))))

            t.manual_seed(rank)
            servers = model_server_helper(model_num=1)
            group = world.create_rpc_group("apex", ["0", "1", "2"])
            apex = DDPGApex(
                Actor(), Actor(), Critic(), Critic(), t.optim.Adam,
                nn.MSELoss(reduction="sum"), group, servers,
                batch_size=100, update_rate=0.005,
                actor_learning_rate=5e-4, critic_learning_rate=1e-3,
                replay_size=20000,
            )
            group.barrier()
            if rank == 2:
                apex.set_sync(False)
                deadline = time.monotonic() + 480
                while (
                    not group.is_paired("solved")
                    and time.monotonic() < deadline
                ):
                    if apex.replay_buffer.all_size() > 500:
                        apex.update()
                    else:
                        time.sleep(0.05)
                solved = group.is_paired("solved")
            else:
                apex.set_sync(False)
                env = PendulumEnv(seed=rank)
                smoothed, wins = -1600.0, 0
                solved = False
                deadline = time.monotonic() + 480
                while time.monotonic() < deadline:
                    if group.is_paired("solved"):
                        solved = True
                        break
                    obs = t.tensor(env.reset(), dtype=t.float32).view(1, 3)
                    total = 0.0
                    transitions = []
                    done = False
                    while not done:
                        with t.no_grad():
                            action = apex.act_with_noise(
                                {"state": obs}, noise_param=(0.0, 0.3),
                                mode="normal",
                            ).clamp(-2, 2)
                        o, r, done, _ = env.step(action.view(-1).numpy())
                        o = t.tensor(o, dtype=t.float32).view(1, 3)
                        total += r
                        transitions.append(
                            {
                                "state": {"state": obs},
                                "action": {"action": action.view(1, 1)},
                                "next_state": {"state": o},
                                "reward": r / 10.0,
                                "terminal": False,
                            }
                        )
                        obs = o
                    apex.store_episode(transitions)
                    apex.manual_sync()
                    smoothed = smoothed * 0.9 + total * 0.1
                    if smoothed > -400:
                        wins += 1
                        if wins >= 5:
                            try:
                                group.pair("solved", True)
                            except RuntimeError:
                                pass
                            solved = True
                            break
                    else:
                        wins = 0
            group.barrier()
            return solved

        # stochastic gate (reference: "weakly reproducible",
        # README.md:131-135): one retry before declaring failure
        for _attempt in range(2):
            results = run_multi(fn, timeout=600)
            if any(results[:2]):
                break
        assert any(results[:2]), "APEX-DDPG samplers never reached -400"
