import os

import pytest
import torch as t
import torch.nn as nn

from machin_amd.env.envs.classic_control import CartPoleEnv
from machin_amd.frame.algorithms.dqn import DQN


class QNet(nn.Module):
    def __init__(self, state_dim=4, action_num=2):
        super().__init__()
        self.fc1 = nn.Linear(state_dim, 16)
        self.fc2 = nn.Linear(16, 16)
        self.fc3 = nn.Linear(16, action_num)

    def forward(self, state):
        a = t.relu(self.fc1(state))
        a = t.relu(self.fc2(a))
        return self.fc3(a)


def make_dqn(mode="double", **kwargs):
    return DQN(
        QNet(),
        QNet(),
        t.optim.Adam,
        nn.MSELoss(reduction="sum"),
        replay_size=10000,
        mode=mode,
        **kwargs,
    )


def random_transition():
    return {
        "state": {"state": t.rand(1, 4)},
        "action": {"action": t.randint(0, 2, (1, 1))},
        "next_state": {"state": t.rand(1, 4)},
        "reward": 1.0,
        "terminal": False,
    }


class TestDQNApi:
    @pytest.mark.parametrize("mode", ["vanilla", "fixed_target", "double"])
    def test_update(self, mode):
        dqn = make_dqn(mode, batch_size=8)
        dqn.store_episode([random_transition() for _ in range(20)])
        loss = dqn.update()
        assert isinstance(loss, float)

    def test_act(self):
        dqn = make_dqn()
        act = dqn.act_discrete({"state": t.rand(1, 4)})
        assert act.shape == (1, 1) and act.dtype == t.long
        act = dqn.act_discrete_with_noise({"state": t.rand(1, 4)})
        assert act.shape == (1, 1)
        assert dqn.epsilon < 1.0

    def test_criticize(self):
        dqn = make_dqn()
        q = dqn._criticize({"state": t.rand(3, 4)})
        assert q.shape == (3, 2)

    def test_save_load(self, tmp_path):
        dqn = make_dqn()
        dqn.store_episode([random_transition() for _ in range(20)])
        dqn.update()
        dqn.save(str(tmp_path))
        files = os.listdir(tmp_path)
        assert any(f.startswith("qnet_target_") for f in files)
        dqn2 = make_dqn()
        dqn2.load(str(tmp_path))
        for p1, p2 in zip(dqn.qnet_target.parameters(), dqn2.qnet.parameters()):
            assert t.allclose(p1, p2)

    def test_config_init(self, tmp_path):
        import __main__

        __main__.QNet = QNet
        config = DQN.generate_config({})
        config["frame_config"]["models"] = ["QNet", "QNet"]
        config["frame_config"]["batch_size"] = 8
        dqn = DQN.init_from_config(config)
        dqn.store_episode([random_transition() for _ in range(20)])
        assert isinstance(dqn.update(), float)

    def test_lr_scheduler(self):
        dqn = DQN(
            QNet(),
            QNet(),
            t.optim.Adam,
            nn.MSELoss(),
            lr_scheduler=t.optim.lr_scheduler.StepLR,
            lr_scheduler_args=((10,),),
            lr_scheduler_kwargs=({"gamma": 0.9},),
        )
        dqn.update_lr_scheduler()


@pytest.mark.slow
class TestDQNFullTrain:
    def test_full_train(self):
        """Solve CartPole: smoothed reward > 150 for 5 consecutive
        episodes (reference gate: test/frame/algorithms/test_dqn.py:50)."""
        t.manual_seed(0)
        dqn = make_dqn(
            "double",
            batch_size=64,
            learning_rate=1e-3,
            epsilon_decay=0.995,
            update_rate=0.01,
        )
        env = CartPoleEnv(seed=0)
        max_episodes = 600
        smoothed = 0.0
        wins = 0
        for episode in range(max_episodes):
            obs = t.tensor(env.reset(), dtype=t.float32).view(1, 4)
            total_reward = 0.0
            transitions = []
            done = False
            while not done:
                with t.no_grad():
                    action = dqn.act_discrete_with_noise({"state": obs})
                obs_next, reward, done, _ = env.step(action.item())
                obs_next = t.tensor(obs_next, dtype=t.float32).view(1, 4)
                total_reward += reward
                transitions.append(
                    {
                        "state": {"state": obs},
                        "action": {"action": action},
                        "next_state": {"state": obs_next},
                        "reward": reward,
                        "terminal": done and env.steps < env.max_episode_steps,
                    }
                )
                obs = obs_next
            dqn.store_episode(transitions)
            if dqn.replay_buffer.size() > 500:
                for _ in range(min(len(transitions), 50)):
                    dqn.update()
            smoothed = smoothed * 0.9 + total_reward * 0.1
            if smoothed > 150:
                wins += 1
                if wins >= 5:
                    return
            else:
                wins = 0
        pytest.fail(f"DQN did not solve CartPole, smoothed={smoothed:.1f}")
