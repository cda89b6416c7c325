"""Full-training gates mirroring the reference CI
(CartPole smoothed > 150, Pendulum smoothed > -400)."""
import pytest
import torch as t
import torch.nn as nn

from machin_amd.frame.algorithms import (
    A2C,
    DDPG,
    PPO,
    SAC,
    TD3,
    DQNPer,
)

from util_models import (
    Critic,
    DetActor,
    GaussianActor,
    QNet,
    StochDiscreteActor,
    VCritic,
)
from util_train import train_cartpole, train_pendulum

pytestmark = pytest.mark.slow


class TestDQNPerFullTrain:
    def test_full_train(self):
        fr = DQNPer(
            QNet(), QNet(), t.optim.Adam, nn.MSELoss(reduction="none"),
            batch_size=64, learning_rate=1e-3, epsilon_decay=0.995,
            update_rate=0.01,
        )
        solved = train_cartpole(
            fr, lambda s: fr.act_discrete_with_noise({"state": s}).item()
        )
        assert solved, "DQNPer did not solve CartPole"


class TestA2CFullTrain:
    def test_full_train(self):
        fr = A2C(
            StochDiscreteActor(), VCritic(), t.optim.Adam, nn.MSELoss(),
            entropy_weight=0.01, gae_lambda=0.97,
            actor_learning_rate=2e-3, critic_learning_rate=2e-3,
            actor_update_times=3, critic_update_times=6,
        )

        def update_fn(_):
            fr.update()

        solved = train_cartpole(
            fr,
            lambda s: fr.act({"state": s})[0].item(),
            update_fn=update_fn,
            max_episodes=1200,
        )
        assert solved, "A2C did not solve CartPole"


class TestPPOFullTrain:
    def test_full_train(self):
        fr = PPO(
            StochDiscreteActor(), VCritic(), t.optim.Adam, nn.MSELoss(),
            entropy_weight=0.01, gae_lambda=0.97,
            actor_learning_rate=2e-3, critic_learning_rate=2e-3,
            actor_update_times=6, critic_update_times=10,
        )

        def update_fn(_):
            fr.update()

        solved = train_cartpole(
            fr,
            lambda s: fr.act({"state": s})[0].item(),
            update_fn=update_fn,
            max_episodes=1000,
        )
        assert solved, "PPO did not solve CartPole"


class TestDDPGFullTrain:
    def test_full_train(self):
        fr = DDPG(
            DetActor(), DetActor(), Critic(), Critic(),
            t.optim.Adam, nn.MSELoss(),
            batch_size=100, update_rate=0.005,
            actor_learning_rate=5e-4, critic_learning_rate=1e-3,
        )

        def act(s):
            return fr.act_with_noise(
                {"state": s}, noise_param=(0.0, 0.3), mode="normal"
            ).clamp(-2, 2)

        solved = train_pendulum(fr, act)
        assert solved, "DDPG did not solve Pendulum"


class TestTD3FullTrain:
    def test_full_train(self):
        fr = TD3(
            DetActor(), DetActor(), Critic(), Critic(), Critic(), Critic(),
            t.optim.Adam, nn.MSELoss(),
            batch_size=100, update_rate=0.005,
            actor_learning_rate=5e-4, critic_learning_rate=1e-3,
        )

        def act(s):
            return fr.act_with_noise(
                {"state": s}, noise_param=(0.0, 0.3), mode="normal"
            ).clamp(-2, 2)

        solved = train_pendulum(fr, act)
        assert solved, "TD3 did not solve Pendulum"


class TestSACFullTrain:
    def test_full_train(self):
        fr = SAC(
            GaussianActor(), Critic(), Critic(), Critic(), Critic(),
            t.optim.Adam, nn.MSELoss(),
            batch_size=100, target_entropy=-1.0,
            actor_learning_rate=1e-3, critic_learning_rate=2e-3,
        )

        def act(s):
            return fr.act({"state": s})[0].clamp(-2, 2)

        solved = train_pendulum(fr, act)
        assert solved, "SAC did not solve Pendulum"
