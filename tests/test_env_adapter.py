"""Generic env adapter + PixelCatch pixel env (round-1 VERDICT next
#7: arbitrary user envs and a non-toy built-in)."""
import numpy as np
import pytest
import torch as t
import torch.nn as nn

from machin_amd.env.envs import PixelCatchEnv, make
from machin_amd.env.wrappers import (
    GymAdapter,
    ParallelWrapperDummy,
    ParallelWrapperSubProc,
    adapt,
    validate_env,
)


class ClassicUserEnv:
    """Plain gym-classic user env: 4-tuple step, seed(), spaces."""

    def __init__(self):
        self.x = 0.0

    def reset(self):
        self.x = 0.0
        return np.array([self.x], dtype=np.float32)

    def step(self, action):
        self.x += float(action)
        return (
            np.array([self.x], dtype=np.float32),
            -abs(self.x),
            abs(self.x) > 3,
            {},
        )

    def seed(self, seed=None):
        return [seed]


class GymnasiumStyleEnv:
    """Modern API: reset(seed=...)->(obs, info), 5-tuple step, no
    seed(), no spaces."""

    def __init__(self):
        self.n = 0

    def reset(self, seed=None, options=None):
        self.n = 0
        return np.zeros(3, dtype=np.float32), {}

    def step(self, action):
        self.n += 1
        terminated = self.n >= 4
        truncated = self.n >= 3 and not terminated
        return (
            np.full(3, self.n, dtype=np.float32),
            1.0,
            terminated,
            truncated,
            {},
        )


class TestGymAdapter:
    def test_classic_env_passthrough(self):
        env = GymAdapter(ClassicUserEnv())
        obs = env.reset()
        assert obs.shape == (1,)
        obs, r, d, info = env.step(1.0)
        assert obs[0] == 1.0 and not d
        env.seed(3)
        env.render()
        env.close()

    def test_gymnasium_style_env(self):
        env = GymAdapter(GymnasiumStyleEnv())
        obs = env.reset()
        assert obs.shape == (3,)  # info tuple unwrapped
        _, r, d, info = env.step(0)
        assert not d
        env.step(0)
        obs, r, d, info = env.step(0)  # truncated at n=3
        assert d and info.get("TimeLimit.truncated")
        env.seed(7)  # routed through reset(seed=...)
        # observation_space inferred from the last observation
        assert env.observation_space.shape == (3,)

    def test_validate_rejects_non_env(self):
        with pytest.raises(TypeError, match="reset"):
            validate_env(object())

    def test_parallel_wrappers_take_adapted_creators(self):
        wrapper = ParallelWrapperDummy(
            [adapt(lambda *_: GymnasiumStyleEnv())] * 2
        )
        obs = wrapper.reset()
        assert len(obs) == 2 and obs[0].shape == (3,)
        obs, rew, done, info = wrapper.step([0, 0])
        assert rew == [1.0, 1.0]
        wrapper.close()

    def test_subproc_wrapper_with_adapter(self):
        wrapper = ParallelWrapperSubProc(
            [adapt(lambda *_: ClassicUserEnv())] * 2
        )
        try:
            obs = wrapper.reset()
            assert len(obs) == 2
            obs, rew, done, info = wrapper.step([1.0, -1.0])
            assert obs[0][0] == 1.0 and obs[1][0] == -1.0
        finally:
            wrapper.close()


class TestPixelCatch:
    def test_observation_contract(self):
        env = PixelCatchEnv(seed=0)
        obs = env.reset()
        assert obs.shape == (4, 84, 84) and obs.dtype == np.uint8
        obs, r, d, _ = env.step(1)
        assert obs.shape == (4, 84, 84)
        assert env.action_space.n == 3
        assert env.observation_space.shape == (4, 84, 84)
        assert make("PixelCatch-v0").__class__ is PixelCatchEnv

    def test_catch_and_miss_rewards(self):
        env = PixelCatchEnv(seed=0, balls=1)
        env.reset()
        # oracle policy: walk the paddle under the ball -> +1
        total, d = 0.0, False
        while not d:
            a = 2 if env.ball_x > env.paddle_x else (
                0 if env.ball_x < env.paddle_x else 1
            )
            _, r, d, _ = env.step(a)
            total += r
        assert total == 1.0
        # adversarial policy: run away -> -1
        env = PixelCatchEnv(seed=0, balls=1)
        env.reset()
        total, d = 0.0, False
        while not d:
            a = 0 if env.ball_x > env.paddle_x else 2
            _, r, d, _ = env.step(a)
            total += r
        assert total == -1.0

    def test_deterministic_under_seed(self):
        a = PixelCatchEnv(seed=5)
        b = PixelCatchEnv(seed=5)
        oa, ob = a.reset(), b.reset()
        assert (oa == ob).all()
        for _ in range(30):
            ra = a.step(2)
            rb = b.step(2)
            assert (ra[0] == rb[0]).all() and ra[1] == rb[1]

    def test_nature_cnn_trains_on_pixel_catch(self):
        """DQN + Nature CNN steps end-to-end on real pixel frames
        (shape/dtype pipeline check; convergence is the gpu gate)."""
        from machin_amd.frame.algorithms import DQN
        from machin_amd.model.nets.nature_cnn import NatureCNN

        class QNet(nn.Module):
            def __init__(self):
                super().__init__()
                self.cnn = NatureCNN(4, feature_dim=128)
                self.head = nn.Linear(128, 3)

            def forward(self, state):
                return self.head(self.cnn(state.float() / 255.0))

        frame = DQN(QNet(), QNet(), t.optim.Adam, nn.MSELoss(),
                    batch_size=8)
        env = PixelCatchEnv(seed=0, balls=1)
        obs = env.reset()
        episode, d = [], False
        while not d and len(episode) < 40:
            st = t.from_numpy(obs).unsqueeze(0)
            act = frame.act_discrete_with_noise(
                {"state": st}
            )
            obs2, r, d, _ = env.step(int(act.item()))
            episode.append({
                "state": {"state": st},
                "action": {"action": act},
                "next_state": {"state": t.from_numpy(obs2).unsqueeze(0)},
                "reward": r,
                "terminal": d,
            })
            obs = obs2
        frame.store_episode(episode)
        loss = frame.update()
        assert np.isfinite(loss)


class TestPixelCatchSubProc:
    def test_pixel_env_through_subprocess_pool(self):
        """PixelCatch frames (84x84x4 u8) round-trip the subprocess
        env pool (dill creators, shared result queue)."""
        wrapper = ParallelWrapperSubProc(
            [lambda *_: PixelCatchEnv(seed=0, balls=1)] * 2
        )
        try:
            obs = wrapper.reset()
            assert len(obs) == 2
            assert obs[0].shape == (4, 84, 84)
            obs, rew, done, info = wrapper.step([1, 2])
            assert obs[1].shape == (4, 84, 84)
        finally:
            wrapper.close()
