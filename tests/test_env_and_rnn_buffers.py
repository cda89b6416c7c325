"""Tests for env wrappers and RNN sequence buffers."""
import numpy as np
import pytest
import torch as t

from machin_amd.env.envs.classic_control import CartPoleEnv, PendulumEnv, make
from machin_amd.env.wrappers.parallel import (
    ParallelWrapperDummy,
    ParallelWrapperSubProc,
)
from machin_amd.frame.buffers.rnn_buffers import (
    RNNBuffer,
    RNNPrioritizedBuffer,
)

from test_buffer import make_episode


class TestEnvs:
    def test_cartpole_contract(self):
        env = CartPoleEnv(seed=3)
        obs = env.reset()
        assert obs.shape == (4,)
        total = 0
        done = False
        while not done:
            obs, r, done, info = env.step(env.action_space.sample())
            total += r
        assert total >= 1

    def test_pendulum_contract(self):
        env = PendulumEnv(seed=3)
        obs = env.reset()
        assert obs.shape == (3,)
        for _ in range(5):
            obs, r, done, _ = env.step(np.array([0.5]))
            assert r <= 0
        assert not done

    def test_make(self):
        assert isinstance(make("CartPole-v1"), CartPoleEnv)
        assert isinstance(make("Pendulum-v1"), PendulumEnv)
        with pytest.raises(ValueError):
            make("Breakout-v4")


class TestParallelWrappers:
    def test_dummy(self):
        w = ParallelWrapperDummy(
            [lambda i: CartPoleEnv(seed=i) for _ in range(4)]
        )
        obs = w.reset()
        assert len(obs) == 4
        obs, rew, done, info = w.step([0, 1, 0, 1])
        assert len(rew) == 4
        assert w.size() == 4
        assert set(w.active()) <= set(range(4))
        w.seed(5)
        # partial step
        obs = w.reset([0, 1])
        assert len(obs) == 2
        w.close()

    def test_dummy_terminal_guard(self):
        w = ParallelWrapperDummy([lambda i: CartPoleEnv(seed=i)])
        w.reset()
        done = False
        while not done:
            _, _, dones, _ = w.step([0])
            done = dones[0]
        with pytest.raises(RuntimeError):
            w.step([0])
        w.close()

    def test_subproc(self):
        w = ParallelWrapperSubProc(
            [lambda i: CartPoleEnv(seed=i) for _ in range(3)]
        )
        try:
            obs = w.reset()
            assert len(obs) == 3 and obs[0].shape == (4,)
            obs, rew, done, info = w.step([0, 0, 1])
            assert len(obs) == 3
            assert w.size() == 3
            w.seed(11)
        finally:
            w.close()


class TestRNNBuffer:
    def test_window_sampling(self):
        b = RNNBuffer(sample_length=3, buffer_size=100)
        b.store_episode(make_episode(10))
        bs, batch = b.sample_batch(4, sample_attrs=["state", "reward"])
        state, reward = batch
        # [windows, length, ...]
        assert state["state"].shape == (4, 3, 4)
        assert reward.shape == (4, 3, 1)
        # windows are consecutive steps
        diffs = state["state"][:, 1:, 0] - state["state"][:, :-1, 0]
        assert t.allclose(diffs, t.ones_like(diffs))

    def test_time_major(self):
        b = RNNBuffer(sample_length=3, buffer_size=100, sample_dimension=0)
        b.store_episode(make_episode(10))
        bs, batch = b.sample_batch(4, sample_attrs=["state"])
        assert batch[0]["state"].shape == (3, 4, 4)

    def test_short_episode_excluded(self):
        b = RNNBuffer(sample_length=5, buffer_size=100)
        b.store_episode(make_episode(3))  # too short for any window
        bs, batch = b.sample_batch(4)
        assert bs == 0

    def test_sample_all(self):
        b = RNNBuffer(sample_length=4, buffer_size=100)
        b.store_episode(make_episode(6))
        bs, batch = b.sample_batch(-1, sample_method="all",
                                   sample_attrs=["reward"])
        # 3 valid starts x length 4
        assert batch[0].shape == (3, 4, 1)


class TestRNNPrioritizedBuffer:
    def test_windows_and_priorities(self):
        b = RNNPrioritizedBuffer(sample_length=3, buffer_size=100)
        b.store_episode(make_episode(10))
        bs, batch, index, is_weight = b.sample_batch(
            5, sample_attrs=["state", "reward"]
        )
        assert bs == 5
        assert batch[0]["state"].shape == (5, 3, 4)
        assert len(index) == 5 and len(is_weight) == 5
        # tail steps (which cannot start a window) have ~zero priority
        leaves = b.wt_tree.get_leaf_all_weights()[:10]
        assert (leaves[-2:] < 1e-9).all()
        b.update_priority(np.full(bs, 2.0), index)

    def test_windows_stay_in_episode(self):
        b = RNNPrioritizedBuffer(sample_length=4, buffer_size=100)
        b.store_episode(make_episode(6, reward_base=0))
        b.store_episode(make_episode(6, reward_base=100))
        for _ in range(10):
            bs, batch, index, w = b.sample_batch(4, sample_attrs=["reward"])
            r = batch[0]
            # each window is either all <100 or all >=100
            for win in r:
                vals = win.view(-1)
                assert (vals < 100).all() or (vals >= 100).all()
