"""Device data path wired into the algorithm classes (round-1 VERDICT
next #1). These CPU tests run the SAME adapter/wiring code the GPU
uses — DeviceSumTree falls back to torch ops on a cpu device; the
HIP-kernel variants are covered by tests/test_device_buffer_gpu.py
and test_gpu_train.py.
"""
import numpy as np
import pytest
import torch as t
import torch.nn as nn

from util_run_multi import run_multi


def _episode(n=5, state_dim=4, actions=2, extra=False):
    eps = []
    for i in range(n):
        d = {
            "state": {"state": t.rand(1, state_dim)},
            "action": {"action": t.randint(0, actions, (1, 1))},
            "next_state": {"state": t.rand(1, state_dim)},
            "reward": float(t.rand(1)),
            "terminal": i == n - 1,
        }
        if extra:
            d["action_log_prob"] = -0.7
        eps.append(d)
    return eps


class TestCpuSumTreeFallback:
    def test_matches_numpy_weight_tree(self):
        from machin_amd.frame.buffers.prioritized_buffer import WeightTree
        from machin_amd.ops.sumtree import DeviceSumTree

        n = 37
        w = t.rand(n).abs() + 0.01
        ref = WeightTree(n)
        ref.update_leaf_batch(
            w.double().numpy(), np.arange(n, dtype=np.int64)
        )
        tree = DeviceSumTree(n, "cpu")
        tree.update_all_leaves(w)
        assert abs(tree.get_weight_sum() - ref.get_weight_sum()) < 1e-4
        # prefix-sum lookups agree
        qs = t.linspace(0.0, float(ref.get_weight_sum()) - 1e-4, 50)
        got = tree.find_leaf_index(qs)
        want = ref.find_leaf_index(qs.double().numpy())
        assert (got.numpy() == want).all()
        # batched leaf update propagates
        tree.update_leaf_batch(t.tensor([5.0, 7.0]), t.tensor([3, 11]))
        ref.update_leaf_batch(
            np.array([5.0, 7.0]), np.array([3, 11], dtype=np.int64)
        )
        assert abs(tree.get_weight_sum() - ref.get_weight_sum()) < 1e-4


class TestDQNPerDeviceBuffer:
    def test_update_through_device_buffer(self):
        from machin_amd.frame.algorithms import DQNPer
        from machin_amd.frame.buffers import DeviceTransitionBuffer

        class QNet(nn.Module):
            def __init__(self):
                super().__init__()
                self.fc = nn.Sequential(
                    nn.Linear(4, 32), nn.ReLU(), nn.Linear(32, 2)
                )

            def forward(self, state):
                return self.fc(state)

        t.manual_seed(0)
        qnet = QNet()
        qnet_t = QNet()
        buf = DeviceTransitionBuffer(1000, "cpu", prioritized=True)
        frame = DQNPer(
            qnet, qnet_t, t.optim.Adam, nn.MSELoss(),
            replay_buffer=buf, batch_size=16,
        )
        for _ in range(4):
            frame.store_episode(_episode())
        assert frame.replay_buffer.size() == 20
        losses = [frame.update() for _ in range(5)]
        assert all(isinstance(v, float) and np.isfinite(v) for v in losses)

    def test_replay_device_cpu_keeps_reference_buffer(self):
        from machin_amd.frame.algorithms import DQNPer
        from machin_amd.frame.buffers.prioritized_buffer import (
            PrioritizedBuffer,
        )

        class QNet(nn.Module):
            def __init__(self):
                super().__init__()
                self.fc = nn.Linear(4, 2)

            def forward(self, state):
                return self.fc(state)

        frame = DQNPer(
            QNet(), QNet(), t.optim.Adam, nn.MSELoss(),
            replay_device="cpu",
        )
        assert isinstance(frame.replay_buffer, PrioritizedBuffer)


class TestRainbowDeviceBuffer:
    def test_update_through_device_buffer(self):
        from machin_amd.auto.model_zoo import DistQNet
        from machin_amd.frame.algorithms import RAINBOW
        from machin_amd.frame.buffers import DeviceTransitionBuffer

        t.manual_seed(0)
        buf = DeviceTransitionBuffer(1000, "cpu", prioritized=True)
        frame = RAINBOW(
            DistQNet(state_dim=4, action_num=2),
            DistQNet(state_dim=4, action_num=2),
            t.optim.Adam, value_min=-10.0, value_max=10.0,
            replay_buffer=buf, batch_size=16,
        )
        for _ in range(4):
            frame.store_episode(_episode())
        losses = [frame.update() for _ in range(3)]
        assert all(np.isfinite(v) for v in losses)


class TestApexDeviceBuffer:
    def test_learner_shard_over_control_plane(self):
        """3 ranks: rank 0 learner hosts the shard, ranks 1-2 push
        flattened episodes over the control plane; sampling and
        priority updates never leave the learner."""
        def fn(rank, world):
            from machin_amd.frame.algorithms import DQNApex
            from machin_amd.frame.buffers.device_buffer_d import (
                DeviceDistributedPrioritizedBuffer,
            )
            from machin_amd.frame.helpers.servers import (
                model_server_helper,
            )

            class QNet(nn.Module):
                def __init__(self):
                    super().__init__()
                    self.fc = nn.Linear(4, 2)

                def forward(self, state):
                    return self.fc(state)

            servers = model_server_helper(model_num=1)
            group = world.create_rpc_group("apex", ["0", "1", "2"])
            apex = DQNApex(
                QNet(), QNet(), t.optim.Adam, nn.MSELoss(),
                group, servers, batch_size=8, replay_size=128,
                replay_device="cpu", replay_learners=["0"],
            )
            assert isinstance(
                apex.replay_buffer, DeviceDistributedPrioritizedBuffer
            )
            group.barrier()
            if rank in (1, 2):
                t.manual_seed(rank)
                for _ in range(3):
                    apex.store_episode(_episode())
            group.barrier()
            loss = None
            if rank == 0:
                assert apex.replay_buffer.size() == 30
                loss = apex.update()
                assert isinstance(loss, float)
            group.barrier()
            return loss

        results = run_multi(fn, world_size=3, timeout=240)
        assert results[0] is not None

    def test_two_learner_shards_round_robin(self):
        """2 learner shards: the episode stream is split round-robin,
        each learner samples only its local shard."""
        def fn(rank, world):
            from machin_amd.frame.buffers.device_buffer_d import (
                DeviceDistributedPrioritizedBuffer,
            )

            group = world.create_rpc_group("shards", ["0", "1", "2"])
            buf = DeviceDistributedPrioritizedBuffer(
                "b", group, 128, learners=["0", "1"], device="cpu"
            )
            group.barrier()
            if rank == 2:
                for _ in range(4):
                    buf.store_episode(_episode())
            group.barrier()
            got = None
            if rank in (0, 1):
                assert buf.size() == 10  # 2 episodes x 5 each
                assert buf.all_size() == 20
                bs, batch, idx, w = buf.sample_batch(
                    6,
                    sample_attrs=["state", "action", "reward",
                                  "next_state", "terminal", "*"],
                )
                assert bs == 6
                assert batch[0]["state"].shape == (6, 4)
                buf.update_priority(t.rand(6), idx)
                got = True
            group.barrier()
            return got

        results = run_multi(fn, world_size=3, timeout=240)
        assert results[0] and results[1]


class TestDeviceBufferWraparound:
    def test_ring_eviction_and_sampling_after_wrap(self):
        from machin_amd.frame.buffers import DeviceTransitionBuffer

        buf = DeviceTransitionBuffer(12, "cpu", prioritized=True)
        for k in range(5):  # 25 transitions through a 12-slot ring
            ep = [
                {
                    "state": {"state": t.full((1, 4), float(k * 5 + i))},
                    "action": {"action": t.randint(0, 2, (1, 1))},
                    "next_state": {"state": t.rand(1, 4)},
                    "reward": float(k * 5 + i),
                    "terminal": i == 4,
                }
                for i in range(5)
            ]
            buf.store_episode(ep)
        assert buf.size() == 12  # clamped to capacity
        bs, batch, idx, w = buf.sample_batch(
            64,
            sample_attrs=["state", "action", "reward", "next_state",
                          "terminal", "*"],
        )
        # only the newest 12 rewards (13..24) can be sampled
        rewards = batch[2].view(-1)
        assert rewards.min() >= 13.0 and rewards.max() <= 24.0
        # ring positions stay in range after wrap
        assert int(idx.max()) < 16  # capacity rounded to tree size
        buf.update_priority(t.rand(64), idx)

    def test_capacity_16_exact_positions(self):
        from machin_amd.frame.buffers.device_buffer import (
            DeviceReplayBuffer,
        )

        buf = DeviceReplayBuffer(
            8, {"x": ((2,), t.float32)}, "cpu"
        )
        pos1 = buf.store_batch({"x": t.arange(12.0).view(6, 2)})
        assert pos1.tolist() == [0, 1, 2, 3, 4, 5]
        pos2 = buf.store_batch({"x": t.arange(8.0).view(4, 2)})
        assert pos2.tolist() == [6, 7, 0, 1]  # wrapped
        assert buf.size() == 8
        # wrapped rows hold the new data
        assert buf.data["x"][0].tolist() == [4.0, 5.0]
