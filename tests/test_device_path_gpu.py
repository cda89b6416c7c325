"""GPU tests: the HBM device replay is THE path when
replay_device="cuda:0" (round-1 VERDICT next #1). Exercises the
DeviceSumTree HIP kernels + flat-ring store/sample through the
algorithm classes, not just through tools/ benches."""
import numpy as np
import pytest
import torch as t
import torch.nn as nn

pytestmark = pytest.mark.gpu


class QNet(nn.Module):
    def __init__(self, state_dim=4, action_num=2):
        super().__init__()
        self.fc = nn.Sequential(
            nn.Linear(state_dim, 32), nn.ReLU(),
            nn.Linear(32, action_num),
        )

    def forward(self, state):
        return self.fc(state)


def _episode(n=5):
    return [
        {
            "state": {"state": t.rand(1, 4)},
            "action": {"action": t.randint(0, 2, (1, 1))},
            "next_state": {"state": t.rand(1, 4)},
            "reward": float(t.rand(1)),
            "terminal": i == n - 1,
        }
        for i in range(n)
    ]


class TestDQNPerDeviceReplay:
    def test_cuda_replay_device_uses_hbm_rings(self):
        from machin_amd.frame.algorithms import DQNPer
        from machin_amd.frame.buffers.device_buffer import (
            DeviceTransitionBuffer,
        )

        frame = DQNPer(
            QNet().to("cuda:0"), QNet().to("cuda:0"),
            t.optim.Adam, nn.MSELoss(),
            replay_device="cuda:0", replay_size=4096, batch_size=32,
        )
        assert isinstance(frame.replay_buffer, DeviceTransitionBuffer)
        for _ in range(10):
            frame.store_episode(_episode())
        # the ring and the sum tree live on device
        inner = frame.replay_buffer._inner
        assert inner.data["state/state"].is_cuda
        assert inner.wt_tree.weights.is_cuda
        losses = [frame.update() for _ in range(10)]
        assert all(np.isfinite(v) for v in losses)

    def test_priorities_track_td_error(self):
        """After updates, sampled indices must skew toward
        higher-priority entries (PER semantics on device)."""
        from machin_amd.frame.buffers.device_buffer import (
            DeviceTransitionBuffer,
        )

        buf = DeviceTransitionBuffer(256, "cuda:0", prioritized=True)
        for _ in range(8):
            buf.store_episode(_episode())
        n = buf.size()
        # set one entry's priority overwhelmingly high
        buf.update_priority(
            t.full((1,), 1000.0, device="cuda:0"),
            t.tensor([5], device="cuda:0"),
        )
        _, idx, _ = buf._inner.sample_batch(512)
        frac = (idx == 5).float().mean().item()
        assert frac > 0.5, f"high-priority entry sampled {frac:.2%}"


class TestRainbowDeviceReplay:
    def test_rainbow_trains_from_hbm(self):
        from machin_amd.auto.model_zoo import DistQNet
        from machin_amd.frame.algorithms import RAINBOW
        from machin_amd.frame.buffers.device_buffer import (
            DeviceTransitionBuffer,
        )

        frame = RAINBOW(
            DistQNet(state_dim=4, action_num=2).to("cuda:0"),
            DistQNet(state_dim=4, action_num=2).to("cuda:0"),
            t.optim.Adam, value_min=-10.0, value_max=10.0,
            replay_device="cuda:0", replay_size=4096, batch_size=32,
        )
        assert isinstance(frame.replay_buffer, DeviceTransitionBuffer)
        for _ in range(6):
            frame.store_episode(_episode())
        losses = [frame.update() for _ in range(5)]
        assert all(np.isfinite(v) for v in losses)


class TestApexDeviceReplayGPU:
    def test_learner_hbm_shard(self):
        """3 processes on one GPU box: rank 0's shard is in HBM, CPU
        sampler ranks push episodes over the control plane."""
        import sys

        sys.path.insert(0, "tests")
        from util_run_multi import run_multi

        def fn(rank, world):
            from machin_amd.frame.algorithms import DQNApex
            from machin_amd.frame.buffers.device_buffer_d import (
                DeviceDistributedPrioritizedBuffer,
            )
            from machin_amd.frame.helpers.servers import (
                model_server_helper,
            )

            import torch.nn as nn

            class Q(nn.Module):
                def __init__(self):
                    super().__init__()
                    self.fc = nn.Linear(4, 2)

                def forward(self, state):
                    return self.fc(state)

            servers = model_server_helper(model_num=1)
            group = world.create_rpc_group("apex", ["0", "1", "2"])
            dev = "cuda:0" if rank == 0 else "cpu"
            q = Q().to(dev)
            qt = Q().to(dev)
            apex = DQNApex(
                q, qt, t.optim.Adam, nn.MSELoss(), group, servers,
                batch_size=16, replay_size=4096,
                replay_device="cuda:0", replay_learners=["0"],
            )
            assert isinstance(
                apex.replay_buffer, DeviceDistributedPrioritizedBuffer
            )
            group.barrier()
            if rank in (1, 2):
                t.manual_seed(rank)
                for _ in range(5):
                    ep = [
                        {
                            "state": {"state": t.rand(1, 4)},
                            "action": {
                                "action": t.randint(0, 2, (1, 1))
                            },
                            "next_state": {"state": t.rand(1, 4)},
                            "reward": float(t.rand(1)),
                            "terminal": i == 4,
                        }
                        for i in range(5)
                    ]
                    apex.store_episode(ep)
            group.barrier()
            loss = None
            if rank == 0:
                assert apex.replay_buffer.size() == 50
                shard = apex.replay_buffer.local._inner
                assert shard.data["state/state"].is_cuda
                assert shard.wt_tree.weights.is_cuda
                for _ in range(3):
                    loss = apex.update()
            group.barrier()
            return loss

        results = run_multi(fn, world_size=3, timeout=300)
        assert results[0] is not None and np.isfinite(results[0])


class TestIMPALARingGPU:
    def test_ring_update_on_gpu_learner(self):
        """Ring mode end-to-end on a CUDA learner: codec write ->
        pinned staging -> one async H2D per attribute -> batched
        V-trace (HIP kernel) -> optimizer step."""
        import sys

        sys.path.insert(0, "tests")
        from util_run_multi import run_multi

        def fn(rank, world):
            import torch.nn as nn

            from machin_amd.frame.algorithms.impala import IMPALA
            from machin_amd.frame.helpers.servers import (
                model_server_helper,
            )
            from machin_amd.parallel.rollout_ring import (
                make_episode_ring,
            )

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
            group = world.create_rpc_group("impala", ["0"])
            frame = IMPALA(
                Actor().to("cuda:0"), Critic().to("cuda:0"),
                t.optim.Adam, nn.MSELoss(), group, servers,
                batch_size=8,
            )
            sample = [
                {
                    "state": {"state": t.rand(1, 4)},
                    "action": {"action": t.randint(0, 2, (1, 1))},
                    "next_state": {"state": t.rand(1, 4)},
                    "reward": 0.5,
                    "terminal": i == 4,
                    "action_log_prob": -0.5,
                }
                for i in range(5)
            ]
            ring, _ = make_episode_ring(sample, unroll=6, slots=16)
            frame.use_rollout_ring(ring)
            for e in range(6):
                t.manual_seed(e)
                ep = [
                    {
                        "state": {"state": t.rand(1, 4)},
                        "action": {"action": t.randint(0, 2, (1, 1))},
                        "next_state": {"state": t.rand(1, 4)},
                        "reward": float(t.rand(1)),
                        "terminal": i == 4,
                        "action_log_prob": float(-t.rand(1)),
                    }
                    for i in range(5)
                ]
                frame.store_episode(ep)
            al, vl = frame.update()
            assert np.isfinite(al) and np.isfinite(vl)
            # pinned staging was engaged (learner on cuda)
            assert frame._ring_pinned is not None
            return True

        assert all(run_multi(fn, world_size=1, timeout=300))


class TestTD3DeviceReplay:
    def test_td3_trains_from_hbm(self):
        """TD3 (DDPG family) with replay_device=cuda: twin critics,
        target smoothing, HBM ring replay."""
        from machin_amd.auto.model_zoo import (
            DeterministicActor,
            QCritic,
        )
        from machin_amd.frame.algorithms import TD3
        from machin_amd.frame.buffers.device_buffer import (
            DeviceTransitionBuffer,
        )

        dev = "cuda:0"
        frame = TD3(
            DeterministicActor(3, 1).to(dev),
            DeterministicActor(3, 1).to(dev),
            QCritic(3, 1).to(dev), QCritic(3, 1).to(dev),
            QCritic(3, 1).to(dev), QCritic(3, 1).to(dev),
            t.optim.Adam, nn.MSELoss(),
            replay_device=dev, batch_size=16,
        )
        assert isinstance(frame.replay_buffer, DeviceTransitionBuffer)
        for _ in range(4):
            ep = [
                {
                    "state": {"state": t.rand(1, 3)},
                    "action": {"action": t.rand(1, 1) * 2 - 1},
                    "next_state": {"state": t.rand(1, 3)},
                    "reward": float(t.rand(1)),
                    "terminal": i == 4,
                }
                for i in range(5)
            ]
            frame.store_episode(ep)
        for _ in range(4):
            out = frame.update()
            assert all(v == v for v in out)
