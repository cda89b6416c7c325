"""End-to-end learning ON the GPU: DQN (model + replay in HBM) solves
CartPole — exercises act/store/update/soft-update with the HIP polyak
and device-resident batches under a real training loop."""
import time

import pytest
import torch as t
import torch.nn as nn

pytestmark = pytest.mark.gpu

if not t.cuda.is_available():
    pytest.skip("no GPU", allow_module_level=True)


class TestGPUTraining:
    def test_dqn_learns_on_gpu(self):
        import sys

        sys.path.insert(0, "tests")
        from util_models import QNet

        from machin_amd.env.envs.classic_control import CartPoleEnv
        from machin_amd.frame.algorithms import DQN

        dev = "cuda:0"
        t.manual_seed(0)
        dqn = DQN(
            QNet().to(dev), QNet().to(dev), t.optim.Adam, nn.MSELoss(),
            batch_size=64, learning_rate=1e-3, epsilon_decay=0.995,
            update_rate=0.01, replay_device=dev, replay_size=10000,
        )
        env = CartPoleEnv(seed=0)
        smoothed, wins = 0.0, 0
        deadline = time.monotonic() + 240
        episodes = 0
        while time.monotonic() < deadline and episodes < 600:
            episodes += 1
            obs = t.tensor(env.reset(), dtype=t.float32,
                           device=dev).view(1, 4)
            total, transitions, done = 0.0, [], False
            while not done:
                with t.no_grad():
                    action = dqn.act_discrete_with_noise({"state": obs})
                o, r, done, _ = env.step(int(action.item()))
                o = t.tensor(o, dtype=t.float32, device=dev).view(1, 4)
                total += r
                transitions.append(
                    {
                        "state": {"state": obs},
                        "action": {"action": action.to(dev)},
                        "next_state": {"state": o},
                        "reward": r,
                        "terminal": done
                        and env.steps < env.max_episode_steps,
                    }
                )
                obs = o
            dqn.store_episode(transitions)
            if dqn.replay_buffer.size() > 500:
                for _ in range(min(len(transitions), 30)):
                    dqn.update()
            smoothed = smoothed * 0.9 + total * 0.1
            if smoothed > 150:
                wins += 1
                if wins >= 3:
                    return
            else:
                wins = 0
        pytest.fail(
            f"DQN on GPU did not reach 150 (smoothed={smoothed:.1f}, "
            f"episodes={episodes})"
        )
