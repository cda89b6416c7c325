"""DQN on the built-in PixelCatch pixel env with device-resident
replay: `replay_device="cuda:0"` keeps the whole replay (uint8 frame
stacks!) in HBM flat rings with DeviceSumTree-priority-free uniform
sampling — zero python-per-transition work on the update path.

Runs on CPU too (the device buffer falls back to torch ops), just
slower; on an MI355X pass --device cuda:0.

    python examples/dqn_pixelcatch_device_replay.py [--device cuda:0]
"""
import argparse
import os
import sys

sys.path.insert(
    0, os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
)

import torch as t
import torch.nn as nn

from machin_amd.env.envs import PixelCatchEnv
from machin_amd.frame.algorithms import DQN
from machin_amd.model.nets.nature_cnn import NatureCNN


class QNet(nn.Module):
    def __init__(self):
        super().__init__()
        self.cnn = NatureCNN(4, feature_dim=256)
        self.head = nn.Linear(256, 3)

    def forward(self, state):
        return self.head(self.cnn(state.float() / 255.0))


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--device", default="cpu")
    parser.add_argument("--episodes", type=int, default=30)
    args = parser.parse_args()
    dev = t.device(args.device)

    t.manual_seed(0)
    dqn = DQN(
        QNet().to(dev), QNet().to(dev), t.optim.Adam, nn.MSELoss(),
        batch_size=32, replay_size=20000,
        # the interesting line: replay lives on the model's device
        replay_device=args.device,
        epsilon_decay=0.999,
    )
    print("replay buffer:", type(dqn.replay_buffer).__name__)

    env = PixelCatchEnv(seed=0, balls=3)
    for ep in range(args.episodes):
        obs = env.reset()
        episode, total, done = [], 0.0, False
        while not done:
            st = t.from_numpy(obs).unsqueeze(0).to(dev)
            act = dqn.act_discrete_with_noise({"state": st})
            obs2, r, done, _ = env.step(int(act.item()))
            total += r
            episode.append({
                "state": {"state": st},
                "action": {"action": act},
                "next_state": {
                    "state": t.from_numpy(obs2).unsqueeze(0).to(dev)
                },
                "reward": r,
                "terminal": done,
            })
            obs = obs2
        dqn.store_episode(episode)
        losses = [dqn.update() for _ in range(4)]
        print(f"ep {ep:3d} reward {total:+.0f} "
              f"loss {sum(losses) / len(losses):.4f} "
              f"eps {dqn.epsilon:.3f}")


if __name__ == "__main__":
    main()
