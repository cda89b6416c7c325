"""APEX-DQN on CartPole: 2 sampler processes + 1 learner, distributed
prioritized replay + model server (reference analog:
examples/tutorials/unleash_distributed_power/dqn_apex.py)."""
import os
import sys

sys.path.insert(
    0, os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
)

import multiprocessing as mp
import socket


def worker(rank, world_size, port):
    import time

    import torch as t
    import torch.nn as nn

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    sys.path.insert(
        0, os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    )
    from machin_amd.env.envs import CartPoleEnv
    from machin_amd.frame.algorithms import DQNApex
    from machin_amd.frame.helpers.servers import model_server_helper
    from machin_amd.parallel.distributed import World

    class QNet(nn.Module):
        def __init__(self):
            super().__init__()
            self.fc1 = nn.Linear(4, 64)
            self.fc2 = nn.Linear(64, 64)
            self.fc3 = nn.Linear(64, 2)

        def forward(self, state):
            a = t.relu(self.fc1(state))
            return self.fc3(t.relu(self.fc2(a)))

    world = World(world_size=world_size, rank=rank, name=str(rank),
                  dist_backend="gloo")
    servers = model_server_helper(model_num=1)
    group = world.create_rpc_group("apex", ["0", "1", "2"])
    apex = DQNApex(
        QNet(), QNet(), t.optim.Adam, nn.MSELoss(reduction="sum"),
        group, servers, batch_size=64, learning_rate=1e-3,
        update_rate=0.005, replay_size=10000,
    )
    group.barrier()

    if rank == 2:  # learner
        apex.set_sync(False)
        while not group.is_paired("solved"):
            if apex.replay_buffer.all_size() > 500:
                apex.update()
            else:
                time.sleep(0.05)
        print("[learner] done")
    else:  # samplers
        apex.set_sync(False)
        env = CartPoleEnv(seed=rank)
        smoothed, episode = 0.0, 0
        while not group.is_paired("solved"):
            episode += 1
            apex.epsilon = max(0.08, 0.995 ** episode)
            obs = t.tensor(env.reset()).view(1, 4)
            total, transitions, done = 0.0, [], False
            while not done:
                with t.no_grad():
                    action = apex.act_discrete_with_noise(
                        {"state": obs}, decay_epsilon=False
                    )
                o, r, done, _ = env.step(int(action.item()))
                o = t.tensor(o).view(1, 4)
                total += r
                transitions.append(
                    {"state": {"state": obs},
                     "action": {"action": action},
                     "next_state": {"state": o}, "reward": r,
                     "terminal": done
                     and env.steps < env.max_episode_steps}
                )
                obs = o
            apex.store_episode(transitions)
            apex.manual_sync()
            smoothed = smoothed * 0.9 + total * 0.1
            if rank == 0 and episode % 50 == 0:
                print(f"[sampler 0] ep {episode}: smoothed {smoothed:.1f}")
            if smoothed > 195:
                print(f"[sampler {rank}] solved at episode {episode}")
                try:
                    group.pair("solved", True)
                except RuntimeError:
                    pass
                break
    group.barrier()


def main():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=worker, args=(r, 3, port))
             for r in range(3)]
    for p in procs:
        p.start()
    for p in procs:
        p.join()


if __name__ == "__main__":
    main()
