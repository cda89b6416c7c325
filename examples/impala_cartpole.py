"""IMPALA on CartPole: 2 actor processes + 1 V-trace learner
(reference analog: test/frame/algorithms/test_impala.py flow)."""
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
    from machin_amd.frame.algorithms import IMPALA
    from machin_amd.frame.helpers.servers import model_server_helper
    from machin_amd.parallel.distributed import World

    class Actor(nn.Module):
        def __init__(self):
            super().__init__()
            self.fc1 = nn.Linear(4, 32)
            self.fc2 = nn.Linear(32, 2)

        def forward(self, state, action=None):
            logits = self.fc2(t.relu(self.fc1(state)))
            dist = t.distributions.Categorical(logits=logits)
            if action is None:
                action = dist.sample().view(-1, 1)
            return (action, dist.log_prob(action.view(-1)).view(-1, 1),
                    dist.entropy().view(-1, 1))

    class Critic(nn.Module):
        def __init__(self):
            super().__init__()
            self.fc1 = nn.Linear(4, 32)
            self.fc2 = nn.Linear(32, 1)

        def forward(self, state):
            return self.fc2(t.relu(self.fc1(state)))

    world = World(world_size=world_size, rank=rank, name=str(rank),
                  dist_backend="gloo")
    servers = model_server_helper(model_num=1)
    group = world.create_rpc_group("impala", ["0", "1", "2"])
    impala = IMPALA(
        Actor(), Critic(), t.optim.Adam, nn.MSELoss(reduction="sum"),
        group, servers, batch_size=4, learning_rate=5e-3,
        entropy_weight=0.01,
    )
    group.barrier()

    if rank == 2:  # learner
        impala.set_sync(False)
        while not group.is_paired("solved"):
            if impala.replay_buffer.all_size() >= 2:
                impala.update()
            else:
                time.sleep(0.02)
        print("[learner] done")
    else:  # actors
        env = CartPoleEnv(seed=rank)
        smoothed, episode = 0.0, 0
        while not group.is_paired("solved"):
            episode += 1
            obs = t.tensor(env.reset()).view(1, 4)
            total, transitions, done = 0.0, [], False
            while not done:
                with t.no_grad():
                    action, logp, _ = impala.act({"state": obs})
                o, r, done, _ = env.step(int(action.item()))
                o = t.tensor(o).view(1, 4)
                total += r
                transitions.append(
                    {"state": {"state": obs},
                     "action": {"action": action},
                     "next_state": {"state": o}, "reward": r,
                     "action_log_prob": float(logp.item()),
                     "terminal": done
                     and env.steps < env.max_episode_steps}
                )
                obs = o
            impala.store_episode(transitions)
            smoothed = smoothed * 0.9 + total * 0.1
            if rank == 0 and episode % 50 == 0:
                print(f"[actor 0] ep {episode}: smoothed {smoothed:.1f}")
            if smoothed > 195:
                print(f"[actor {rank}] solved at episode {episode}")
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
