"""A3C on CartPole with 3 worker processes and gradient servers
(reference analog: examples/framework_examples/a3c.py)."""
import os
import sys

sys.path.insert(
    0, os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
)

import multiprocessing as mp
import os
import socket


def worker(rank, world_size, port):
    import torch as t
    import torch.nn as nn

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    from machin_amd.env.envs import CartPoleEnv
    from machin_amd.frame.algorithms import A3C
    from machin_amd.frame.helpers.servers import grad_server_helper
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
    servers = grad_server_helper(
        [Actor, Critic], learning_rate=5e-3, reduce_batch_size=3,
        reduce_method="mean",
    )
    a3c = A3C(Actor(), Critic(), nn.MSELoss(reduction="sum"), servers,
              entropy_weight=0.01, gae_lambda=0.97)
    group = world.groups["grad_server_group"]
    group.barrier()

    env = CartPoleEnv(seed=rank)
    smoothed = 0.0
    for episode in range(3000):
        if group.is_paired("solved"):
            break
        obs = t.tensor(env.reset()).view(1, 4)
        total, transitions, done = 0.0, [], False
        while not done:
            with t.no_grad():
                action = a3c.act({"state": obs})[0]
            o, r, done, _ = env.step(int(action.item()))
            o = t.tensor(o).view(1, 4)
            total += r
            transitions.append(
                {"state": {"state": obs}, "action": {"action": action},
                 "next_state": {"state": o}, "reward": r,
                 "terminal": done and env.steps < env.max_episode_steps}
            )
            obs = o
        a3c.store_episode(transitions)
        a3c.update()
        smoothed = smoothed * 0.9 + total * 0.1
        if rank == 0 and episode % 50 == 0:
            print(f"[worker 0] episode {episode}: smoothed {smoothed:.1f}")
        if smoothed > 195:
            print(f"[worker {rank}] solved at episode {episode}")
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
