"""IMPALA with the shared-memory rollout ring: 2 actor PROCESSES
write fixed-shape episode segments into shared memory; the learner's
update() drains ready slots, stages them through pinned memory and
runs ONE batched V-trace update — no pickled episodes cross any
process boundary, only slot indices.

    python examples/impala_ring.py
"""
import os
import socket
import sys

sys.path.insert(
    0, os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
)

import multiprocessing as mp


def _sample_episode():
    import torch as t

    return [
        {
            "state": {"state": t.zeros(1, 4)},
            "action": {"action": t.zeros(1, 1, dtype=t.long)},
            "next_state": {"state": t.zeros(1, 4)},
            "reward": 0.0,
            "terminal": i == 4,
            "action_log_prob": 0.0,
        }
        for i in range(5)
    ]


def worker(rank, world_size, port, ring, done_flag):
    import time

    import torch as t
    import torch.nn as nn

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
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
            return (
                action,
                dist.log_prob(action.view(-1)).view(-1, 1),
                dist.entropy().view(-1, 1),
            )

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
    frame = IMPALA(Actor(), Critic(), t.optim.Adam,
                   nn.MSELoss(), group, servers,
                   batch_size=16, learning_rate=5e-3,
                   entropy_weight=1e-3)
    frame.use_rollout_ring(ring)
    group.barrier()

    if rank in (1, 2):  # actors
        env = CartPoleEnv(seed=rank)
        for episode_i in range(60):
            obs = t.tensor(env.reset()).view(1, 4)
            episode, done, total = [], False, 0.0
            while not done:
                with t.no_grad():
                    action, logp, _ = frame.act({"state": obs})
                obs2, r, done, _ = env.step(int(action.item()))
                obs2 = t.tensor(obs2).view(1, 4)
                total += r
                episode.append({
                    "state": {"state": obs},
                    "action": {"action": action},
                    "next_state": {"state": obs2},
                    "reward": r / 100.0,
                    "terminal": done,
                    "action_log_prob": float(logp.item()),
                })
                obs = obs2
            frame.store_episode(episode)  # -> shared-memory ring
            if rank == 1 and episode_i % 20 == 0:
                print(f"actor episode {episode_i}: reward {total:.0f}",
                      flush=True)
        done_flag[rank - 1] = 1
    else:  # learner: consume until both actors are done AND the ring
        # has drained (actors block on acquire if nobody consumes)
        step = 0
        while True:
            act_loss, value_loss = frame.update()
            if step % 50 == 0:
                print(f"learner step {step}: act {act_loss:.4f} "
                      f"value {value_loss:.4f}", flush=True)
            step += 1
            if int(done_flag.sum()) == 2 and act_loss == 0.0 \
                    and value_loss == 0.0:
                break
    group.barrier()
    import torch.distributed as dist

    dist.barrier()
    world.stop()


def main():
    from machin_amd.parallel.rollout_ring import make_episode_ring

    ctx = mp.get_context("spawn")
    ring, _ = make_episode_ring(
        _sample_episode(), unroll=64, slots=64, ctx=ctx
    )
    import torch as t

    done_flag = t.zeros(2, dtype=t.uint8).share_memory_()
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    procs = [
        ctx.Process(target=worker, args=(r, 3, port, ring, done_flag))
        for r in range(3)
    ]
    for p in procs:
        p.start()
    for p in procs:
        p.join()


if __name__ == "__main__":
    main()
