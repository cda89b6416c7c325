"""APEX-DQN learner benchmark on the HBM-resident prioritized replay
(BASELINE config #4 shape, single GPU): Nature-CNN double-DQN learner,
PER sampling + priority updates fully on device, fused polyak target
update, bf16 forward/backward.

Run on a GPU box: python tools/bench_apex.py [--batch 512] [--steps 50]
Writes gpurun_out/apex_bench.json.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(
    0, os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
)

import torch as t
import torch.nn as nn


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--batch", type=int, default=512)
    parser.add_argument("--steps", type=int, default=100)
    parser.add_argument("--warmup", type=int, default=20)
    parser.add_argument("--replay", type=int, default=200000)
    args = parser.parse_args()

    import machin_amd.ops as ops
    from machin_amd.frame.buffers.device_buffer import (
        DevicePrioritizedBuffer,
    )
    from machin_amd.model.nets.nature_cnn import NatureCNN

    assert ops.available()
    dev = t.device("cuda:0")
    t.backends.cudnn.benchmark = True

    class QNet(nn.Module):
        def __init__(self, actions=6):
            super().__init__()
            self.torso = NatureCNN(4)
            self.head = nn.Linear(512, actions)

        def forward(self, frames):
            return self.head(self.torso(frames))

    qnet = QNet().to(dev).to(memory_format=t.channels_last)
    qnet_t = QNet().to(dev).to(memory_format=t.channels_last)
    qnet_t.load_state_dict(qnet.state_dict())
    optim = t.optim.Adam(qnet.parameters(), lr=1e-4)

    spec = {
        "state": ((4, 84, 84), t.uint8),
        "action": ((1,), t.long),
        "reward": ((), t.float32),
        "next_state": ((4, 84, 84), t.uint8),
        "terminal": ((), t.float32),
    }
    buf = DevicePrioritizedBuffer(args.replay, spec, dev)
    # fill with synthetic transitions (actor-feed stand-in)
    n = 8192
    for _ in range(6):
        buf.store_batch(
            {
                "state": t.randint(0, 256, (n, 4, 84, 84), dtype=t.uint8,
                                   device=dev),
                "action": t.randint(0, 6, (n, 1), device=dev),
                "reward": t.rand(n, device=dev),
                "next_state": t.randint(0, 256, (n, 4, 84, 84),
                                        dtype=t.uint8, device=dev),
                "terminal": (t.rand(n, device=dev) > 0.98).float(),
            }
        )

    tgt = [p.data for p in qnet_t.parameters()]
    src = [p.data for p in qnet.parameters()]

    def step():
        batch, idx, is_w = buf.sample_batch(args.batch)
        s = ops.dequant_u8(
            batch["state"].permute(0, 2, 3, 1)
        ).view(-1, 84, 84, 4).permute(0, 3, 1, 2)
        sn = ops.dequant_u8(
            batch["next_state"].permute(0, 2, 3, 1)
        ).view(-1, 84, 84, 4).permute(0, 3, 1, 2)
        # no_grad OUTSIDE autocast: autocast's weight-cast cache must
        # not leak grad-less casts into the training forward
        with t.no_grad():
            with t.autocast(device_type="cuda", dtype=t.bfloat16):
                online_next = qnet(sn).float()
                best = online_next.argmax(dim=1, keepdim=True)
                q_next = qnet_t(sn).float().gather(1, best)
            y = batch["reward"].view(-1, 1) + 0.99 * (
                1.0 - batch["terminal"].view(-1, 1)
            ) * q_next
        with t.autocast(device_type="cuda", dtype=t.bfloat16):
            q = qnet(s).float().gather(1, batch["action"])
        td = q - y
        loss = (td.pow(2).view(-1) * is_w).mean()
        optim.zero_grad(set_to_none=False)
        loss.backward()
        optim.step()
        buf.update_priority(td.detach().abs().view(-1), idx)
        ops.polyak_update_(tgt, src, 0.005)
        return loss.detach()

    for _ in range(args.warmup):
        step()
    t.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    t.cuda.synchronize()
    dt = time.perf_counter() - t0
    out = {
        "metric": "apex_dqn_learner_samples_per_sec",
        "value": args.batch * args.steps / dt,
        "ms_per_step": dt / args.steps * 1000,
        "batch": args.batch,
        "replay_size": args.replay,
        "dtype": "bf16",
        "device_resident_replay": True,
    }
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/apex_bench.json", "w") as f:
        json.dump(out, f, indent=2)
    print(json.dumps(out))


if __name__ == "__main__":
    main()
