"""Training launchers.

Parity target: reference ``machin/auto/launcher.py`` (Launcher :13,
DistributedLauncher :105) + ``machin/auto/pl_plugin.py``. The
reference drives training through pytorch-lightning with patched DDP
plugins; this image has no lightning, so the Launcher is a
self-contained fit loop with the same responsibilities: episodes from
the dataset -> store -> update -> log -> checkpoint on the monitored
metric -> early stop. The distributed launcher spawns one process per
world member, each constructing a World (gloo on CPU, RCCL on GPU)
and running the same loop; sampler ranks of APEX/IMPALA get no-op
updates from init_from_config.
"""
import os
from typing import Callable, Optional

import torch as t

from ..utils.conf import Config
from ..utils.logging import default_logger
from ..utils.save_env import SaveEnv


class Launcher:
    """Single-process fit loop."""

    def __init__(self, config, dataset_factory: Callable,
                 frame=None, save_env: Optional[SaveEnv] = None):
        from .config import init_algorithm_from_config

        self.config = config
        data = config.data if isinstance(config, Config) else dict(config)
        self.data = data
        self.frame = frame or init_algorithm_from_config(config)
        self.save_env = save_env or SaveEnv(data.get("root_dir", "trial"))
        self.dataset = dataset_factory(self.frame, data)
        self.episode_per_epoch = data.get("episode_per_epoch", 10)
        self.max_episodes = data.get("max_episodes", 10000)
        self.patience = data.get("early_stopping_patience", 10)
        self.monitor = getattr(
            self.dataset, "early_stopping_monitor", "total_reward"
        )

    def init_frame(self):
        return self.frame

    def training_step(self, result):
        """Store the episode and run updates (one per transition,
        after a warmup)."""
        for episode in result.observations:
            self.frame.store_episode(episode)
        buffer = getattr(self.frame, "replay_buffer", None)
        if buffer is None or not hasattr(buffer, "size"):
            self.frame.update()
            return
        is_online = type(self.frame).__name__ in (
            "A2C", "PPO", "TRPO", "A3C", "IMPALA",
        )
        if is_online:
            self.frame.update()
        elif buffer.size() > self.data.get("update_warmup_size", 500):
            n = min(
                sum(len(ep) for ep in result.observations),
                self.data.get("max_updates_per_episode", 50),
            )
            for _ in range(n):
                self.frame.update()

    def fit(self) -> float:
        smoothed = None
        best = None
        stale_epochs = 0
        episodes = 0
        version = 0
        while episodes < self.max_episodes:
            epoch_metric = None
            for _ in range(self.episode_per_epoch):
                result = next(self.dataset)
                episodes += len(result.observations)
                self.training_step(result)
                for log in result.logs:
                    if self.monitor in log:
                        v = log[self.monitor]
                        smoothed = (
                            v if smoothed is None
                            else smoothed * 0.9 + v * 0.1
                        )
                        epoch_metric = smoothed
            if epoch_metric is not None:
                default_logger.info(
                    f"episodes={episodes} {self.monitor}={epoch_metric:.2f}"
                )
                if best is None or epoch_metric > best:
                    best = epoch_metric
                    stale_epochs = 0
                    version += 1
                    self.frame.save(
                        self.save_env.get_trial_model_dir(), version=version
                    )
                else:
                    stale_epochs += 1
                    if stale_epochs >= self.patience:
                        default_logger.info("Early stopping.")
                        break
        return best if best is not None else float("-inf")


class DistributedLauncher(Launcher):
    """Per-process launcher used inside launch_distributed."""


def _dist_worker(rank: int, world_size: int, port: int, config_data: dict):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    t.set_num_threads(2)
    import torch.distributed as dist

    from ..parallel.distributed.world import World
    from .envs.classic_control import make_dataset

    world = World(world_size=world_size, rank=rank, name=str(rank),
                  dist_backend="nccl" if t.cuda.is_available() else "gloo",
                  dist_timeout=1800.0)
    config = Config(**config_data)
    launcher = DistributedLauncher(config, dataset_factory=make_dataset)
    launcher.fit()
    # keep this process's control-plane services alive until every
    # member finishes (A3C/APEX peers keep calling them), then tear
    # the world down EXPLICITLY: destroying the gloo process group
    # from interpreter-exit GC intermittently aborts the child
    # ("terminate called without an active exception")
    dist.barrier()
    try:
        world.stop()
    except Exception:  # noqa: BLE001 - teardown best-effort
        pass
    # exit WITHOUT interpreter teardown: lingering comm threads
    # (gloo/pg destructors running from GC at exit) intermittently
    # abort the child after training already succeeded
    os._exit(0)


def launch_distributed(config):
    """Spawn ``world_size`` processes (one per GPU when available) and
    run the fit loop in each."""
    import torch.multiprocessing as mp

    data = config.data if isinstance(config, Config) else dict(config)
    world_size = data.get("world_size", 3)
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    mp.spawn(
        _dist_worker,
        args=(world_size, port, dict(data)),
        nprocs=world_size,
        join=True,
    )
    return None
