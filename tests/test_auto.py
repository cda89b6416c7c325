"""Auto layer tests: config generation, validation, CLI, launch."""
import json
import subprocess
import sys

import pytest

from machin_amd.auto.config import (
    generate_algorithm_config,
    generate_env_config,
    generate_training_config,
    get_available_algorithms,
    get_available_environments,
    init_algorithm_from_config,
    is_algorithm_distributed,
    launch,
    validate_config,
)
from machin_amd.utils.conf import Config


def full_config(algo="DQN", env="CartPole-v1", tmpdir="/tmp/trial_auto"):
    config = generate_env_config(env)
    config = generate_algorithm_config(algo, config)
    config = generate_training_config(root_dir=tmpdir, config=config)
    return config


class TestConfigGeneration:
    def test_lists(self):
        assert "DQN" in get_available_algorithms()
        assert "CartPole-v1" in get_available_environments()

    def test_generate_and_validate(self, tmp_path):
        config = full_config(tmpdir=str(tmp_path))
        validate_config(config)
        assert config["frame"] == "DQN"
        assert config["frame_config"]["models"][0].endswith("QNet")

    def test_unknown_algo(self):
        with pytest.raises(ValueError):
            generate_algorithm_config("NotAnAlgo")
        with pytest.raises(ValueError):
            generate_env_config("Doom-v0")

    def test_invalid_config(self):
        with pytest.raises(ValueError):
            validate_config(Config(frame="DQN"))

    def test_distributed_flag(self, tmp_path):
        assert not is_algorithm_distributed(full_config(tmpdir=str(tmp_path)))
        c = full_config("A3C", tmpdir=str(tmp_path))
        assert is_algorithm_distributed(c)

    @pytest.mark.parametrize(
        "algo,env",
        [
            ("DQN", "CartPole-v1"),
            ("DQNPer", "CartPole-v1"),
            ("RAINBOW", "CartPole-v1"),
            ("A2C", "CartPole-v1"),
            ("PPO", "CartPole-v1"),
            ("TRPO", "CartPole-v1"),
            ("DDPG", "Pendulum-v1"),
            ("HDDPG", "Pendulum-v1"),
            ("DDPGPer", "Pendulum-v1"),
            ("TD3", "Pendulum-v1"),
            ("SAC", "Pendulum-v1"),
        ],
    )
    def test_init_from_config(self, algo, env, tmp_path):
        config = full_config(algo, env, tmpdir=str(tmp_path))
        frame = init_algorithm_from_config(config)
        assert type(frame).__name__ == algo


class TestLaunch:
    def test_short_dqn_launch(self, tmp_path):
        config = full_config(tmpdir=str(tmp_path))
        config["max_episodes"] = 20
        config["episode_per_epoch"] = 10
        config["frame_config"]["batch_size"] = 16
        best = launch(config)
        assert best is not None

    def test_short_ddpg_launch(self, tmp_path):
        config = full_config("DDPG", "Pendulum-v1", tmpdir=str(tmp_path))
        config["max_episodes"] = 4
        config["episode_per_epoch"] = 2
        config["frame_config"]["batch_size"] = 16
        best = launch(config)
        assert best is not None


class TestCLI:
    def test_generate_cli(self, tmp_path):
        out = tmp_path / "conf.json"
        r = subprocess.run(
            [sys.executable, "-m", "machin_amd.auto", "generate",
             "--algo", "PPO", "--env", "CartPole-v1",
             "--output", str(out)],
            capture_output=True, text=True, cwd="/root/repo", timeout=120,
        )
        assert r.returncode == 0, r.stderr
        data = json.loads(out.read_text())
        assert data["frame"] == "PPO"

    def test_list_cli(self):
        r = subprocess.run(
            [sys.executable, "-m", "machin_amd.auto", "list"],
            capture_output=True, text=True, cwd="/root/repo", timeout=120,
        )
        assert r.returncode == 0
        assert "CartPole-v1" in r.stdout


class TestPixelCatchAutoEnv:
    def test_pixelcatch_config_and_one_update(self):
        """PixelCatch is launchable through the auto layer (flattened
        pixels on the MLP zoo; CNN models are wired manually)."""
        from machin_amd.auto.config import (
            generate_algorithm_config,
            generate_env_config,
            get_available_environments,
            init_algorithm_from_config,
        )
        from machin_amd.auto.envs.classic_control import make_dataset

        assert "PixelCatch-v0" in get_available_environments()
        cfg = generate_env_config("PixelCatch-v0")
        cfg = generate_algorithm_config("DQN", cfg)
        cfg["frame_config"]["batch_size"] = 8
        frame = init_algorithm_from_config(cfg)
        ds = make_dataset(frame, cfg.data)
        episode = next(iter(ds)).observations[0]
        assert episode[0]["state"]["state"].shape == (1, 4 * 84 * 84)
        frame.store_episode(episode)
        import numpy as np

        assert np.isfinite(frame.update())
