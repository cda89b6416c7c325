"""Auto-layer distributed launch: A3C spawned across 3 processes via
machin_amd.auto.launcher.launch_distributed."""
import pytest

from machin_amd.auto.config import (
    generate_algorithm_config,
    generate_env_config,
    generate_training_config,
    launch,
)

pytestmark = pytest.mark.slow


class TestDistributedLaunch:
    def test_a3c_short_launch(self, tmp_path):
        config = generate_env_config("CartPole-v1")
        config = generate_algorithm_config("A3C", config)
        config = generate_training_config(
            root_dir=str(tmp_path), config=config
        )
        config["world_size"] = 3
        config["max_episodes"] = 30
        config["episode_per_epoch"] = 10
        config["frame_config"]["batch_size"] = 16
        config["frame_config"]["learning_rate"] = 5e-3
        # returns after all workers join; no exception = pass
        launch(config)
