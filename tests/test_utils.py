"""Tests for machin_amd.utils and small parallel helpers."""
import json
import os
import time

import numpy as np
import pytest
import torch as t
import torch.nn as nn

from machin_amd.utils.checker import CheckError, check_model, check_nan
from machin_amd.utils.conf import (
    Config,
    load_config_cmd,
    load_config_file,
    merge_config,
    save_config,
)
from machin_amd.utils.helper_classes import (
    Counter,
    Object,
    Switch,
    Timer,
    Trigger,
)
from machin_amd.utils.learning_rate import gen_learning_rate_func
from machin_amd.utils.prepare import prep_load_model
from machin_amd.utils.save_env import SaveEnv


class TestHelperClasses:
    def test_counter(self):
        c = Counter(start=5, step=2)
        c.count()
        assert c.get() == 7
        assert c > 6 and c < 8 and c == 7 and c >= 7 and c <= 7
        assert c % 3 == 1
        c.reset()
        assert c.get() == 0

    def test_switch_trigger(self):
        s = Switch()
        s.on(); assert s.get()
        s.flip(); assert not s.get()
        tr = Trigger(True)
        assert tr.get()
        assert not tr.get()  # auto-off after read

    def test_timer(self):
        tm = Timer()
        time.sleep(0.01)
        assert tm.end() >= 0.01

    def test_object(self):
        o = Object({"a": 1})
        assert o.a == 1
        assert o.missing is None
        o.b = 2
        assert o["b"] == 2 and "b" in o
        o2 = Object(const_attrs={"k"})
        with pytest.raises(RuntimeError):
            o2.k = 5


class TestConf:
    def test_file_roundtrip(self, tmp_path):
        c = Config(x=1, y="z")
        p = str(tmp_path / "c.json")
        save_config(c, p)
        c2 = load_config_file(p)
        assert c2.x == 1 and c2.y == "z"

    def test_cmdline(self):
        c = load_config_cmd(args=["--conf", "lr=0.5", "--conf", "name=abc"])
        assert c.lr == 0.5 and c.name == "abc"

    def test_merge(self):
        c = merge_config(Config(a=1), {"b": 2})
        assert c.a == 1 and c.b == 2


class TestSaveEnv:
    def test_dirs(self, tmp_path):
        env = SaveEnv(str(tmp_path))
        for d in (
            env.get_trial_config_dir(),
            env.get_trial_model_dir(),
            env.get_trial_image_dir(),
            env.get_trial_train_log_dir(),
        ):
            assert os.path.isdir(d)

    def test_restart(self, tmp_path):
        env1 = SaveEnv(str(tmp_path))
        env2 = SaveEnv(str(tmp_path),
                       restart_from_trial=env1.env_create_time)
        assert env1.get_trial_root() == env2.get_trial_root()


class TestPrepare:
    def test_load_latest_common(self, tmp_path):
        m1, m2 = nn.Linear(2, 2), nn.Linear(2, 2)
        t.save(m1.state_dict(), tmp_path / "a_1.pt")
        t.save(m1.state_dict(), tmp_path / "a_2.pt")
        t.save(m2.state_dict(), tmp_path / "b_1.pt")
        targets = {"a": nn.Linear(2, 2), "b": nn.Linear(2, 2)}
        version = prep_load_model(str(tmp_path), targets)
        assert version == 1  # only version 1 common to both
        assert t.allclose(targets["a"].weight, m1.weight)

    def test_missing_raises(self, tmp_path):
        with pytest.raises(RuntimeError):
            prep_load_model(str(tmp_path), {"x": nn.Linear(2, 2)})


class TestChecker:
    def test_nan_detection(self):
        model = nn.Linear(4, 2)
        cancel = check_model(None, model)
        model(t.rand(1, 4))  # fine
        with pytest.raises(CheckError):
            model(t.full((1, 4), float("nan")))
        cancel()
        model(t.full((1, 4), float("nan")))  # hooks removed

    def test_check_nan_fn(self):
        with pytest.raises(CheckError):
            check_nan(t.tensor([float("nan")]), "x")


class TestLearningRate:
    def test_step_table(self):
        f = gen_learning_rate_func([(0, 1e-3), (100, 1e-4)])
        assert f(0) == 1e-3
        assert f(99) == 1e-3
        assert f(100) == 1e-4
        assert f(500) == 1e-4


class TestTensorBoard:
    def test_json_fallback_writer(self, tmp_path):
        from machin_amd.utils.tensor_board import _JsonBoardWriter

        w = _JsonBoardWriter(str(tmp_path))
        w.add_scalar("a", 1.5, 3)
        w.add_histogram("h", t.rand(10), 3)
        w.add_text("t", "hello")
        w.close()
        files = [f for f in os.listdir(tmp_path) if f.endswith(".jsonl")]
        lines = open(tmp_path / files[0]).read().strip().split("\n")
        assert len(lines) == 3
        assert json.loads(lines[0])["tag"] == "a"


class TestMedia:
    def test_create_image(self, tmp_path):
        from machin_amd.utils.media import create_image

        img = np.random.rand(16, 16, 3).astype(np.float32)
        create_image(img, str(tmp_path), "frame")
        assert os.path.exists(tmp_path / "frame.png")

    def test_create_video_npz(self, tmp_path):
        from machin_amd.utils.media import create_video

        frames = [np.random.rand(8, 8, 3) for _ in range(4)]
        create_video(frames, str(tmp_path), "vid", extension="npz")
        assert os.path.exists(tmp_path / "vid.npz")


class TestVisualize:
    def test_text_graph(self, tmp_path):
        from machin_amd.utils.visualize import visualize_graph

        x = t.rand(2, 2, requires_grad=True)
        y = (x * 2 + 1).sum()
        text = visualize_graph(y, str(tmp_path), name="g")
        assert "Backward" in text
        assert os.path.exists(tmp_path / "g.txt")


class TestAssigner:
    def test_cpu_assignment(self):
        from machin_amd.parallel.assigner import ModelAssigner

        models = [nn.Linear(4, 4) for _ in range(3)]
        assigner = ModelAssigner(
            models,
            model_connection={(0, 1): 1, (1, 2): 1},
            devices=[t.device("cpu")],
            iterations=10,
        )
        assert all(d == t.device("cpu") for d in assigner.assignment)


class TestPoolRespawn:
    def test_respawn_liveness(self):
        import os as _os

        from machin_amd.parallel.pool import Pool

        pool = Pool(processes=2)
        try:
            r = pool.apply_async(lambda: _os._exit(1))
            time.sleep(0.5)
            # pool still serves new work after a worker death
            assert pool.apply(lambda: 42, ()) == 42
        finally:
            pool.terminate()


class TestTracing:
    def test_noop_on_cpu(self):
        from machin_amd.utils.tracing import trace_range, traced

        with trace_range("phase"):
            x = 1 + 1

        @traced("fn")
        def f(a):
            return a * 2

        assert f(21) == 42


class TestMediaLogger:
    def test_local_media_logger(self, tmp_path):
        from machin_amd.auto.pl_logger import LocalMediaLogger

        img_dir = tmp_path / "img"
        art_dir = tmp_path / "art"
        logger = LocalMediaLogger(str(img_dir), str(art_dir))
        logger.log_metrics({"reward": 1.5}, step=3)
        src = tmp_path / "thing.txt"
        src.write_text("data")
        logger.log_artifact(str(src))
        logger.finalize()
        assert (art_dir / "thing.txt").exists()
        assert "reward" in (art_dir / "metrics.jsonl").read_text()


class TestDataset:
    def test_determine_precision(self):
        import torch.nn as nn

        from machin_amd.auto.dataset import determine_precision

        assert determine_precision([nn.Linear(2, 2)]) == t.float32
        mixed = nn.Linear(2, 2)
        half = nn.Linear(2, 2).half()
        with pytest.raises(RuntimeError):
            determine_precision([mixed, half])

    def test_dataset_result(self):
        from machin_amd.auto.dataset import DatasetResult

        r = DatasetResult()
        r.add_observation([1, 2])
        r.add_log({"total_reward": 5})
        assert len(r) == 1 and r.logs[0]["total_reward"] == 5


class TestSpace:
    def test_discrete_and_box(self):
        from machin_amd.env.envs.classic_control import Space

        d = Space(n=4, seed=0)
        assert d.discrete
        assert all(0 <= d.sample() < 4 for _ in range(20))
        b = Space(shape=(2,), low=np.array([-1.0, 0.0]),
                  high=np.array([1.0, 2.0]), seed=0)
        s = b.sample()
        assert s.shape == (2,)
        assert -1 <= s[0] <= 1 and 0 <= s[1] <= 2


class TestSaveEnvCleanup:
    def test_remove_old_trials(self, tmp_path):
        import time as _time

        env = SaveEnv(str(tmp_path))
        # fabricate an old trial directory (2 hours ago)
        old_name = _time.strftime(
            "%Y_%m_%d_%H_%M_%S", _time.localtime(_time.time() - 7200)
        )
        os.makedirs(tmp_path / old_name)
        env.remove_trials_older_than(diff_hour=1)
        assert not (tmp_path / old_name).exists()
        assert os.path.isdir(env.get_trial_root())


class TestMediaSubproc:
    def test_image_subproc(self, tmp_path):
        from machin_amd.utils.media import create_image_subproc

        img = np.random.rand(8, 8, 3).astype(np.float32)
        wait = create_image_subproc(img, str(tmp_path), "async_frame")
        wait()
        assert os.path.exists(tmp_path / "async_frame.png")
