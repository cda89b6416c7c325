"""GPU integration: every single-process algorithm updates with
models on cuda:0, driving the HIP kernels (fused polyak, C51
projection, GAE scans, tanh-gaussian) through the real frameworks."""
import numpy as np
import pytest
import torch as t
import torch.nn as nn

pytestmark = pytest.mark.gpu

if not t.cuda.is_available():
    pytest.skip("no GPU", allow_module_level=True)

DEV = "cuda:0"


def cont_transition():
    return {
        "state": {"state": t.rand(1, 3, device=DEV)},
        "action": {"action": t.rand(1, 1, device=DEV) * 2 - 1},
        "next_state": {"state": t.rand(1, 3, device=DEV)},
        "reward": float(np.random.rand()),
        "terminal": False,
    }


def disc_transition():
    return {
        "state": {"state": t.rand(1, 4, device=DEV)},
        "action": {"action": t.randint(0, 2, (1, 1), device=DEV)},
        "next_state": {"state": t.rand(1, 4, device=DEV)},
        "reward": float(np.random.rand()),
        "terminal": False,
    }


class TestGPUAlgorithms:
    def test_hip_extension_loaded(self):
        import machin_amd.ops as ops

        assert ops.available(), (
            "HIP extension must be importable on the GPU box — "
            "the native path is required, not the eager fallback"
        )

    def test_dqn_gpu(self):
        import sys
        sys.path.insert(0, "tests")
        from util_models import QNet

        from machin_amd.frame.algorithms import DQN

        dqn = DQN(
            QNet().to(DEV), QNet().to(DEV), t.optim.Adam, nn.MSELoss(),
            batch_size=8, replay_device=DEV,
        )
        dqn.store_episode([disc_transition() for _ in range(20)])
        loss = dqn.update()
        assert loss == loss
        # soft_update ran the fused HIP polyak on device params
        act = dqn.act_discrete({"state": t.rand(1, 4, device=DEV)})
        assert act.shape == (1, 1)

    def test_td3_gpu(self):
        import sys
        sys.path.insert(0, "tests")
        from util_models import Critic, DetActor

        from machin_amd.frame.algorithms import TD3

        td3 = TD3(
            DetActor().to(DEV), DetActor().to(DEV),
            Critic().to(DEV), Critic().to(DEV),
            Critic().to(DEV), Critic().to(DEV),
            t.optim.Adam, nn.MSELoss(), batch_size=8, replay_device=DEV,
        )
        td3.store_episode([cont_transition() for _ in range(20)])
        for _ in range(3):
            pl, vl = td3.update()
        assert vl == vl

    def test_sac_gpu(self):
        import sys
        sys.path.insert(0, "tests")
        from util_models import Critic, GaussianActor

        from machin_amd.frame.algorithms import SAC

        sac = SAC(
            GaussianActor().to(DEV), Critic().to(DEV), Critic().to(DEV),
            Critic().to(DEV), Critic().to(DEV),
            t.optim.Adam, nn.MSELoss(), batch_size=8, replay_device=DEV,
            target_entropy=-1.0,
        )
        sac.store_episode([cont_transition() for _ in range(20)])
        pl, vl = sac.update()
        assert pl == pl

    def test_rainbow_gpu(self):
        import sys
        sys.path.insert(0, "tests")
        from util_models import DistQNet

        from machin_amd.frame.algorithms import RAINBOW

        fr = RAINBOW(
            DistQNet().to(DEV), DistQNet().to(DEV), t.optim.Adam,
            -10.0, 10.0, batch_size=8, replay_device=DEV,
        )
        eps = [disc_transition() for _ in range(20)]
        eps[-1]["terminal"] = True
        fr.store_episode(eps)
        loss = fr.update()  # C51 projection runs the HIP kernel
        assert loss == loss

    def test_ppo_gpu(self):
        import sys
        sys.path.insert(0, "tests")
        from util_models import StochDiscreteActor, VCritic

        from machin_amd.frame.algorithms import PPO

        fr = PPO(
            StochDiscreteActor().to(DEV), VCritic().to(DEV),
            t.optim.Adam, nn.MSELoss(), replay_device=DEV,
        )
        eps = [disc_transition() for _ in range(10)]
        eps[-1]["terminal"] = True
        fr.store_episode(eps)  # GAE runs the HIP scan kernel
        pl, vl = fr.update()
        assert pl == pl

    def test_bench_smoke(self):
        """One IMPALA learner step (the graft smoke path)."""
        import sys

        sys.path.insert(0, ".")
        from bench import ImpalaLearnerBench

        b = ImpalaLearnerBench(
            device=DEV, unroll=8, env_batch=16, dtype=t.bfloat16
        )
        loss = b.step()
        t.cuda.synchronize()
        assert loss == loss
