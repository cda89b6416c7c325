"""Multi-learner (learner_process_number=2) integration over gloo:
the exact DDP-wrap + finalize + zero_grad wiring the 8-GPU RCCL
scaling run uses (round-1 ADVICE high / VERDICT next #2).

Each learner rank updates on DIFFERENT data; after the update the
learner models must be bit-identical across learner ranks — which can
only happen if gradients were actually all-reduced and averaged.
"""
import numpy as np
import torch as t

from util_run_multi import run_multi

_ZOO = "machin_amd.auto.model_zoo."


def _flat_params(model):
    model = getattr(model, "module", model)
    return t.cat(
        [p.detach().float().view(-1) for p in model.parameters()]
    ).numpy()


class TestDQNApexMultiLearner:
    def test_two_learners_stay_in_sync(self):
        def fn(rank, world):
            from machin_amd.frame.algorithms import DQNApex

            config = DQNApex.generate_config({})
            fc = config["frame_config"]
            fc["models"] = [_ZOO + "QNet", _ZOO + "QNet"]
            fc["model_kwargs"] = (
                {"state_dim": 4, "action_num": 2},
                {"state_dim": 4, "action_num": 2},
            )
            fc["learner_process_number"] = 2
            fc["batch_size"] = 8
            fc["replay_size"] = 100
            frame = DQNApex.init_from_config(config)
            group = world.groups["apex_group"]
            group.barrier()

            # every rank stores different episodes
            t.manual_seed(1000 + rank)
            for _ in range(2):
                episode = [
                    {
                        "state": {"state": t.rand(1, 4)},
                        "action": {"action": t.randint(0, 2, (1, 1))},
                        "next_state": {"state": t.rand(1, 4)},
                        "reward": float(t.rand(1)),
                        "terminal": i == 4,
                    }
                    for i in range(5)
                ]
                frame.store_episode(episode)
            group.barrier()

            out = None
            if rank in (0, 1):
                # learners sample different random batches -> without
                # grad sync their params would diverge immediately
                for _ in range(3):
                    loss = frame.update()
                    assert isinstance(loss, float)
                out = _flat_params(frame.qnet)
            group.barrier()
            return out

        results = run_multi(fn, world_size=3, timeout=240)
        assert results[0] is not None and results[1] is not None
        assert np.allclose(results[0], results[1], atol=1e-6), (
            "learner models diverged: gradients were not synchronized"
        )
        assert results[2] is None


class TestDQNApexOneShotReduction:
    def test_one_shot_config_path(self):
        """ddp_reduction="one_shot" flows from the config into the
        learner GradReducer and keeps ranks in sync."""
        def fn(rank, world):
            from machin_amd.frame.algorithms import DQNApex

            config = DQNApex.generate_config({})
            fc = config["frame_config"]
            fc["models"] = [_ZOO + "QNet", _ZOO + "QNet"]
            fc["model_kwargs"] = (
                {"state_dim": 4, "action_num": 2},
                {"state_dim": 4, "action_num": 2},
            )
            fc["learner_process_number"] = 2
            fc["ddp_reduction"] = "one_shot"
            fc["batch_size"] = 8
            fc["replay_size"] = 100
            frame = DQNApex.init_from_config(config)
            group = world.groups["apex_group"]
            group.barrier()
            if rank in (0, 1):
                assert frame.qnet.reducer.reduction == "one_shot"
            t.manual_seed(1500 + rank)
            episode = [
                {
                    "state": {"state": t.rand(1, 4)},
                    "action": {"action": t.randint(0, 2, (1, 1))},
                    "next_state": {"state": t.rand(1, 4)},
                    "reward": float(t.rand(1)),
                    "terminal": i == 4,
                }
                for i in range(5)
            ]
            frame.store_episode(episode)
            group.barrier()
            out = None
            if rank in (0, 1):
                for _ in range(2):
                    frame.update()
                out = _flat_params(frame.qnet)
            group.barrier()
            return out

        results = run_multi(fn, world_size=3, timeout=240)
        assert np.allclose(results[0], results[1], atol=1e-6)


class TestIMPALAMultiLearner:
    def test_two_learners_stay_in_sync(self):
        def fn(rank, world):
            from machin_amd.frame.algorithms import IMPALA

            config = IMPALA.generate_config({})
            fc = config["frame_config"]
            fc["models"] = [_ZOO + "StochasticActor", _ZOO + "VCritic"]
            fc["model_kwargs"] = (
                {"state_dim": 4, "action_num": 2},
                {"state_dim": 4},
            )
            fc["learner_process_number"] = 2
            fc["batch_size"] = 2
            frame = IMPALA.init_from_config(config)
            group = world.groups["impala_group"]
            group.barrier()

            # learner ranks store 2 episodes each (different data);
            # the two learners partition the 4 episodes between them
            if rank in (0, 1):
                t.manual_seed(2000 + rank)
                for _ in range(2):
                    episode = [
                        {
                            "state": {"state": t.rand(1, 4)},
                            "action": {
                                "action": t.randint(0, 2, (1, 1))
                            },
                            "next_state": {"state": t.rand(1, 4)},
                            "reward": float(t.rand(1)),
                            "terminal": i == 4,
                            "action_log_prob": float(
                                t.log(t.rand(1) * 0.5 + 0.25)
                            ),
                        }
                        for i in range(5)
                    ]
                    frame.store_episode(episode)
            group.barrier()

            out = None
            if rank in (0, 1):
                act_loss, value_loss = frame.update()
                assert isinstance(act_loss, float)
                out = (
                    _flat_params(frame.actor),
                    _flat_params(frame.critic),
                )
            group.barrier()
            return out

        results = run_multi(fn, world_size=3, timeout=240)
        assert results[0] is not None and results[1] is not None
        for part, name in zip(range(2), ("actor", "critic")):
            assert np.allclose(
                results[0][part], results[1][part], atol=1e-6
            ), f"{name} diverged across learner ranks"

    def test_update_policy_flag(self):
        """update(update_policy=False) must leave the actor untouched
        (reference contract, round-1 ADVICE low)."""
        def fn(rank, world):
            from machin_amd.frame.algorithms import IMPALA

            config = IMPALA.generate_config({})
            fc = config["frame_config"]
            fc["models"] = [_ZOO + "StochasticActor", _ZOO + "VCritic"]
            fc["model_kwargs"] = (
                {"state_dim": 4, "action_num": 2},
                {"state_dim": 4},
            )
            fc["batch_size"] = 2
            fc["learner_process_number"] = 1
            frame = IMPALA.init_from_config(config)
            group = world.groups["impala_group"]
            group.barrier()
            if rank == 0:
                t.manual_seed(3000)
                episode = [
                    {
                        "state": {"state": t.rand(1, 4)},
                        "action": {"action": t.randint(0, 2, (1, 1))},
                        "next_state": {"state": t.rand(1, 4)},
                        "reward": 1.0,
                        "terminal": i == 4,
                        "action_log_prob": -0.5,
                    }
                    for i in range(5)
                ]
                frame.store_episode(episode)
                before_actor = _flat_params(frame.actor)
                before_critic = _flat_params(frame.critic)
                frame.update(update_policy=False)
                assert np.allclose(
                    before_actor, _flat_params(frame.actor)
                ), "actor stepped despite update_policy=False"
                assert not np.allclose(
                    before_critic, _flat_params(frame.critic)
                ), "critic did not step"
            group.barrier()
            return True

        assert all(run_multi(fn, world_size=3, timeout=240))
