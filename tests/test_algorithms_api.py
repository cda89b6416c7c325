"""API tests for the single-process algorithm family."""
import numpy as np
import pytest
import torch as t
import torch.nn as nn

from machin_amd.frame.algorithms import (
    A2C,
    DDPG,
    DDPGPer,
    GAIL,
    HDDPG,
    PPO,
    RAINBOW,
    SAC,
    TD3,
    TRPO,
    DQNPer,
)

from util_models import (
    Critic,
    DetActor,
    Discriminator,
    DistQNet,
    GaussianActor,
    QNet,
    StochDiscreteActor,
    TRPODiscreteActor,
    TRPOGaussianActor,
    VCritic,
)


def cont_transition():
    return {
        "state": {"state": t.rand(1, 3)},
        "action": {"action": t.rand(1, 1) * 2 - 1},
        "next_state": {"state": t.rand(1, 3)},
        "reward": float(np.random.rand()),
        "terminal": False,
    }


def disc_transition(action=None):
    return {
        "state": {"state": t.rand(1, 4)},
        "action": {"action": action if action is not None
                   else t.randint(0, 2, (1, 1))},
        "next_state": {"state": t.rand(1, 4)},
        "reward": float(np.random.rand()),
        "terminal": False,
    }


class TestDQNPer:
    def test_update(self):
        fr = DQNPer(QNet(), QNet(), t.optim.Adam, nn.MSELoss(), batch_size=8)
        fr.store_episode([disc_transition() for _ in range(20)])
        loss = fr.update()
        assert isinstance(loss, float)
        # priorities changed
        assert fr.replay_buffer.wt_tree.get_weight_sum() > 0


class TestRainbow:
    def test_update_and_act(self):
        fr = RAINBOW(
            DistQNet(), DistQNet(), t.optim.Adam, -10.0, 10.0,
            batch_size=8, reward_future_steps=3,
        )
        eps = [disc_transition() for _ in range(20)]
        eps[-1]["terminal"] = True
        fr.store_episode(eps)
        act = fr.act_discrete({"state": t.rand(1, 4)})
        assert act.shape == (1, 1)
        act = fr.act_discrete_with_noise({"state": t.rand(1, 4)})
        assert act.shape == (1, 1)
        loss = fr.update()
        assert isinstance(loss, float) and loss == loss


class TestDDPG:
    def make(self, cls=DDPG, **kw):
        return cls(
            DetActor(), DetActor(), Critic(), Critic(),
            t.optim.Adam, nn.MSELoss(), batch_size=8, **kw,
        )

    def test_act_modes(self):
        fr = self.make()
        s = {"state": t.rand(1, 3)}
        a = fr.act(s)
        assert a.shape == (1, 1)
        for mode, param in [
            ("uniform", (-0.1, 0.1)),
            ("normal", (0.0, 0.1)),
            ("clipped_normal", (0.0, 0.1, -0.2, 0.2)),
            ("ou", {}),
        ]:
            an = fr.act_with_noise(s, noise_param=param, mode=mode)
            assert an.shape == (1, 1)
        with pytest.raises(ValueError):
            fr.act_with_noise(s, mode="bogus")

    def test_update(self):
        fr = self.make()
        fr.store_episode([cont_transition() for _ in range(20)])
        pl, vl = fr.update()
        assert isinstance(pl, float) and isinstance(vl, float)

    def test_save_load(self, tmp_path):
        fr = self.make()
        fr.store_episode([cont_transition() for _ in range(20)])
        fr.update()
        fr.save(str(tmp_path))
        fr2 = self.make()
        fr2.load(str(tmp_path))
        for p1, p2 in zip(
            fr.actor_target.parameters(), fr2.actor.parameters()
        ):
            assert t.allclose(p1, p2)


class TestHDDPG:
    def test_update(self):
        fr = HDDPG(
            DetActor(), DetActor(), Critic(), Critic(),
            t.optim.Adam, nn.MSELoss(), batch_size=8,
            q_increase_rate=1.0, q_decrease_rate=0.1,
        )
        fr.store_episode([cont_transition() for _ in range(20)])
        pl, vl = fr.update()
        assert isinstance(pl, float)


class TestTD3:
    def test_update(self):
        fr = TD3(
            DetActor(), DetActor(), Critic(), Critic(), Critic(), Critic(),
            t.optim.Adam, nn.MSELoss(), batch_size=8,
        )
        fr.store_episode([cont_transition() for _ in range(20)])
        for _ in range(3):
            pl, vl = fr.update()
        assert isinstance(vl, float)


class TestDDPGPer:
    def test_update(self):
        fr = DDPGPer(
            DetActor(), DetActor(), Critic(), Critic(),
            t.optim.Adam, nn.MSELoss(), batch_size=8,
        )
        fr.store_episode([cont_transition() for _ in range(20)])
        pl, vl = fr.update()
        assert isinstance(vl, float)


class TestSAC:
    def test_update(self):
        fr = SAC(
            GaussianActor(), Critic(), Critic(), Critic(), Critic(),
            t.optim.Adam, nn.MSELoss(), batch_size=8, target_entropy=-1.0,
        )
        fr.store_episode([cont_transition() for _ in range(20)])
        pl, vl = fr.update()
        assert isinstance(pl, float)
        a, lp = fr.act({"state": t.rand(1, 3)})[:2]
        assert a.shape == (1, 1) and lp.shape == (1, 1)


class TestA2C:
    def make(self, cls=A2C, **kw):
        return cls(
            StochDiscreteActor(), VCritic(), t.optim.Adam, nn.MSELoss(),
            entropy_weight=0.01, gae_lambda=0.95, **kw,
        )

    def _episode(self, fr, length=10):
        eps = [disc_transition() for _ in range(length)]
        eps[-1]["terminal"] = True
        fr.store_episode(eps)

    def test_store_computes_gae(self):
        fr = self.make()
        self._episode(fr)
        bs, batch = fr.replay_buffer.sample_batch(
            -1, sample_method="all",
            sample_attrs=["value", "gae"],
            additional_concat_custom_attrs=["value", "gae"],
        )
        assert bs == 10
        assert batch[0].shape == (10, 1)
        assert batch[1].shape == (10, 1)

    def test_update_clears_buffer(self):
        fr = self.make()
        self._episode(fr)
        pl, vl = fr.update()
        assert fr.replay_buffer.size() == 0
        assert isinstance(pl, float)

    def test_act(self):
        fr = self.make()
        a, lp, ent = fr.act({"state": t.rand(1, 4)})
        assert a.shape == (1, 1)

    def test_store_transition_raises(self):
        fr = self.make()
        with pytest.raises(NotImplementedError):
            fr.store_transition(disc_transition())


class TestPPO:
    def test_update(self):
        fr = PPO(
            StochDiscreteActor(), VCritic(), t.optim.Adam, nn.MSELoss(),
            surrogate_loss_clip=0.2,
        )
        eps = [disc_transition() for _ in range(10)]
        eps[-1]["terminal"] = True
        fr.store_episode(eps)
        pl, vl = fr.update()
        assert isinstance(pl, float)
        assert fr.replay_buffer.size() == 0


class TestTRPO:
    @pytest.mark.parametrize("actor_cls", [TRPODiscreteActor])
    def test_update_discrete(self, actor_cls):
        fr = TRPO(
            actor_cls(), VCritic(), t.optim.Adam, nn.MSELoss(),
        )
        eps = []
        for _ in range(15):
            st = t.rand(1, 4)
            with t.no_grad():
                a, _, _ = fr.actor(st)
            d = disc_transition(action=a)
            d["state"] = {"state": st}
            eps.append(d)
        eps[-1]["terminal"] = True
        fr.store_episode(eps)
        pl, vl = fr.update()
        assert isinstance(pl, float) and pl == pl

    def test_update_continuous(self):
        actor = TRPOGaussianActor()
        fr = TRPO(
            actor, VCritic(state_dim=3), t.optim.Adam, nn.MSELoss(),
        )
        eps = []
        for _ in range(15):
            st = t.rand(1, 3)
            with t.no_grad():
                a, _, _ = actor(st)
            eps.append(
                {
                    "state": {"state": st},
                    "action": {"action": a},
                    "next_state": {"state": t.rand(1, 3)},
                    "reward": float(np.random.rand()),
                    "terminal": False,
                }
            )
        fr.store_episode(eps)
        pl, vl = fr.update()
        assert pl == pl and vl == vl


class TestGAIL:
    def test_update(self):
        ppo = PPO(
            StochDiscreteActor(), VCritic(), t.optim.Adam, nn.MSELoss(),
        )
        fr = GAIL(Discriminator(), ppo, t.optim.Adam)
        # expert data
        fr.store_expert_episode(
            [
                {"state": {"state": t.rand(1, 4)},
                 "action": {"action": t.randint(0, 2, (1, 1))}}
                for _ in range(10)
            ]
        )
        eps = [disc_transition() for _ in range(10)]
        eps[-1]["terminal"] = True
        fr.store_episode(eps)
        result = fr.update()
        assert isinstance(result[-1], float)


class TestNoise:
    def test_generators(self):
        from machin_amd.frame.noise import (
            ClippedNormalNoiseGen,
            NormalNoiseGen,
            OrnsteinUhlenbeckNoiseGen,
            UniformNoiseGen,
        )

        for gen in [
            NormalNoiseGen((2, 3)),
            ClippedNormalNoiseGen((2, 3)),
            UniformNoiseGen((2, 3)),
            OrnsteinUhlenbeckNoiseGen((2, 3)),
        ]:
            n = gen("cpu")
            assert n.shape == (2, 3)
            repr(gen)

    def test_ou_correlated(self):
        from machin_amd.frame.noise import OrnsteinUhlenbeckNoiseGen

        gen = OrnsteinUhlenbeckNoiseGen((1,), theta=0.15, sigma=0.2)
        a = gen("cpu").item()
        b = gen("cpu").item()
        gen.reset()
        assert gen.x_prev.sum() == 0

    def test_param_space_noise(self):
        from machin_amd.frame.noise import perturb_model
        from machin_amd.utils.helper_classes import Switch

        net = nn.Linear(4, 2)
        p_switch, r_switch = Switch(), Switch()
        cancel, spec = perturb_model(net, p_switch, r_switch)
        p_switch.on()
        r_switch.on()
        x = t.rand(1, 4)
        with t.no_grad():
            out1 = net(x)
            p_switch.off()
            clean = net(x)
        cancel()
        with t.no_grad():
            out_after_cancel = net(x)
        assert t.allclose(clean, out_after_cancel)
        assert not t.allclose(out1, clean)  # perturbed differs
