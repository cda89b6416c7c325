"""Model layer tests: NeuralNetworkModule, wrappers, nets, TRPO
bases, size estimator."""
import pytest
import torch as t
import torch.nn as nn

from machin_amd.frame.algorithms.utils import safe_call, safe_return
from machin_amd.model.nets import (
    ActorCriticCNN,
    NatureCNN,
    NeuralNetworkModule,
    ResNet,
    dynamic_module_wrapper,
    mlp,
    static_module_wrapper,
)
from machin_amd.parallel.assigner import ModelSizeEstimator


class TestNeuralNetworkModule:
    def test_device_via_modules(self):
        class Net(NeuralNetworkModule):
            def __init__(self):
                super().__init__()
                self.fc = nn.Linear(4, 2)
                self.set_input_module(self.fc)
                self.set_output_module(self.fc)

            def forward(self, state):
                return self.fc(state)

        net = Net()
        assert net.input_device == t.device("cpu")
        assert net.output_device == t.device("cpu")

    def test_static_wrapper(self):
        net = static_module_wrapper(nn.Linear(4, 2), "cpu", "cpu")
        assert net.input_device == t.device("cpu")

    def test_dynamic_wrapper(self):
        wrapped = dynamic_module_wrapper(nn.Linear(4, 2))
        assert wrapped.input_device == t.device("cpu")
        out = wrapped(t.rand(1, 4))
        assert out.shape == (1, 2)

    def test_find_child(self):
        seq = nn.Sequential(nn.Sequential(nn.Linear(4, 8)), nn.ReLU())
        first = NeuralNetworkModule.find_child(seq, True)
        assert isinstance(first, nn.Linear)


class TestSafeCall:
    def test_moves_and_filters(self):
        class Net(nn.Module):
            def __init__(self):
                super().__init__()
                self.fc = nn.Linear(4, 2)

            def forward(self, state):
                return self.fc(state)

        net = Net()
        out = safe_call(net, {"state": t.rand(1, 4), "extra": 123})
        assert out.shape == (1, 2)

    def test_missing_arg_raises(self):
        class Net(nn.Module):
            def __init__(self):
                super().__init__()
                self.fc = nn.Linear(4, 2)

            def forward(self, state, other):
                return self.fc(state)

        with pytest.raises(RuntimeError, match="other"):
            safe_call(Net(), {"state": t.rand(1, 4)})

    def test_method_dispatch(self):
        class Net(nn.Module):
            def __init__(self):
                super().__init__()
                self.fc = nn.Linear(4, 2)

            def forward(self, state):
                return self.fc(state)

            def evaluate(self, state):
                return self.fc(state) * 2

        net = Net()
        a = safe_call(net, {"state": t.ones(1, 4)})
        b = safe_call(net, {"state": t.ones(1, 4)}, method="evaluate")
        assert t.allclose(b, a * 2)

    def test_safe_return(self):
        assert safe_return((5,)) == 5
        assert safe_return((1, 2)) == (1, 2)
        assert safe_return(7) == 7


class TestNets:
    def test_nature_cnn_shapes(self):
        net = NatureCNN(4)
        out = net(t.rand(2, 4, 84, 84))
        assert out.shape == (2, 512)

    def test_actor_critic_cnn(self):
        net = ActorCriticCNN(4, 6)
        logits, value = net(t.rand(2, 4, 84, 84))
        assert logits.shape == (2, 6)
        assert value.shape == (2, 1)

    def test_mlp_builder(self):
        net = mlp([4, 16, 2])
        assert net(t.rand(3, 4)).shape == (3, 2)

    @pytest.mark.parametrize("depth", [18, 50])
    def test_resnet(self, depth):
        net = ResNet(3, depth, 10)
        out = net(t.rand(1, 3, 64, 64))
        assert out.shape == (1, 10)

    def test_resnet_bad_depth(self):
        with pytest.raises(ValueError):
            ResNet(3, 19, 10)


class TestTRPOBases:
    def test_discrete_contract(self):
        from util_models import TRPODiscreteActor

        actor = TRPODiscreteActor()
        s = t.rand(4, 4)
        a, lp, ent = actor(s)
        assert a.shape == (4, 1) and lp.shape == (4, 1)
        kl = actor.get_kl(s)
        assert kl.item() == pytest.approx(0.0, abs=1e-6)
        old = actor.get_dist_params(s)
        assert actor.compare_kl(old, s).item() == pytest.approx(
            0.0, abs=1e-6
        )

    def test_continuous_contract(self):
        from util_models import TRPOGaussianActor

        actor = TRPOGaussianActor()
        s = t.rand(4, 3)
        a, lp, ent = actor(s)
        assert a.shape == (4, 1)
        assert actor.get_kl(s).item() == pytest.approx(0.0, abs=1e-6)
        mean, log_std = actor.get_dist_params(s)
        assert actor.compare_kl(mean, log_std, s).item() == pytest.approx(
            0.0, abs=1e-6
        )


class TestSizeEstimator:
    def test_estimate(self):
        est = ModelSizeEstimator(nn.Linear(1024, 1024),
                                 size_multiplier=1)
        # ~1M params x 4 bytes = ~4 MiB
        assert 3.9 < est.estimate_size() < 4.2
