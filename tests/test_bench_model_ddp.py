"""The bench model + GradReducer over gloo x2 on CPU — de-risks the
exact module/bucket structure the 8-GPU RCCL run uses."""
import pytest
import torch as t

from util_run_multi import run_multi


class TestBenchModelReducer:
    def test_actor_critic_cnn_reduces(self):
        def fn(rank, world):
            import torch.nn as nn

            from machin_amd.model.nets.nature_cnn import ActorCriticCNN
            from machin_amd.parallel.ddp import GradReducer

            t.manual_seed(0)
            model = ActorCriticCNN(4, 6)
            reducer = GradReducer(model, bucket_cap_mb=1.0)
            assert reducer.world_size == 2
            t.manual_seed(rank)
            frames = t.rand(4, 4, 84, 84)
            logits, values = model(frames)
            loss = logits.sum() + values.sum()
            reducer.zero_grad_()
            loss.backward()
            reducer.finalize()
            # grads identical on both ranks after all-reduce
            g = model.policy.weight.grad.flatten()[:16].clone()
            import torch.distributed as dist

            out = [t.zeros_like(g) for _ in range(2)]
            dist.all_gather(out, g)
            assert t.allclose(out[0], out[1], atol=1e-6)
            # and every parameter received a grad view
            assert all(p.grad is not None for p in model.parameters())
            return True

        assert all(run_multi(fn, world_size=2, timeout=240))
