"""GradReducer / DistributedDataParallel correctness over gloo x2
(the same code path runs RCCL over xGMI on the GPU node)."""
import pytest
import torch as t

from util_run_multi import run_multi


class TestGradReducer:
    def test_gradients_averaged(self):
        def fn(rank, world):
            import torch.nn as nn

            from machin_amd.parallel.ddp import GradReducer

            t.manual_seed(0)  # same init everywhere
            model = nn.Sequential(
                nn.Linear(8, 32), nn.ReLU(), nn.Linear(32, 4)
            )
            reducer = GradReducer(model, bucket_cap_mb=0.0001)  # many buckets
            assert len(reducer.buckets) > 1
            t.manual_seed(100 + rank)  # different data per rank
            x = t.rand(16, 8)
            y = t.rand(16, 4)
            reducer.zero_grad_()
            loss = ((model(x) - y) ** 2).sum()
            loss.backward()
            reducer.finalize()
            # expected: average of per-rank grads computed locally
            ref = nn.Sequential(
                nn.Linear(8, 32), nn.ReLU(), nn.Linear(32, 4)
            )
            t.manual_seed(0)
            for p in ref.parameters():
                pass
            ref.load_state_dict(model.state_dict())
            grads_sum = None
            for r in range(world.world_size):
                t.manual_seed(100 + r)
                xr = t.rand(16, 8)
                yr = t.rand(16, 4)
                ref.zero_grad()
                ((ref(xr) - yr) ** 2).sum().backward()
                g = [p.grad.clone() for p in ref.parameters()]
                grads_sum = (
                    g if grads_sum is None
                    else [a + b for a, b in zip(grads_sum, g)]
                )
            expect = [g / world.world_size for g in grads_sum]
            for p, e in zip(model.parameters(), expect):
                assert t.allclose(p.grad, e, atol=1e-5)
            return True

        assert all(run_multi(fn, world_size=2))

    def test_multiple_backward_rounds(self):
        def fn(rank, world):
            import torch.nn as nn

            from machin_amd.parallel.ddp import GradReducer

            t.manual_seed(0)
            model = nn.Linear(4, 4)
            reducer = GradReducer(model)
            opt = t.optim.SGD(model.parameters(), lr=0.1)
            for step in range(5):
                t.manual_seed(step * 10 + rank)
                x = t.rand(8, 4)
                reducer.zero_grad_()
                model(x).sum().backward()
                reducer.finalize()
                opt.step()
            # ranks stay in sync
            w = model.weight.detach()
            out = [t.zeros_like(w) for _ in range(world.world_size)]
            import torch.distributed as dist

            dist.all_gather(out, w)
            assert t.allclose(out[0], out[1], atol=1e-6)
            return True

        assert all(run_multi(fn, world_size=2))

    def test_ddp_wrapper_broadcast(self):
        def fn(rank, world):
            import torch.nn as nn

            from machin_amd.parallel.ddp import DistributedDataParallel

            t.manual_seed(rank * 7)  # DIFFERENT init per rank
            model = nn.Linear(4, 2)
            ddp = DistributedDataParallel(model)
            # after wrapping, params match rank 0's
            w = ddp.module.weight.detach()
            out = [t.zeros_like(w) for _ in range(world.world_size)]
            import torch.distributed as dist

            dist.all_gather(out, w)
            assert t.allclose(out[0], out[1])
            y = ddp(t.rand(3, 4))
            assert y.shape == (3, 2)
            return True

        assert all(run_multi(fn, world_size=2))


class TestGradReducerRobustness:
    def test_survives_set_to_none_zero_grad(self):
        """optimizer.zero_grad(set_to_none=True) detaches the bucket
        views; the post-accumulate hook must fold the fresh grads back
        in so reduction still happens (round-1 ADVICE high)."""
        def fn(rank, world):
            import torch.nn as nn

            from machin_amd.parallel.ddp import GradReducer

            t.manual_seed(0)
            model = nn.Linear(8, 4)
            reducer = GradReducer(model)
            opt = t.optim.SGD(model.parameters(), lr=0.1)
            t.manual_seed(100 + rank)
            x = t.rand(16, 8)
            opt.zero_grad(set_to_none=True)  # detaches bucket views
            model(x).sum().backward()
            reducer.finalize()
            # grads must be identical (averaged) across ranks
            import torch.distributed as dist

            g = model.weight.grad.flatten().clone()
            out = [t.zeros_like(g) for _ in range(world.world_size)]
            dist.all_gather(out, g)
            assert t.allclose(out[0], out[1], atol=1e-6)
            # and the views must be re-bound for the optimizer step
            assert model.weight.grad is reducer.buckets[
                reducer._param_bucket[id(model.weight)]
            ]["view_of"][id(model.weight)]
            return True

        assert all(run_multi(fn, world_size=2))

    def test_reduce_scatter_mode(self):
        """reduce_scatter mode: averaged numerics on any backend
        (falls back to all-reduce on gloo, real RS+AG on RCCL)."""
        def fn(rank, world):
            import torch.nn as nn

            from machin_amd.parallel.ddp import GradReducer

            t.manual_seed(0)
            model = nn.Sequential(nn.Linear(8, 16), nn.Linear(16, 3))
            reducer = GradReducer(model, reduction="reduce_scatter")
            t.manual_seed(200 + rank)
            x = t.rand(4, 8)
            reducer.zero_grad_()
            model(x).sum().backward()
            reducer.finalize()
            import torch.distributed as dist

            g = t.cat([p.grad.flatten() for p in model.parameters()])
            out = [t.zeros_like(g) for _ in range(world.world_size)]
            dist.all_gather(out, g)
            assert t.allclose(out[0], out[1], atol=1e-6)
            return True

        assert all(run_multi(fn, world_size=2))

    def test_one_shot_mode(self):
        """one_shot (single-hop all-gather + local sum)
        produces the same averaged gradients as all_reduce."""
        def fn(rank, world):
            import torch.nn as nn

            from machin_amd.parallel.ddp import GradReducer

            t.manual_seed(0)
            m1 = nn.Sequential(nn.Linear(8, 16), nn.Linear(16, 3))
            t.manual_seed(0)
            m2 = nn.Sequential(nn.Linear(8, 16), nn.Linear(16, 3))
            r1 = GradReducer(m1, reduction="one_shot")
            r2 = GradReducer(m2, reduction="all_reduce")
            t.manual_seed(400 + rank)
            x = t.rand(4, 8)
            for r, m in ((r1, m1), (r2, m2)):
                r.zero_grad_()
                m(x).sum().backward()
                r.finalize()
            for p, q in zip(m1.parameters(), m2.parameters()):
                assert t.allclose(p.grad, q.grad, atol=1e-6)
            return True

        assert all(run_multi(fn, world_size=2))

    def test_matches_torch_ddp(self):
        """GradReducer's reduced grads equal torch-DDP's on the same
        model + per-rank data (round-1 VERDICT next #2)."""
        def fn(rank, world):
            import torch.nn as nn

            from machin_amd.parallel.ddp import GradReducer

            t.manual_seed(7)
            model = nn.Sequential(
                nn.Linear(8, 32), nn.ReLU(), nn.Linear(32, 4)
            )
            t.manual_seed(7)
            ref = nn.Sequential(
                nn.Linear(8, 32), nn.ReLU(), nn.Linear(32, 4)
            )
            ref = nn.parallel.DistributedDataParallel(ref)
            reducer = GradReducer(model)
            t.manual_seed(300 + rank)
            x = t.rand(16, 8)
            y = t.rand(16, 4)
            reducer.zero_grad_()
            ((model(x) - y) ** 2).mean().backward()
            reducer.finalize()
            ((ref(x) - y) ** 2).mean().backward()
            for p, q in zip(model.parameters(), ref.parameters()):
                assert t.allclose(p.grad, q.grad, atol=1e-6), (
                    f"grad mismatch vs torch DDP on rank {rank}"
                )
            return True

        assert all(run_multi(fn, world_size=2, timeout=240))
