"""Randomized consistency tests for the ops CPU fallbacks against
independent brute-force formulations (the GPU kernels are separately
tested against these fallbacks in tests/test_ops_gpu.py, so this
chain pins kernel semantics to first-principles math)."""
import numpy as np
import torch as t

import machin_amd.ops as ops


def brute_nstep(rewards, terminals, gamma, n):
    T, B = rewards.shape
    out = t.zeros_like(rewards)
    for ti in range(T):
        for b in range(B):
            g, factor, alive = 0.0, 1.0, 1.0
            for k in range(n):
                if ti + k >= T:
                    break
                g += factor * alive * float(rewards[ti + k, b])
                alive *= 1.0 - float(terminals[ti + k, b])
                factor *= gamma
            out[ti, b] = g
    return out


def brute_vtrace(blp, tlp, rew, val, boot, term, gamma, rc, cc, pc):
    T, B = rew.shape
    vs = t.zeros(T, B)
    pg = t.zeros(T, B)
    for b in range(B):
        acc = 0.0
        vs_next = float(boot[b])
        v_next = float(boot[b])
        for i in range(T - 1, -1, -1):
            rho = float(np.exp(float(tlp[i, b]) - float(blp[i, b])))
            nd = 1.0 - float(term[i, b])
            vk = float(val[i, b])
            delta = min(rho, rc) * (
                float(rew[i, b]) + gamma * nd * v_next - vk
            )
            acc = delta + gamma * nd * min(rho, cc) * acc
            vs[i, b] = acc + vk
            pg[i, b] = min(rho, pc) * (
                float(rew[i, b]) + gamma * nd * vs_next - vk
            )
            vs_next = float(vs[i, b])
            v_next = vk
    return vs, pg


def brute_projection(next_dist, rewards, terminals, gamma, vmin, vmax):
    B, A = next_dist.shape
    dz = (vmax - vmin) / (A - 1)
    out = t.zeros(B, A)
    for b in range(B):
        for j in range(A):
            z = vmin + j * dz
            tz = float(rewards[b]) + gamma * (
                1.0 - float(terminals[b])
            ) * z
            tz = min(max(tz, vmin), vmax)
            pos = (tz - vmin) / dz
            lo, hi = int(np.floor(pos)), int(np.ceil(pos))
            p = float(next_dist[b, j])
            if lo == hi:
                out[b, lo] += p
            else:
                out[b, lo] += p * (hi - pos)
                out[b, hi] += p * (pos - lo)
    return out


class TestOpsFuzz:
    def test_nstep_random(self):
        for seed in range(5):
            g = t.Generator().manual_seed(seed)
            T = int(t.randint(1, 30, (1,), generator=g))
            B = int(t.randint(1, 9, (1,), generator=g))
            n = int(t.randint(1, 6, (1,), generator=g))
            rew = t.rand(T, B, generator=g)
            term = (t.rand(T, B, generator=g) > 0.7).float()
            got = ops.nstep_returns(rew, term, 0.9, n)
            want = brute_nstep(rew, term, 0.9, n)
            assert t.allclose(got, want, atol=1e-5), (seed, T, B, n)

    def test_vtrace_random(self):
        for seed in range(5):
            g = t.Generator().manual_seed(100 + seed)
            T = int(t.randint(2, 25, (1,), generator=g))
            B = int(t.randint(1, 7, (1,), generator=g))
            blp = t.randn(T, B, generator=g) * 0.5
            tlp = t.randn(T, B, generator=g) * 0.5
            rew = t.randn(T, B, generator=g)
            val = t.randn(T, B, generator=g)
            boot = t.randn(B, generator=g)
            term = (t.rand(T, B, generator=g) > 0.8).float()
            vs, pg = ops.vtrace(blp, tlp, rew, val, boot, term, 0.95,
                                rho_clip=1.2, c_clip=1.05,
                                pg_rho_clip=1.1)
            wvs, wpg = brute_vtrace(blp, tlp, rew, val, boot, term,
                                    0.95, 1.2, 1.05, 1.1)
            assert t.allclose(vs, wvs, atol=1e-4), seed
            assert t.allclose(pg, wpg, atol=1e-4), seed

    def test_projection_random(self):
        for seed in range(5):
            g = t.Generator().manual_seed(200 + seed)
            B = int(t.randint(1, 16, (1,), generator=g))
            A = int(t.randint(2, 31, (1,), generator=g))
            dist = t.softmax(t.randn(B, A, generator=g), dim=1)
            rew = t.randn(B, generator=g) * 3
            term = (t.rand(B, generator=g) > 0.8).float()
            got = ops.categorical_projection(dist, rew, term, 0.99,
                                             -5.0, 5.0)
            want = brute_projection(dist, rew, term, 0.99, -5.0, 5.0)
            assert t.allclose(got, want, atol=1e-5), (seed, B, A)
            # probability mass is conserved
            assert t.allclose(got.sum(1), dist.sum(1), atol=1e-5)

    def test_cpu_sumtree_random_ops(self):
        from machin_amd.frame.buffers.prioritized_buffer import (
            WeightTree,
        )
        from machin_amd.ops.sumtree import DeviceSumTree

        for seed in range(4):
            g = t.Generator().manual_seed(300 + seed)
            n = int(t.randint(2, 200, (1,), generator=g))
            tree = DeviceSumTree(n, "cpu")
            ref = WeightTree(n)
            w0 = t.rand(n, generator=g) + 0.01
            tree.update_all_leaves(w0)
            ref.update_leaf_batch(w0.double().numpy(),
                                  np.arange(n, dtype=np.int64))
            for _ in range(3):
                k = int(t.randint(1, n + 1, (1,), generator=g))
                idx = t.randperm(n, generator=g)[:k]
                w = t.rand(k, generator=g) + 0.01
                tree.update_leaf_batch(w, idx)
                ref.update_leaf_batch(w.double().numpy(),
                                      idx.numpy().astype(np.int64))
            assert abs(tree.get_weight_sum()
                       - ref.get_weight_sum()) < 1e-3
            qs = t.rand(64, generator=g) * (tree.get_weight_sum() - 1e-4)
            got = tree.find_leaf_index(qs)
            want = ref.find_leaf_index(qs.double().numpy())
            assert (got.numpy() == want).all(), seed
