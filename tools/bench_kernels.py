"""Microbenchmarks of the gfx950 kernels vs the reference CPU numbers.

Run on a GPU box:  python tools/bench_kernels.py
Writes JSON to gpurun_out/kernel_bench.json.

Reference CPU baselines:
  sum-tree build 10M leaves           90 ms
  sum-tree lookup 10M x 10M-leaf     230 ms
  sum-tree batched update 1M         20 ms
"""
import json
import os
import time

import sys
sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))

import numpy as np
import torch as t


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    t.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    t.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000.0  # ms


def main():
    from machin_amd.ops.sumtree import DeviceSumTree
    import machin_amd.ops as ops
    from machin_amd.frame.buffers import WeightTree

    dev = t.device("cuda:0")
    results = {}

    # --- sum-tree: the BASELINE.md comparison sizes -------------------
    n_leaf = 10_000_000
    tree = DeviceSumTree(n_leaf, dev)
    w = t.rand(n_leaf, device=dev)
    results["sumtree_build_10M_ms"] = timeit(
        lambda: tree.update_all_leaves(w), iters=10
    )
    u = t.rand(10_000_000, device=dev) * tree.get_weight_sum_tensor()
    results["sumtree_lookup_10M_ms"] = timeit(
        lambda: tree.find_leaf_index(u), iters=10
    )
    idx = t.randint(0, n_leaf, (1_000_000,), device=dev)
    nw = t.rand(1_000_000, device=dev)
    results["sumtree_update_1M_ms"] = timeit(
        lambda: tree.update_leaf_batch(nw, idx), iters=10
    )
    # typical learner-shaped ops
    idx_s = t.randint(0, n_leaf, (512,), device=dev)
    nw_s = t.rand(512, device=dev)
    results["sumtree_update_512_ms"] = timeit(
        lambda: tree.update_leaf_batch(nw_s, idx_s), iters=50
    )
    results["sumtree_sample_512_ms"] = timeit(
        lambda: tree.sample(512), iters=50
    )
    del tree, w, u, idx, nw

    # CPU comparison (same machine, numpy tree)
    cpu_tree = WeightTree(n_leaf)
    w_np = np.random.rand(n_leaf)
    t0 = time.perf_counter()
    cpu_tree.update_all_leaves(w_np)
    results["cpu_sumtree_build_10M_ms"] = (time.perf_counter() - t0) * 1000
    q = np.random.uniform(0, cpu_tree.get_weight_sum(), 10_000_000)
    t0 = time.perf_counter()
    cpu_tree.find_leaf_index(q)
    results["cpu_sumtree_lookup_10M_ms"] = (time.perf_counter() - t0) * 1000
    idx_np = np.random.randint(0, n_leaf, 1_000_000)
    wn_np = np.random.rand(1_000_000)
    t0 = time.perf_counter()
    cpu_tree.update_leaf_batch(wn_np, idx_np)
    results["cpu_sumtree_update_1M_ms"] = (time.perf_counter() - t0) * 1000
    del cpu_tree

    # --- scans --------------------------------------------------------
    T, B = 20, 65536
    rew = t.rand(T, B, device=dev)
    term = (t.rand(T, B, device=dev) > 0.98).float()
    val = t.rand(T, B, device=dev)
    nxt = t.rand(T, B, device=dev)
    boot = t.rand(B, device=dev)
    blp = -t.rand(T, B, device=dev)
    tlp = -t.rand(T, B, device=dev)
    results["gae_20x65536_ms"] = timeit(
        lambda: ops.gae(rew, val, nxt, term, 0.99, 0.95)
    )
    results["vtrace_20x65536_ms"] = timeit(
        lambda: ops.vtrace(blp, tlp, rew, val, boot, term, 0.99)
    )

    # --- projection ---------------------------------------------------
    Bp, A = 4096, 51
    dist = t.softmax(t.randn(Bp, A, device=dev), dim=1)
    rp = t.randn(Bp, device=dev)
    tp = (t.rand(Bp, device=dev) > 0.9).float()
    results["projection_4096x51_ms"] = timeit(
        lambda: ops.categorical_projection(dist, rp, tp, 0.99, -10, 10)
    )

    # --- polyak -------------------------------------------------------
    from machin_amd.model.nets.nature_cnn import ActorCriticCNN

    m1 = ActorCriticCNN().to(dev)
    m2 = ActorCriticCNN().to(dev)
    tl = [p.data for p in m1.parameters()]
    sl = [p.data for p in m2.parameters()]
    results["polyak_naturecnn_ms"] = timeit(
        lambda: ops.polyak_update_(tl, sl, 0.005), iters=100
    )

    def torch_polyak():
        t._foreach_mul_(tl, 0.995)
        t._foreach_add_(tl, sl, alpha=0.005)

    results["polyak_naturecnn_foreach_ms"] = timeit(torch_polyak, iters=100)

    # cached plan: one launch, zero per-call host setup (round 2)
    plan = ops.FusedPolyak(tl, sl)
    results["polyak_naturecnn_cached_ms"] = timeit(
        lambda: plan(0.005), iters=100
    )

    # --- n-step returns ----------------------------------------------
    Tn, Bn = 64, 4096
    rw = t.rand(Tn, Bn, device=dev)
    al = (t.rand(Tn, Bn, device=dev) > 0.02).float()
    from machin_amd.ops import _machin_hip as ext

    results["nstep_64x4096_n3_kernel_ms"] = timeit(
        lambda: ext.nstep_returns(rw, al, 0.99, 3)
    )

    def nstep_torch():
        pad = t.zeros(1, Bn, device=dev)
        out = t.zeros_like(rw)
        for _ in range(3):
            out = rw + 0.99 * al * t.cat([out[1:], pad], dim=0)
        return out

    results["nstep_64x4096_n3_torch_ms"] = timeit(nstep_torch)

    # --- conv1 stem kernels: v1 vs v3 A/B at 1/5 bench K ------------
    Bc = 8192
    Kc = Bc * 400
    frames_c = t.randint(0, 256, (Bc, 84, 84, 4), dtype=t.uint8,
                         device=dev)
    gy = (t.randn(Kc, 32, device=dev) * 0.1).to(t.bfloat16).contiguous()
    wgt = (t.randn(256, 32, device=dev) * 0.1).to(t.bfloat16).contiguous()
    bias_c = t.zeros(32, device=dev)
    for v in ("1", "3"):
        os.environ["MACHIN_CONV1_V"] = v
        results[f"conv1_fwd_v{v}_K3.3M_ms"] = timeit(
            lambda: ext.conv1_fwd(frames_c, wgt, bias_c, 1.0 / 255.0),
            iters=30,
        )
        results[f"conv1_wrw_v{v}_K3.3M_ms"] = timeit(
            lambda: ext.conv1_wrw(gy, frames_c, 1.0 / 255.0), iters=30
        )
    os.environ.pop("MACHIN_CONV1_V", None)

    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/kernel_bench.json", "w") as f:
        json.dump(results, f, indent=2)
    for k, v in results.items():
        print(f"{k:40s} {v:10.3f}")


if __name__ == "__main__":
    main()
