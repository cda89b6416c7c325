"""Pool dispatch micro-benchmark (CPU): the reference claims P2PPool
has a "50% speed advantage" over regular pools for small tasks
(machin docs/source/tutorials/parallel_distributed.rst:60). Measures
small-task round-trip throughput of this repo's Pool (lock-based
shared queues), P2PPool (lock-free per-worker pipes) and ThreadPool.
"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def _tiny(x):
    return x + 1


def bench(pool, n_tasks=2000):
    # warm
    pool.map(_tiny, range(32))
    t0 = time.perf_counter()
    out = pool.map(_tiny, range(n_tasks), chunksize=1)
    dt = time.perf_counter() - t0
    assert out[:3] == [1, 2, 3]
    return n_tasks / dt


def main():
    from machin_amd.parallel.pool import P2PPool, Pool, ThreadPool

    results = {}
    for name, cls in (("pool_lock_based", Pool),
                      ("p2p_pool", P2PPool),
                      ("thread_pool", ThreadPool)):
        p = cls(processes=4)
        try:
            results[f"{name}_tasks_per_s"] = bench(p)
        finally:
            p.terminate()
    results["p2p_vs_lock_speedup"] = (
        results["p2p_pool_tasks_per_s"]
        / results["pool_lock_based_tasks_per_s"]
    )
    print(json.dumps(results, indent=2))
    return results


if __name__ == "__main__":
    main()
