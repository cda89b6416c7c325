import numpy as np
import pytest
import torch as t

from machin_amd.frame.buffers import PrioritizedBuffer, WeightTree

from test_buffer import make_episode


class TestWeightTree:
    def test_build_and_sum(self):
        tree = WeightTree(10)
        w = np.arange(1, 11, dtype=np.float64)
        tree.update_all_leaves(w)
        assert tree.get_weight_sum() == pytest.approx(55.0)
        assert tree.get_leaf_max() == pytest.approx(10.0)
        assert np.allclose(tree.get_leaf_all_weights(), w)

    def test_single_update(self):
        tree = WeightTree(5)
        for i in range(5):
            tree.update_leaf(float(i + 1), i)
        assert tree.get_weight_sum() == pytest.approx(15.0)
        tree.update_leaf(10.0, 0)
        assert tree.get_weight_sum() == pytest.approx(24.0)
        assert tree.get_leaf_weight(0) == pytest.approx(10.0)

    def test_batch_update(self):
        tree = WeightTree(100)
        idx = np.arange(100)
        w = np.random.uniform(0.1, 5.0, 100)
        tree.update_leaf_batch(w, idx)
        assert tree.get_weight_sum() == pytest.approx(w.sum())
        # partial update
        idx2 = np.array([3, 50, 99])
        w2 = np.array([7.0, 8.0, 9.0])
        tree.update_leaf_batch(w2, idx2)
        expect = w.copy()
        expect[idx2] = w2
        assert tree.get_weight_sum() == pytest.approx(expect.sum())
        assert np.allclose(tree.get_leaf_weight(idx2), w2)

    def test_batch_update_duplicates(self):
        tree = WeightTree(8)
        tree.update_leaf_batch(
            np.array([1.0, 2.0, 3.0]), np.array([0, 0, 1])
        )
        # last write wins for leaf 0
        assert tree.get_leaf_weight(0) == pytest.approx(2.0)
        assert tree.get_weight_sum() == pytest.approx(5.0)

    def test_find_leaf(self):
        tree = WeightTree(4)
        tree.update_all_leaves(np.array([1.0, 2.0, 3.0, 4.0]))
        # cumulative: [1, 3, 6, 10]
        assert tree.find_leaf_index(0.5) == 0
        assert tree.find_leaf_index(1.5) == 1
        assert tree.find_leaf_index(4.0) == 2
        assert tree.find_leaf_index(9.9) == 3
        idx = tree.find_leaf_index(np.array([0.5, 2.5, 6.5]))
        assert list(idx) == [0, 1, 3]

    def test_find_leaf_statistics(self):
        tree = WeightTree(64)
        w = np.random.uniform(0.0, 2.0, 64)
        tree.update_all_leaves(w)
        s = tree.get_weight_sum()
        queries = np.random.uniform(0, s, 20000)
        idx = tree.find_leaf_index(queries)
        counts = np.bincount(idx, minlength=64) / 20000
        assert np.abs(counts - w / s).max() < 0.02

    def test_non_power_of_two(self):
        for size in (1, 3, 7, 1000):
            tree = WeightTree(size)
            w = np.random.uniform(0.5, 1.5, size)
            tree.update_all_leaves(w)
            assert tree.get_weight_sum() == pytest.approx(w.sum())

    def test_errors(self):
        with pytest.raises(ValueError):
            WeightTree(0)
        tree = WeightTree(4)
        with pytest.raises(ValueError):
            tree.update_leaf(1.0, 10)
        with pytest.raises(ValueError):
            tree.get_leaf_weight(9)
        with pytest.raises(ValueError):
            tree.update_leaf_batch(np.array([1.0]), np.array([4]))


class TestPrioritizedBuffer:
    def test_store_and_sample(self):
        b = PrioritizedBuffer(buffer_size=100)
        b.store_episode(make_episode(20))
        bs, batch, index, is_weight = b.sample_batch(10)
        assert bs == 10
        assert batch[0]["state"].shape == (10, 4)
        assert len(index) == 10
        assert len(is_weight) == 10
        assert np.all(is_weight > 0) and np.all(is_weight <= 1.0 + 1e-9)

    def test_priority_update(self):
        b = PrioritizedBuffer(buffer_size=100)
        b.store_episode(make_episode(10))
        bs, batch, index, is_weight = b.sample_batch(5)
        before = b.wt_tree.get_weight_sum()
        b.update_priority(np.full(5, 100.0), index)
        assert b.wt_tree.get_weight_sum() > before

    def test_priority_bias(self):
        b = PrioritizedBuffer(buffer_size=100, beta_increment_per_sampling=0)
        b.store_episode(make_episode(10))
        # make transition 0 dominate
        b.update_priority(np.array([1000.0]), np.array([0]))
        counts = np.zeros(10)
        for _ in range(50):
            _, _, idx, _ = b.sample_batch(10)
            counts += np.bincount(idx, minlength=10)
        assert counts[0] > counts[1:].sum()

    def test_explicit_priorities(self):
        b = PrioritizedBuffer(buffer_size=100)
        b.store_episode(make_episode(3), priorities=[1.0, 2.0, 3.0])
        assert b.wt_tree.get_weight_sum() > 0

    def test_empty_sample(self):
        b = PrioritizedBuffer(buffer_size=100)
        assert b.sample_batch(4) == (0, None, None, None)

    def test_clear(self):
        b = PrioritizedBuffer(buffer_size=100)
        b.store_episode(make_episode(5))
        b.clear()
        assert b.size() == 0
        assert b.wt_tree.get_weight_sum() == 0

    def test_beta_annealing(self):
        b = PrioritizedBuffer(
            buffer_size=100, beta=0.4, beta_increment_per_sampling=0.1
        )
        b.store_episode(make_episode(10))
        for _ in range(10):
            b.sample_batch(2)
        assert b.curr_beta == pytest.approx(1.0)
