import torch as t

from machin_amd.frame.buffers import Buffer

import pytest


def make_episode(length, reward_base=0.0):
    return [
        {
            "state": {"state": t.full((1, 4), float(i))},
            "action": {"action": t.full((1, 1), float(i))},
            "next_state": {"state": t.full((1, 4), float(i + 1))},
            "reward": reward_base + i,
            "terminal": i == length - 1,
        }
        for i in range(length)
    ]


class TestBuffer:
    def test_store_and_size(self):
        b = Buffer(buffer_size=100)
        b.store_episode(make_episode(5))
        assert b.size() == 5
        b.clear()
        assert b.size() == 0

    def test_empty_episode_raises(self):
        b = Buffer(buffer_size=10)
        with pytest.raises(ValueError):
            b.store_episode([])

    def test_missing_attr_raises(self):
        b = Buffer(buffer_size=10)
        with pytest.raises(ValueError):
            b.store_episode([{"state": {"state": t.zeros(1, 4)}}])

    def test_ring_eviction_whole_episode(self):
        b = Buffer(buffer_size=6)
        b.store_episode(make_episode(4))   # ep0 at 0-3
        b.store_episode(make_episode(2))   # ep1 at 4-5 -> full
        b.store_episode(make_episode(3))   # overwrites 0,1,2 -> evicts ep0
        assert b.size() == 6
        assert len(b.episode_transition_handles) == 2

    def test_sample_random_unique(self):
        b = Buffer(buffer_size=100)
        b.store_episode(make_episode(10))
        bs, batch = b.sample_batch(5, sample_method="random_unique")
        assert bs == 5
        state, action, next_state, reward, terminal = batch[:5]
        assert state["state"].shape == (5, 4)
        assert reward.shape == (5, 1)
        assert terminal.shape == (5, 1)

    def test_sample_more_than_available(self):
        b = Buffer(buffer_size=100)
        b.store_episode(make_episode(3))
        bs, batch = b.sample_batch(10, sample_method="random_unique")
        assert bs == 3
        bs, batch = b.sample_batch(10, sample_method="random")
        assert bs == 10
        assert batch[0]["state"].shape == (10, 4)

    def test_sample_all(self):
        b = Buffer(buffer_size=100)
        b.store_episode(make_episode(4))
        bs, batch = b.sample_batch(-1, sample_method="all")
        assert bs == 4

    def test_sample_empty(self):
        b = Buffer(buffer_size=100)
        bs, batch = b.sample_batch(4)
        assert bs == 0 and batch is None

    def test_sample_attrs_and_wildcard(self):
        b = Buffer(buffer_size=100)
        eps = make_episode(4)
        for e in eps:
            e["note"] = "n"
            e["idx_val"] = 3
        b.store_episode(eps)
        bs, batch = b.sample_batch(
            2, sample_attrs=["state", "reward", "*"]
        )
        assert bs == 2
        state, reward, extras = batch
        assert set(extras.keys()) == {"note", "idx_val"}
        assert extras["note"] == ["n", "n"]

    def test_additional_concat_custom(self):
        b = Buffer(buffer_size=100)
        eps = make_episode(4)
        for e in eps:
            e["weight"] = 0.5
        b.store_episode(eps)
        bs, batch = b.sample_batch(
            3,
            sample_attrs=["reward", "weight"],
            additional_concat_custom_attrs=["weight"],
        )
        reward, weight = batch
        assert t.is_tensor(weight) and weight.shape == (3, 1)

    def test_no_concatenate(self):
        b = Buffer(buffer_size=100)
        b.store_episode(make_episode(4))
        bs, batch = b.sample_batch(2, concatenate=False, sample_attrs=["reward"])
        assert isinstance(batch[0], list) and len(batch[0]) == 2

    def test_unknown_attr_raises(self):
        b = Buffer(buffer_size=100)
        b.store_episode(make_episode(4))
        with pytest.raises(RuntimeError):
            b.sample_batch(2, sample_attrs=["nonexistent"])

    def test_callable_sample_method(self):
        b = Buffer(buffer_size=100)
        b.store_episode(make_episode(4))

        def take_first_two(buffer, batch_size):
            return 2, [buffer.storage[0], buffer.storage[1]]

        bs, batch = b.sample_batch(99, sample_method=take_first_two)
        assert bs == 2
        # first two states are 0 and 1
        assert batch[0]["state"][0, 0].item() == 0.0
        assert batch[0]["state"][1, 0].item() == 1.0

    def test_storage_isolated_from_source(self):
        b = Buffer(buffer_size=10)
        ep = make_episode(1)
        b.store_episode(ep)
        ep_tensor = ep[0]["state"]["state"]
        ep_tensor += 100
        assert b.storage[0].state["state"][0, 0].item() == 0.0
