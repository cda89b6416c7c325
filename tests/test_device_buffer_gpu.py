"""GPU tests for the HBM-resident replay buffers."""
import pytest
import torch as t

pytestmark = pytest.mark.gpu

if not t.cuda.is_available():
    pytest.skip("no GPU", allow_module_level=True)

DEV = "cuda:0"
SPEC = {
    "state": ((4,), t.float32),
    "action": ((1,), t.long),
    "reward": ((), t.float32),
}


class TestDeviceReplayBuffer:
    def test_store_and_sample(self):
        from machin_amd.frame.buffers.device_buffer import DeviceReplayBuffer

        buf = DeviceReplayBuffer(100, SPEC, DEV)
        batch = {
            "state": t.rand(30, 4),
            "action": t.randint(0, 2, (30, 1)),
            "reward": t.rand(30),
        }
        pos = buf.store_batch(batch)
        assert pos.numel() == 30 and buf.size() == 30
        out = buf.sample_batch(16)
        assert out["state"].shape == (16, 4)
        assert out["state"].device.type == "cuda"

    def test_ring_wraparound(self):
        from machin_amd.frame.buffers.device_buffer import DeviceReplayBuffer

        buf = DeviceReplayBuffer(10, {"x": ((), t.float32)}, DEV)
        buf.store_batch({"x": t.arange(8, dtype=t.float32)})
        buf.store_batch({"x": t.arange(8, 14, dtype=t.float32)})
        assert buf.size() == 10
        # ring holds 4..13
        vals = set(buf.data["x"].cpu().tolist())
        assert vals == set(float(v) for v in range(4, 14))

    def test_spec_validation(self):
        from machin_amd.frame.buffers.device_buffer import DeviceReplayBuffer

        buf = DeviceReplayBuffer(10, SPEC, DEV)
        with pytest.raises(ValueError):
            buf.store_batch({"state": t.rand(5, 4)})


class TestDevicePrioritizedBuffer:
    def test_per_flow(self):
        from machin_amd.frame.buffers.device_buffer import (
            DevicePrioritizedBuffer,
        )

        buf = DevicePrioritizedBuffer(1000, SPEC, DEV)
        batch = {
            "state": t.rand(200, 4),
            "action": t.randint(0, 2, (200, 1)),
            "reward": t.rand(200),
        }
        buf.store_batch(batch)
        out, idx, w = buf.sample_batch(64)
        assert out["state"].shape == (64, 4)
        assert idx.shape == (64,) and w.shape == (64,)
        assert w.max().item() <= 1.0 + 1e-5
        buf.update_priority(t.rand(64, device=DEV) * 10, idx)
        assert buf.wt_tree.get_weight_sum() > 0

    def test_priority_bias(self):
        from machin_amd.frame.buffers.device_buffer import (
            DevicePrioritizedBuffer,
        )

        buf = DevicePrioritizedBuffer(
            64, {"x": ((), t.float32)}, DEV,
            beta_increment_per_sampling=0,
        )
        buf.store_batch({"x": t.arange(64, dtype=t.float32)})
        buf.update_priority(
            t.full((1,), 1000.0, device=DEV),
            t.zeros(1, dtype=t.long, device=DEV),
        )
        counts = t.zeros(64)
        for _ in range(30):
            _, idx, _ = buf.sample_batch(32)
            counts += t.bincount(idx.cpu(), minlength=64)
        assert counts[0] > counts[1:].sum()

    def test_throughput_shapes_atari(self):
        from machin_amd.frame.buffers.device_buffer import (
            DevicePrioritizedBuffer,
        )

        spec = {
            "state": ((4, 84, 84), t.uint8),
            "action": ((1,), t.long),
            "reward": ((), t.float32),
            "next_state": ((4, 84, 84), t.uint8),
            "terminal": ((), t.float32),
        }
        buf = DevicePrioritizedBuffer(100000, spec, DEV)
        n = 2048
        batch = {
            "state": t.randint(0, 256, (n, 4, 84, 84), dtype=t.uint8),
            "action": t.randint(0, 6, (n, 1)),
            "reward": t.rand(n),
            "next_state": t.randint(0, 256, (n, 4, 84, 84), dtype=t.uint8),
            "terminal": t.zeros(n),
        }
        buf.store_batch(batch)
        out, idx, w = buf.sample_batch(512)
        t.cuda.synchronize()
        assert out["state"].shape == (512, 4, 84, 84)
