"""Shared-memory rollout ring: host actor farm -> GPU learner staging.

SURVEY.md §2.6-C / §5.8: the reference ships rollouts as pickled CPU
tensors over RPC (machin/frame/buffers/buffer_d.py:194-197). On
MI355X the actor->learner path is: actor processes write rollout
segments into preallocated SHARED-MEMORY slots (zero serialization),
only slot indices cross process boundaries (tiny pipe messages), and
the learner copies ready slots into pinned staging and issues ONE
async H2D copy per attribute onto a side stream.

Layout: for each attribute, one shared tensor of shape
``[slots, *per_slot_shape]``. Actors acquire a free slot, fill it,
mark it ready; the learner drains ready slots in batches and recycles
them.
"""
from typing import Dict, List, Tuple

import torch as t
import torch.multiprocessing as mp


class RolloutRing:
    """Multi-producer single-consumer ring of rollout segments."""

    def __init__(
        self,
        slots: int,
        spec: Dict[str, Tuple[Tuple[int, ...], t.dtype]],
        ctx=None,
    ):
        ctx = ctx or mp.get_context("spawn")
        self.slots = int(slots)
        self.spec = dict(spec)
        self.data = {
            k: t.zeros((self.slots, *shape), dtype=dtype).share_memory_()
            for k, (shape, dtype) in self.spec.items()
        }
        self._free = ctx.Queue()
        self._ready = ctx.Queue()
        for i in range(self.slots):
            self._free.put(i)

    # -- actor side ----------------------------------------------------
    def acquire(self, timeout: float = None) -> int:
        """Take a free slot index (blocks when the learner lags)."""
        return self._free.get(timeout=timeout)

    def slot(self, index: int) -> Dict[str, t.Tensor]:
        """Views of one slot's attribute tensors (write in place)."""
        return {k: v[index] for k, v in self.data.items()}

    def commit(self, index: int):
        """Mark a filled slot ready for the learner."""
        self._ready.put(index)

    # -- learner side --------------------------------------------------
    def drain(self, max_slots: int, timeout: float = 1.0) -> List[int]:
        """Collect up to ``max_slots`` ready slot indices; waits for at
        least one up to ``timeout``."""
        out = []
        try:
            out.append(self._ready.get(timeout=timeout))
        except Exception:  # noqa: BLE001 - queue.Empty
            return out
        while len(out) < max_slots:
            try:
                out.append(self._ready.get_nowait())
            except Exception:  # noqa: BLE001
                break
        return out

    def gather(
        self,
        indices: List[int],
        device: t.device,
        pinned: Dict[str, t.Tensor] = None,
        non_blocking: bool = True,
    ) -> Dict[str, t.Tensor]:
        """Copy the chosen slots to ``device`` (through a pinned
        staging buffer when provided) and return the batch."""
        idx = t.tensor(indices, dtype=t.long)
        out = {}
        for k, buf in self.data.items():
            host = buf.index_select(0, idx)
            if pinned is not None and k in pinned:
                staging = pinned[k][: len(indices)]
                staging.copy_(host)
                host = staging
            out[k] = host.to(device, non_blocking=non_blocking)
        return out

    def release(self, indices: List[int]):
        """Recycle consumed slots back to the actors."""
        for i in indices:
            self._free.put(i)

    def make_pinned_staging(self, max_batch: int) -> Dict[str, t.Tensor]:
        return {
            k: t.empty((max_batch, *shape), dtype=dtype).pin_memory()
            for k, (shape, dtype) in self.spec.items()
        }

    def host_register(self) -> bool:
        """Learner-side: hipHostRegister the shared-memory buffers so
        H2D copies DMA STRAIGHT from the ring slots (no host staging
        copy at all). Returns False when registration is unavailable
        (callers then use pinned staging via :meth:`gather`)."""
        try:
            cudart = t.cuda.cudart()
            for buf in self.data.values():
                rc = cudart.cudaHostRegister(
                    buf.data_ptr(), buf.numel() * buf.element_size(), 0
                )
                if int(rc) != 0:
                    return False
            return True
        except Exception:  # noqa: BLE001 - optional fast path
            return False

    def upload_slots(
        self,
        indices: List[int],
        pool: Dict[str, t.Tensor],
        positions: List[int],
        non_blocking: bool = True,
    ):
        """Direct DMA: copy each ready slot into its HBM pool position
        (requires :meth:`host_register`; issue on a side stream).
        Attributes missing from ``pool`` are skipped (callers may
        route some attributes through their own staging)."""
        for slot_id, pos in zip(indices, positions):
            for k, buf in self.data.items():
                if k in pool:
                    pool[k][pos].copy_(buf[slot_id],
                                       non_blocking=non_blocking)


class EpisodeSegmentCodec:
    """Maps algorithm-level episodes (lists of Transition/dicts) onto
    fixed-shape ring slots so IMPALA actors can store rollouts through
    shared memory instead of control-plane RPC.

    Slot layout (unroll T):
      * ``state/<k>``  [T, ...]   episode states (zero-padded)
      * ``action/<k>`` [T, ...]   taken actions
      * ``reward``     [T]        float32
      * ``terminal``   [T]        float32
      * ``behavior_logp`` [T]     float32 (required by V-trace)
      * ``boot_state/<k>`` [...]  next_state of the segment's last
                                  transition (bootstrap input)
      * ``length``     []         int64 valid steps

    Episodes longer than T are split into consecutive segments, each
    bootstrapping from its own tail — the standard IMPALA unroll.
    """

    def __init__(self, unroll: int):
        self.unroll = int(unroll)

    # -- spec ----------------------------------------------------------
    def spec_from_episode(self, episode) -> Dict[str, Tuple[Tuple[int, ...], t.dtype]]:
        tr = episode[0]
        if not isinstance(tr, dict):
            tr = {k: getattr(tr, k) for k in tr.keys()}
        T = self.unroll
        spec = {
            "reward": ((T,), t.float32),
            "terminal": ((T,), t.float32),
            "behavior_logp": ((T,), t.float32),
            "length": ((), t.long),
        }
        for k, v in tr["state"].items():
            spec[f"state/{k}"] = ((T, *v.shape[1:]), v.dtype)
        for k, v in tr["action"].items():
            spec[f"action/{k}"] = ((T, *v.shape[1:]), v.dtype)
        for k, v in tr["next_state"].items():
            spec[f"boot_state/{k}"] = (tuple(v.shape[1:]), v.dtype)
        return spec

    # -- actor side ----------------------------------------------------
    @staticmethod
    def _scalar(v):
        return float(v.reshape(-1)[0]) if t.is_tensor(v) else float(v)

    def write_episode(self, ring: "RolloutRing", episode,
                      timeout: float = None):
        """Split an episode into segments and write each into a free
        slot. Blocks when the ring is full (learner backpressure)."""
        T = self.unroll
        dicts = []
        for tr in episode:
            if not isinstance(tr, dict):
                tr = {k: getattr(tr, k) for k in tr.keys()}
            if "action_log_prob" not in tr:
                raise ValueError(
                    "IMPALA ring transitions require 'action_log_prob'."
                )
            dicts.append(tr)
        for start in range(0, len(dicts), T):
            chunk = dicts[start : start + T]
            L = len(chunk)
            slot_id = ring.acquire(timeout=timeout)
            slot = ring.slot(slot_id)
            for k in chunk[0]["state"]:
                col = slot[f"state/{k}"]
                col[:L] = t.cat([c["state"][k] for c in chunk], dim=0)
                if L < T:
                    col[L:] = 0
            for k in chunk[0]["action"]:
                col = slot[f"action/{k}"]
                col[:L] = t.cat([c["action"][k] for c in chunk], dim=0)
                if L < T:
                    col[L:] = 0
            for k in chunk[-1]["next_state"]:
                slot[f"boot_state/{k}"].copy_(
                    chunk[-1]["next_state"][k][0]
                )
            slot["reward"][:L] = t.as_tensor(
                [self._scalar(c["reward"]) for c in chunk]
            )
            slot["terminal"][:L] = t.as_tensor(
                [self._scalar(c["terminal"]) for c in chunk]
            )
            slot["behavior_logp"][:L] = t.as_tensor(
                [self._scalar(c["action_log_prob"]) for c in chunk]
            )
            if L < T:
                slot["reward"][L:] = 0
                slot["terminal"][L:] = 1.0
                slot["behavior_logp"][L:] = 0
            slot["length"].fill_(L)
            ring.commit(slot_id)


def make_episode_ring(sample_episode, unroll: int, slots: int,
                      ctx=None) -> Tuple["RolloutRing", EpisodeSegmentCodec]:
    """Build a RolloutRing sized for the given episode shape."""
    codec = EpisodeSegmentCodec(unroll)
    ring = RolloutRing(
        slots=slots, spec=codec.spec_from_episode(sample_episode),
        ctx=ctx,
    )
    return ring, codec
