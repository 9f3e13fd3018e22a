"""Columnar pinned-buffer ingest feeder.

The boundary the reference crosses at `persist_source` (storage →
compute: `src/storage-operators/src/persist_source.rs`, consumed at
`src/compute/src/render.rs:290-296`) is, on this engine, a host→HBM
hand-off of columnar update batches. The feeder owns a ring of
page-locked (pinned) column buffers and a dedicated copy stream:

- `stage(...)` packs one batch's columns into the next ring slot and
  issues the H2D copies asynchronously on the copy stream; it returns a
  device `Updates` descriptor whose `ready()` blocks until the copies
  land (the engine runs on its own HIP stream, so the hand-off point is
  a host-side event wait — batch t+1's PCIe transfer overlaps batch t's
  compute, which is the part that pays).
- Slot lifetime contract: a staged batch's DEVICE buffers stay owned by
  the feeder slot until `mark_consumed()` is called for it. Synchronous
  engine pushes (`arr_insert`, `join_push`, ...) consume on return;
  `arr_insert_async` consumes only at `arr_flush` (mz_gpu.h lifetime
  note) — call `mark_consumed()` at that point. Re-using a slot whose
  batch was never marked raises rather than silently overwriting
  buffers a pending lane insert may still read.

PCIe-inclusive throughput is a different number from the HBM-resident
bench (`bench.py` stages everything up front per the contract); the
measured hand-off rate is reported in DESIGN.md §7.

CPU fallback (no CUDA device / no pinned allocator): plain host-memory
descriptors (`on_device=0` — the engine stages them itself), same API,
used by the CPU tests.
"""
import numpy as np

from . import _abi as abi


class PinnedFeeder:
    def __init__(self, schema, capacity_rows, device=None, depth=2):
        self.kw = schema.key_words
        self.vb = schema.val_bytes
        self.cap = int(capacity_rows)
        self.depth = depth
        self.slot = 0
        self.device = device
        self.gpu = False
        if device is not None:
            try:
                import torch
                self.gpu = torch.cuda.is_available()
            except Exception:
                self.gpu = False
        if self.gpu:
            import torch
            self.torch = torch
            self.copy_stream = torch.cuda.Stream(device=device)
            self.slots = []
            for _ in range(depth):
                mk = lambda n, dt: torch.empty(n, dtype=dt, pin_memory=True)
                host = {
                    "keys": mk(self.cap * self.kw, torch.int64),
                    "vals": mk(max(self.cap * self.vb, 1), torch.uint8),
                    "times": mk(self.cap, torch.int64),
                    "diffs": mk(self.cap, torch.int64),
                }
                dev = {k: torch.empty_like(v, device=device)
                       for k, v in host.items()}
                self.slots.append({"host": host, "dev": dev,
                                   "event": torch.cuda.Event(),
                                   "inflight": False})
        self._pending = []  # slot indices staged but not yet consumed

    def stage(self, keys, vals, times, diffs, lower, upper):
        """Stage one columnar batch; returns (updates, ready) where
        `ready()` must be called before pushing `updates` into the
        engine (it waits for the H2D copies of THIS batch only)."""
        n = len(times)
        assert n <= self.cap, "batch exceeds feeder capacity"
        keys = np.ascontiguousarray(keys, np.int64).reshape(-1)
        vals = (np.ascontiguousarray(vals, np.uint8).reshape(-1)
                if self.vb else None)
        times = np.ascontiguousarray(times, np.int64).reshape(-1)
        diffs = np.ascontiguousarray(diffs, np.int64).reshape(-1)
        if not self.gpu:
            u = abi.make_updates(keys, vals,
                                 times.view(np.uint64), diffs, lower, upper)
            return u, (lambda: None)
        torch = self.torch
        idx = self.slot
        s = self.slots[idx]
        if s["inflight"]:
            raise RuntimeError(
                "PinnedFeeder slot re-use before mark_consumed(): the "
                "engine may still read this slot's device buffers "
                "(async inserts hold them until arr_flush — see the "
                "lifetime contract in the module docstring)")
        self.slot = (self.slot + 1) % self.depth
        s["inflight"] = True
        self._pending.append(idx)
        # the slot's previous transfer must have landed before re-packing
        s["event"].synchronize()
        h, d = s["host"], s["dev"]
        h["keys"][:n * self.kw] = torch.from_numpy(keys)
        if self.vb:
            h["vals"][:n * self.vb] = torch.from_numpy(vals)
        h["times"][:n] = torch.from_numpy(times)
        h["diffs"][:n] = torch.from_numpy(diffs)
        with torch.cuda.stream(self.copy_stream):
            d["keys"][:n * self.kw].copy_(h["keys"][:n * self.kw],
                                          non_blocking=True)
            if self.vb:
                d["vals"][:n * self.vb].copy_(h["vals"][:n * self.vb],
                                              non_blocking=True)
            d["times"][:n].copy_(h["times"][:n], non_blocking=True)
            d["diffs"][:n].copy_(h["diffs"][:n], non_blocking=True)
            s["event"].record(self.copy_stream)
        u = abi.make_updates_from_torch(
            d["keys"][:n * self.kw],
            d["vals"][:n * self.vb] if self.vb else None,
            d["times"][:n], d["diffs"][:n], lower, upper)
        return u, s["event"].synchronize

    def mark_consumed(self):
        """The engine has consumed the OLDEST outstanding staged batch:
        after a synchronous push returns, or after `arr_flush` for
        batches handed to `arr_insert_async`. Frees that slot for
        re-use."""
        if not self.gpu:
            return
        if not self._pending:
            raise RuntimeError("mark_consumed() with no staged batch")
        self.slots[self._pending.pop(0)]["inflight"] = False
