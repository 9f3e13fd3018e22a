"""In-process exchange for thread-parallel CPU runs (the baseline leg).

W worker threads each own an engine context and a shard; the exchange is
a shared mailbox + barrier (the thread analog of timely's in-process
channels, communication.rs:169-184). The oracle's C calls release the
GIL, so shards run in parallel on host cores.
"""
import threading

import numpy as np

from .dist import shard_of


class ThreadExchangeGroup:
    def __init__(self, world):
        self.world = world
        self.barrier = threading.Barrier(world)
        self.mailbox = [[None] * world for _ in range(world)]

    def member(self, rank):
        return ThreadExchange(self, rank)


class ThreadExchange:
    def __init__(self, group, rank):
        self.group = group
        self.world = group.world
        self.rank = rank

    def exchange(self, keys, vals, times, diffs, kw, vb):
        W = self.world
        n = len(times)
        keys = np.ascontiguousarray(keys, np.int64)
        vals = np.asarray(vals, np.uint8).reshape(n, vb) if vb else \
            np.zeros((n, 0), np.uint8)
        shards = shard_of(keys, kw, W) if n else np.empty(0, np.int64)
        km = keys.view(np.int64).reshape(n, kw)
        for dst in range(W):
            m = shards == dst
            self.group.mailbox[dst][self.rank] = (
                km[m].copy(), vals[m].copy(), times[m].copy(),
                diffs[m].copy())
        self.group.barrier.wait()
        parts = self.group.mailbox[self.rank]
        out = (np.concatenate([p[0] for p in parts]).reshape(-1),
               np.concatenate([p[1] for p in parts]).reshape(-1),
               np.concatenate([p[2] for p in parts]),
               np.concatenate([p[3] for p in parts]))
        self.group.barrier.wait()  # all read before next round overwrites
        return out
