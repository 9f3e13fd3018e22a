"""materialize_amd.dist — the Exchange over RCCL/xGMI (or gloo on CPU).

Replaces timely's hash-partitioned Exchange pacts
(src/compute/src/render/join/linear_join.rs:390,
 src/compute/src/extensions/arrange.rs:134): every re-keying stage routes
update rows by shard = route_hash(key) % world to the owning rank as one
all-to-all-v per stage per batch. One process per GPU;
backend "nccl" IS RCCL on ROCm; tests run the same code under gloo on CPU.

Rows travel packed as fixed-stride byte records (key words ‖ val ‖ time ‖
diff) in one `all_to_all_single` with split sizes — the columnar
`(key,val,time,diff)` chunk format of the reference's columnar_exchange
(timely-util/src/columnar.rs:276), flattened.
"""
import numpy as np


def shard_of(keys, kw, world):
    """Vectorized splitmix64 route hash (matches mz_gpu_route_hash /
    orc_route_hash exactly) -> shard ids."""
    keys = keys.view(np.uint64).reshape(-1, kw)
    h = np.full(len(keys), 0x9E3779B97F4A7C15, np.uint64)
    with np.errstate(over="ignore"):
        for w in range(kw):
            x = keys[:, w] + h
            x ^= x >> np.uint64(30)
            x *= np.uint64(0xBF58476D1CE4E5B9)
            x ^= x >> np.uint64(27)
            x *= np.uint64(0x94D049BB133111EB)
            x ^= x >> np.uint64(31)
            h = x
    return (h % np.uint64(world)).astype(np.int64)


def pack_records(keys, vals, times, diffs, kw, vb):
    """Pack columns into fixed-stride byte records."""
    n = len(times)
    stride = 8 * kw + vb + 16
    rec = np.zeros((n, stride), np.uint8)
    rec[:, :8 * kw] = keys.view(np.uint8).reshape(n, 8 * kw)
    if vb:
        rec[:, 8 * kw:8 * kw + vb] = vals.reshape(n, vb)
    rec[:, 8 * kw + vb:8 * kw + vb + 8] = \
        times.view(np.uint8).reshape(n, 8)
    rec[:, 8 * kw + vb + 8:] = diffs.view(np.uint8).reshape(n, 8)
    return rec


def unpack_records(rec, kw, vb):
    n = len(rec)
    keys = rec[:, :8 * kw].copy().view(np.int64).reshape(-1)
    vals = rec[:, 8 * kw:8 * kw + vb].copy() if vb else \
        np.empty(0, np.uint8)
    times = rec[:, 8 * kw + vb:8 * kw + vb + 8].copy().view(
        np.uint64).reshape(-1)
    diffs = rec[:, 8 * kw + vb + 8:].copy().view(np.int64).reshape(-1)
    return keys, vals, times, diffs


class LocalExchange:
    """world_size 1: identity."""
    world = 1
    rank = 0

    def exchange(self, keys, vals, times, diffs, kw, vb):
        return keys, vals, times, diffs


class TorchExchange:
    """all-to-all-v over torch.distributed (RCCL on GPU, gloo on CPU)."""

    def __init__(self, device="cpu"):
        import torch.distributed as dist
        self.dist = dist
        self.world = dist.get_world_size()
        self.rank = dist.get_rank()
        self.device = device

    def exchange_dev(self, gctx, upd, kw, vb, t):
        """Device-resident all-to-all-v: shard-partition on the GPU
        (mz_gpu_partition — splitmix64 routing identical to the host
        path), then column-wise torch.distributed all_to_all_single.
        On nccl (RCCL over xGMI) the columns never leave HBM; on gloo
        they bounce through pinned host copies (the CPU-validation
        transport). Returns a device Updates at [t, t+1)."""
        import torch

        from . import _abi as abi
        dist = self.dist
        W = self.world
        n = upd.n
        work_dev = torch.device("cuda", torch.cuda.current_device())
        kt = torch.empty(max(n, 1) * kw, dtype=torch.int64,
                         device=work_dev)
        vt = (torch.empty(max(n * vb, 1), dtype=torch.uint8,
                          device=work_dev) if vb else None)
        tt = torch.empty(max(n, 1), dtype=torch.int64, device=work_dev)
        dt = torch.empty(max(n, 1), dtype=torch.int64, device=work_dev)
        counts = gctx.partition_dev(abi.schema(kw, vb), upd, W,
                                    (kt, vt, tt, dt))
        comm_cpu = self.device == "cpu"
        cnt_dev = "cpu" if comm_cpu else work_dev
        cnt_t = torch.tensor(counts, dtype=torch.int64, device=cnt_dev)
        recv_cnt = torch.zeros(W, dtype=torch.int64, device=cnt_dev)
        dist.all_to_all_single(recv_cnt, cnt_t)
        rc = [int(x) for x in recv_cnt.cpu()]
        outs = []
        for ten, mult in ((kt, kw), (vt, vb), (tt, 1), (dt, 1)):
            if mult == 0 or ten is None:
                outs.append(None)
                continue
            send = ten[:n * mult]
            if comm_cpu:
                send = send.cpu()
            recv = torch.empty(sum(rc) * mult, dtype=ten.dtype,
                               device=send.device)
            dist.all_to_all_single(recv, send,
                                   [c * mult for c in rc],
                                   [int(c) * mult for c in counts])
            outs.append(recv.to(work_dev) if comm_cpu else recv)
        ko, vo, to_, do_ = outs
        return abi.make_updates_from_torch(ko, vo, to_, do_, t, t + 1)

    def exchange(self, keys, vals, times, diffs, kw, vb):
        import torch
        dist = self.dist
        W = self.world
        n = len(times)
        stride = 8 * kw + vb + 16
        shards = shard_of(keys, kw, W) if n else np.empty(0, np.int64)
        order = np.argsort(shards, kind="stable")
        counts = np.bincount(shards, minlength=W)
        rec = pack_records(
            keys.view(np.int64).reshape(n, kw)[order].reshape(-1),
            vals.reshape(n, vb)[order] if vb else vals,
            times[order], diffs[order], kw, vb)
        send = torch.from_numpy(rec.reshape(-1))
        if self.device != "cpu":
            send = send.to(self.device)
        in_splits = [int(c) * stride for c in counts]
        # exchange counts first
        cnt_t = torch.tensor(counts, dtype=torch.int64)
        recv_cnt = torch.zeros(W, dtype=torch.int64)
        if self.device != "cpu":
            cnt_t = cnt_t.to(self.device)
            recv_cnt = recv_cnt.to(self.device)
        dist.all_to_all_single(recv_cnt, cnt_t)
        recv_counts = [int(x) for x in recv_cnt.cpu()]
        out_splits = [c * stride for c in recv_counts]
        recv = torch.zeros(sum(out_splits), dtype=torch.uint8,
                           device=send.device)
        dist.all_to_all_single(recv, send, out_splits, in_splits)
        rec_in = recv.cpu().numpy().reshape(-1, stride) if sum(recv_counts) \
            else np.zeros((0, stride), np.uint8)
        return unpack_records(rec_in, kw, vb)
