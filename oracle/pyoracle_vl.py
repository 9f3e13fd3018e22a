"""Varlen-val oracle — TEST INFRASTRUCTURE ONLY (category (b), like
oracle.cpp). A pure-Python restatement of the reference semantics for
variable-length vals (the byte-arena row layout of
/root/reference/src/row-spine/src/lib.rs:110-135): canonical order is
(key i64-tuple, val bytes lexicographic with shorter-prefix-first,
time); consolidation sums diffs of equal (key, val, time) and drops
zeros (ColInternalMerger::merge, timely-util/src/columnation.rs:653-713);
the half-join probe matches per key with le/lt time tie-breaks
(delta_join.rs:356-399) and passes the matched varlen val through.
Only tests may import this module."""
import numpy as np


def _rows_from(keys, kw, arena, offs, times, diffs):
    n = len(times)
    out = []
    for i in range(n):
        k = tuple(int(np.int64(np.uint64(keys[i * kw + w])))
                  for w in range(kw))
        v = bytes(arena[offs[i]:offs[i + 1]])
        out.append((k, v, int(times[i]), int(diffs[i])))
    return out


def consolidate_rows(rows):
    agg = {}
    for (k, v, t, d) in rows:
        key = (k, v, t)
        agg[key] = agg.get(key, 0) + d
    out = [(k, v, t, d) for ((k, v, t), d) in agg.items() if d != 0]
    out.sort(key=lambda r: (r[0], r[1], r[2]))
    return out


def rows_to_cols(rows, kw):
    keys = np.zeros(len(rows) * kw, np.int64)
    offs = np.zeros(len(rows) + 1, np.uint32)
    arena = bytearray()
    times = np.zeros(len(rows), np.uint64)
    diffs = np.zeros(len(rows), np.int64)
    for i, (k, v, t, d) in enumerate(rows):
        for w in range(kw):
            keys[i * kw + w] = k[w]
        offs[i] = len(arena)
        arena.extend(v)
        times[i] = t
        diffs[i] = d
    offs[len(rows)] = len(arena)
    return (keys, np.frombuffer(bytes(arena), np.uint8).copy()
            if arena else np.empty(0, np.uint8), offs, times, diffs)


class VlOracle:
    """Mirrors the varlen slice of the GpuCtx interface."""

    def __init__(self):
        self.arrs = []

    def arr_create(self, kw):
        self.arrs.append({"kw": kw, "rows": []})
        return len(self.arrs) - 1

    def arr_insert(self, arr, keys, kw, arena, offs, times, diffs):
        rows = _rows_from(keys, kw, arena, offs, times, diffs)
        a = self.arrs[arr]
        a["rows"] = consolidate_rows(a["rows"] + rows)

    def consolidate(self, keys, kw, arena, offs, times, diffs):
        return rows_to_cols(
            consolidate_rows(_rows_from(keys, kw, arena, offs, times,
                                        diffs)), kw)

    def set_logical_compaction(self, arr, frontier):
        a = self.arrs[arr]
        rows = [(k, v, max(t, frontier), d) for (k, v, t, d) in a["rows"]]
        a["rows"] = consolidate_rows(rows)

    def halfjoin(self, arr, skeys, kw, stimes, sdiffs, le,
                 out_key_fn=None):
        """Probe: per stream update (key, t, d1), every arrangement row
        with the same key and t2 le/lt t emits (key, val, t, d1*d2);
        output consolidated. out_key_fn maps the key tuple (identity
        default)."""
        a = self.arrs[arr]
        bykey = {}
        for (k, v, t2, d2) in a["rows"]:
            bykey.setdefault(k, []).append((v, t2, d2))
        out = []
        n = len(stimes)
        for i in range(n):
            k = tuple(int(np.int64(np.uint64(skeys[i * kw + w])))
                      for w in range(kw))
            t = int(stimes[i])
            d1 = int(sdiffs[i])
            for (v, t2, d2) in bykey.get(k, []):
                if (t2 <= t) if le else (t2 < t):
                    ok = out_key_fn(k) if out_key_fn else k
                    d = (d1 * d2 + 2**63) % 2**64 - 2**63  # wrapping i64
                    out.append((ok, v, t, d))
        return rows_to_cols(consolidate_rows(out), kw)
