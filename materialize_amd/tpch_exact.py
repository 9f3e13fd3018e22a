"""Exact restatement of the reference's TPC-H load generator.

Reproduces, draw for draw, `Tpch::by_seed` from
/root/reference/src/storage/src/source/generator/tpch.rs:41-331 over the
exact RNG stack the reference uses (rand 0.8 StdRng = ChaCha12):

- `StdRng::seed_from_u64` (rand_core's PCG32-based seed expansion),
- ChaCha12 keystream (rand_chacha 0.3: 64-bit block counter at state
  words 12-13, stream words 14-15 = 0, 12 rounds, little-endian words;
  rand_core::BlockRng consumption collapses to the plain sequential u32
  word stream — verified by the Q3 MD5 golden below),
- rand 0.8 `UniformInt` sampling: widening-multiply rejection with
  `zone = (range << range.leading_zeros()).wrapping_sub(1)`; u32-wide
  draws for i32-typed ranges, u64-wide for i64/usize,
- `Alphanumeric` (6-bit shift + rejection against 62),
- `SliceRandom::choose` (usize exclusive range) and
  `choose_multiple` = `index::sample` -> `sample_floyd` for
  (len=92, amount=5) (rand 0.8 seq/index.rs),
- the generator's row structure: supplier, part(+partsupp), customer,
  orders (outer rng draws one u64 seed per order; the order's content
  comes from a FRESH StdRng seeded with it — order_row, tpch.rs:262-348),
  and the churn protocol (retract an active order's lineitems+order,
  insert regenerated ones — tpch.rs:204-241).

Only the columns Q3/Q17 need are materialized; every other field's RNG
draws are consumed faithfully so the stream stays aligned. Pinned by the
reference's own end-to-end golden: the SF 0.01 snapshot's Q3 result
hashes (testdrive md5 discipline, md5 over concatenated stringified
values in result order) to 637be0ff3f50cd612b004a69958bfccb with 127
rows (/root/reference/test/testdrive/tpch.td:193-215).
"""
import math
from datetime import date as _date
from datetime import timedelta

import numpy as np

MASK32 = 0xFFFFFFFF
MASK64 = 0xFFFFFFFFFFFFFFFF

START_DATE = _date(1992, 1, 1)
CURRENT_DATE = _date(1995, 6, 17)
END_DATE = _date(1998, 12, 31)
ORDER_END_DAYS = (END_DATE - START_DATE).days - 151

NATION_COUNT = 25
SEGMENTS = ["AUTOMOBILE", "BUILDING", "FURNITURE", "MACHINERY", "HOUSEHOLD"]
PRIORITIES_LEN = 4
INSTRUCTIONS_LEN = 4
MODES_LEN = 7
PARTNAMES_LEN = 92
TYPES_LENS = [6, 5, 5]
CONTAINERS_LENS = [4, 7]
ALPHABET_LEN = 64


def _chacha12_blocks(key_words, counter0, nblocks):
    """ChaCha12 keystream blocks for counters counter0..+nblocks-1,
    returned as a flat uint32 word array (block-major, 16 words each)."""
    n = nblocks
    x = np.empty((16, n), np.uint32)
    const = (0x61707865, 0x3320646E, 0x79622D32, 0x6B206574)
    for i in range(4):
        x[i] = np.uint32(const[i])
    for i in range(8):
        x[4 + i] = np.uint32(key_words[i])
    ctr = np.arange(counter0, counter0 + n, dtype=np.uint64)
    x[12] = (ctr & np.uint64(0xFFFFFFFF)).astype(np.uint32)
    x[13] = (ctr >> np.uint64(32)).astype(np.uint32)
    x[14] = 0
    x[15] = 0
    init = x.copy()

    def rotl(v, s):
        return (v << np.uint32(s)) | (v >> np.uint32(32 - s))

    def qr(a, b, c, d):
        x[a] += x[b]
        x[d] = rotl(x[d] ^ x[a], 16)
        x[c] += x[d]
        x[b] = rotl(x[b] ^ x[c], 12)
        x[a] += x[b]
        x[d] = rotl(x[d] ^ x[a], 8)
        x[c] += x[d]
        x[b] = rotl(x[b] ^ x[c], 7)

    for _ in range(6):  # 12 rounds = 6 double rounds
        qr(0, 4, 8, 12)
        qr(1, 5, 9, 13)
        qr(2, 6, 10, 14)
        qr(3, 7, 11, 15)
        qr(0, 5, 10, 15)
        qr(1, 6, 11, 12)
        qr(2, 7, 8, 13)
        qr(3, 4, 9, 14)
    x += init
    return x.T.reshape(-1)  # block-major word order


class StdRng08:
    """rand 0.8 StdRng (ChaCha12) word stream + distribution layer."""

    CHUNK_BLOCKS = 4096  # words generated per refill (x16)

    def __init__(self, seed_u64):
        # rand_core SeedableRng::seed_from_u64 (PCG32 seed expansion)
        state = seed_u64 & MASK64
        MUL = 6364136223846793005
        INC = 11634580027462260723
        kw = []
        for _ in range(8):
            state = (state * MUL + INC) & MASK64
            xorshifted = (((state >> 18) ^ state) >> 27) & MASK32
            rot = state >> 59
            kw.append(((xorshifted >> rot)
                       | (xorshifted << ((32 - rot) & 31))) & MASK32)
        self.key = kw
        self.counter = 0
        self.buf = []  # python ints, fast sequential consumption
        self.pos = 0

    def _refill(self):
        w = _chacha12_blocks(self.key, self.counter, self.CHUNK_BLOCKS)
        self.counter += self.CHUNK_BLOCKS
        self.buf = w.tolist()
        self.pos = 0

    def next_u32(self):
        if self.pos >= len(self.buf):
            self._refill()
        v = self.buf[self.pos]
        self.pos += 1
        return v

    def next_u64(self):
        # BlockRng::next_u64 consumes two sequential words (lo, hi) in
        # every index case — see module docstring
        if self.pos + 1 >= len(self.buf):
            rem = self.buf[self.pos:]
            self._refill()
            if rem:
                lo = rem[0]
                hi = self.buf[0]
                self.pos = 1
                return lo | (hi << 32)
        lo = self.buf[self.pos]
        hi = self.buf[self.pos + 1]
        self.pos += 2
        return lo | (hi << 32)

    # ---- rand 0.8 UniformInt sampling ----
    def r32_incl(self, lo, hi):
        """gen_range(lo..=hi) for i32/u32-typed ranges (one u32/draw)."""
        rng = (hi - lo + 1) & MASK32
        if rng == 0:
            v = self.next_u32()
            return v - (1 << 32) if v >> 31 else v
        lz = 32 - rng.bit_length()
        zone = ((rng << lz) & MASK32) - 1
        while True:
            v = self.next_u32()
            prod = v * rng
            if (prod & MASK32) <= zone:
                return lo + (prod >> 32)

    def r64_incl(self, lo, hi):
        """gen_range(lo..=hi) for i64/u64/usize ranges (one u64/draw)."""
        rng = (hi - lo + 1) & MASK64
        if rng == 0:
            return self.next_u64()
        lz = 64 - rng.bit_length()
        zone = ((rng << lz) & MASK64) - 1
        while True:
            v = self.next_u64()
            prod = v * rng
            if (prod & MASK64) <= zone:
                return lo + (prod >> 64)

    def r64_excl(self, lo, hi):
        """gen_range(lo..hi) for usize/i64 (sample_single)."""
        rng = (hi - lo) & MASK64
        lz = 64 - rng.bit_length()
        zone = ((rng << lz) & MASK64) - 1
        while True:
            v = self.next_u64()
            prod = v * rng
            if (prod & MASK64) <= zone:
                return lo + (prod >> 64)

    def r32_excl(self, lo, hi):
        """gen_range(lo..hi) for u32-typed ranges (sample_single)."""
        rng = (hi - lo) & MASK32
        lz = 32 - rng.bit_length()
        zone = ((rng << lz) & MASK32) - 1
        while True:
            v = self.next_u32()
            prod = v * rng
            if (prod & MASK32) <= zone:
                return lo + (prod >> 32)

    def choose_idx(self, n):
        """SliceRandom::choose -> gen_index: u32-typed gen_range(0..n)
        whenever n fits u32 (rand 0.8 seq/mod.rs gen_index)."""
        return self.r32_excl(0, n)

    def floyd_5_of(self, length):
        """index::sample -> sample_floyd draws for (length, 5): five
        u32-typed inclusive draws gen_range(0..=j); indices discarded."""
        for j in range(length - 5, length):
            self.r32_incl(0, j)

    def alnum_consume(self, count):
        """Alphanumeric.sample_string(rng, count): per char, loop
        { v = next_u32() >> 26; accept if v < 62 } — values discarded."""
        accepted = 0
        while accepted < count:
            if self.pos >= len(self.buf):
                self._refill()
            # vectorized acceptance over the remaining chunk
            arr = np.asarray(self.buf[self.pos:], dtype=np.uint64)
            acc = (arr >> np.uint64(26)) < np.uint64(62)
            need = count - accepted
            csum = np.cumsum(acc)
            total = int(csum[-1]) if len(csum) else 0
            if total >= need:
                # position after the need-th acceptance
                idx = int(np.searchsorted(csum, need))
                self.pos += idx + 1
                accepted = count
            else:
                accepted += total
                self.pos = len(self.buf)

    # ---- generator helpers (tpch.rs:368-445) ----
    def d_decimal(self, lo, hi):
        """decimal(): returns the raw integer n of n/div (i64 range)."""
        return self.r64_incl(lo, hi)

    def d_date(self, start, dlo, dhi):
        return start + timedelta(days=self.r32_incl(dlo, dhi))

    def d_vstring(self, mn, mx):
        take = self.r64_incl(mn, mx)
        for _ in range(take):
            self.choose_idx(ALPHABET_LEN)

    def d_phone(self):
        self.r32_incl(100, 999)
        self.r32_incl(100, 999)
        self.r32_incl(1000, 9999)

    def d_text(self, mn, mx, source_len):
        self.r64_incl(0, source_len - mx)
        self.r64_incl(mn, mx)


def order_key(i):
    """mk_sparse (tpch.rs:386-397)."""
    low = i & 7
    return ((i >> 3) << 5) | low


def partkey_retailprice(key):
    """Integer dollars (i64 division, tpch.rs:351-355)."""
    return (90000 + ((key // 10) % 20001) + 100 * (key % 1000)) // 100


class TpchExact:
    """The reference generator, exactly. Yields the Q3/Q17-relevant
    columns of the snapshot, plus churn batches with identical RNG
    consumption."""

    TEXT_LEN = 3 << 20

    def __init__(self, sf=0.01, seed=0):
        f_to_i = lambda m: max(int(math.floor(sf * m)), 1)
        self.count_supplier = f_to_i(10_000.0)
        self.count_part = f_to_i(200_000.0)
        self.count_customer = f_to_i(150_000.0)
        self.count_orders = f_to_i(1_500_000.0)
        self.count_clerk = f_to_i(1_000.0)
        self.rng = StdRng08(seed)
        # Context init: text_string_source = Alphanumeric 3<<20 chars
        self.rng.alnum_consume(self.TEXT_LEN)
        self.active_orders = []  # (key, seed) in insertion order

    # ---- row generators (draw-faithful) ----
    def _supplier_row(self):
        r = self.rng
        r.r64_excl(0, NATION_COUNT)       # nation
        r.d_vstring(10, 40)               # address
        r.d_phone()
        r.d_decimal(-999_99, 9_999_99)    # acctbal
        r.d_text(25, 100, self.TEXT_LEN)

    def _part_row(self):
        r = self.rng
        r.floyd_5_of(PARTNAMES_LEN)       # name
        m = r.r32_incl(1, 5)
        n = r.r32_incl(1, 5)
        for _ in range(4):                # partsupp rows
            r.r64_incl(0, 3)              # suppkey term
            r.r32_incl(1, 9_999)          # availqty
            r.d_decimal(1_00, 1_000_00)   # supplycost
            r.d_text(49, 198, self.TEXT_LEN)
        for ln in TYPES_LENS:             # type syllables
            r.choose_idx(ln)
        r.r32_incl(1, 50)                 # size
        c1 = r.choose_idx(CONTAINERS_LENS[0])
        c2 = r.choose_idx(CONTAINERS_LENS[1])
        r.d_text(49, 198, self.TEXT_LEN)
        # engine encodings: brand code = 10*m+n ('Brand#23' = 23);
        # container code = 10*c1+c2 ('MED BOX' = 10)
        return m * 10 + n, c1 * 10 + c2

    def _customer_row(self, key):
        r = self.rng
        r.r64_excl(0, NATION_COUNT)       # nation
        r.d_vstring(10, 40)               # address
        r.d_phone()
        r.d_decimal(-999_99, 9_999_99)    # acctbal
        seg = r.choose_idx(len(SEGMENTS))
        r.d_text(29, 116, self.TEXT_LEN)
        return (key, SEGMENTS[seg])

    def order_row(self, seed, key):
        """tpch.rs:262-348: the order and its lineitems from a fresh
        StdRng(seed). Returns (orderkey, custkey, orderdate,
        [(partkey, quantity, extprice_dollars, discount_hundredths,
          shipdate), ...])."""
        r = StdRng08(seed)
        okey = order_key(key)
        while True:
            custkey = r.r64_incl(1, self.count_customer)
            if custkey % 3 != 0:
                break
        orderdate = r.d_date(START_DATE, 1, ORDER_END_DAYS)
        lineitem_count = r.r64_incl(1, 7)  # usize
        lines = []
        for _ in range(lineitem_count):
            partkey = r.r64_incl(1, self.count_part)
            r.r64_incl(0, 3)               # suppkey term
            quantity = r.r32_incl(1, 50)
            discount = r.d_decimal(0, 8)   # /100
            tax = r.d_decimal(0, 10)       # /100
            shipdate = r.d_date(orderdate, 1, 121)
            receiptdate = r.d_date(shipdate, 1, 30)
            # packer draw order: returnflag (only when receipted),
            # commitdate, instructions, modes, comment
            retflag = 2  # "N"
            if receiptdate <= CURRENT_DATE:
                retflag = r.choose_idx(2)  # returnflag ("R","A")
            commitdate = r.d_date(orderdate, 30, 90)
            r.choose_idx(INSTRUCTIONS_LEN)
            mode = r.choose_idx(MODES_LEN)
            r.d_text(10, 43, self.TEXT_LEN)
            ep = quantity * partkey_retailprice(partkey)
            lines.append((partkey, quantity, ep, discount, shipdate,
                          commitdate, receiptdate, mode, retflag, tax))
        prio = r.choose_idx(PRIORITIES_LEN)  # orderpriority
        r.r64_incl(1, self.count_clerk)      # clerk
        r.d_text(19, 78, self.TEXT_LEN)
        return okey, custkey, orderdate, lines, prio

    def snapshot(self):
        """Generate the full snapshot (stream order: suppliers, parts,
        customers, orders; nation/region draws follow but are after all
        Q3 data and are skipped). Returns (customers, orders, lineitems):
        customers = [(custkey, segment)], orders = [(orderkey, custkey,
        orderdate)], lineitems = [(orderkey, ep_dollars, discount,
        shipdate, partkey, quantity)]."""
        for _ in range(self.count_supplier):
            self._supplier_row()
        self.parts = []  # (partkey, brand_code, container_code)
        for key in range(1, self.count_part + 1):
            b, c = self._part_row()
            self.parts.append((key, b, c))
        customers = []
        for key in range(1, self.count_customer + 1):
            customers.append(self._customer_row(key))
        orders = []
        lineitems = []
        for key in range(1, self.count_orders + 1):
            seed = self.rng.next_u64()
            okey, custkey, odate, lines, prio = self.order_row(seed, key)
            orders.append((okey, custkey, odate, prio))
            for ln in lines:
                lineitems.append((okey,) + ln)
            self.active_orders.append((key, seed))
        return customers, orders, lineitems

    def churn_batch(self):
        """One churn tick (tpch.rs:204-241): retract a random active
        order's lineitems+order, insert regenerated ones. Returns
        (retract_order, retract_lines, insert_order, insert_lines) in the
        reference's shapes."""
        r = self.rng
        idx = r.r64_excl(0, len(self.active_orders))
        key, old_seed = self.active_orders[idx]
        # Vec::swap_remove
        last = self.active_orders.pop()
        if idx < len(self.active_orders):
            self.active_orders[idx] = last
        okey, ck_o, od_o, old_lines, prio_o = self.order_row(old_seed, key)
        new_seed = r.next_u64()
        _, ck_n, od_n, new_lines, prio_n = self.order_row(new_seed, key)
        self.active_orders.append((key, new_seed))
        return ((okey, ck_o, od_o, prio_o), [(okey,) + ln for ln in old_lines],
                (okey, ck_n, od_n, prio_n), [(okey,) + ln for ln in new_lines])


CUTOFF = _date(1995, 3, 15)


def q3_result(customers, orders, lineitems):
    """Q3 over the snapshot, exactly as the reference computes and
    renders it: revenue = sum(ep * (1 - d)) through the Accum::Numeric
    pipeline, whose value ends fully decNumber-`reduce`d (trailing zeros
    stripped — reduce.rs:2192); rows are rendered to strings
    (l_orderkey, revenue standard notation, o_orderdate, "0") and then
    LEXICOGRAPHICALLY SORTED AS STRING ROWS — testdrive sorts decoded
    rows before hashing (`actual.sort()`, testdrive sql.rs:291), so the
    query's ORDER BY does not affect the hash. Verified: this exact
    discipline reproduces the reference's pinned Q3/Q6/Q12 MD5s."""
    building = {ck for ck, seg in customers if seg == "BUILDING"}
    odate = {}
    for (okey, ck, od, _prio) in orders:
        if ck in building and od < CUTOFF:
            odate[okey] = od
    groups = {}  # okey -> sum in 1e-2 dollars
    for (okey, _pk, _q, ep, d, sd, *_rest) in lineitems:
        if sd <= CUTOFF:
            continue
        od = odate.get(okey)
        if od is None:
            continue
        groups[okey] = groups.get(okey, 0) + ep * (100 - d)
    rows = []
    for okey, r2 in groups.items():
        rows.append([str(okey), render_revenue_1e2(r2), str(odate[okey]),
                     "0"])
    rows.sort()
    return rows


def render_dec_reduced(units, scale):
    """decNumber cx.reduce rendering of an exact value `units` * 10^-scale
    (non-negative): trailing fractional zeros stripped, integer when the
    fraction vanishes (the Numeric sum rendering, reduce.rs:2192)."""
    s = 10 ** scale
    whole, frac = divmod(units, s)
    if frac == 0:
        return str(whole)
    fs = str(frac).rjust(scale, "0").rstrip("0")
    return f"{whole}.{fs}"


def render_revenue_1e2(r2):
    """decNumber-reduced standard notation of r2 (integer 1e-2 dollars)."""
    intp, frac = divmod(r2, 100)
    if frac == 0:
        return str(intp)
    if frac % 10 == 0:
        return f"{intp}.{frac // 10}"
    return f"{intp}.{frac:02d}"


def q3_md5(rows):
    """testdrive's hash discipline: md5 over the concatenated stringified
    values of every (string-sorted) row (sql.rs:374-379)."""
    import hashlib
    h = hashlib.md5()
    for row in rows:
        for e in row:
            h.update(e.encode())
    return h.hexdigest()


# --------------------------------------------------------------- adapter

def _days(d):
    return (d - START_DATE).days


def _seg_code(seg):
    """Engine mktsegment encoding: BUILDING = 0 (the Q3 closures filter
    mkt == 0); any distinct nonzero code for the rest."""
    return 0 if seg == "BUILDING" else SEGMENTS.index(seg) + 10


class ExactEngineData:
    """Adapts the exact snapshot/churn to the engine's column formats
    (materialize_amd.tpch encodings: dates int32 days since 1992-01-01,
    extendedprice int64 cents, discount int64 basis points, BUILDING=0).
    Duck-types the TpchGen surface Q3Dataflow.load/churn consume."""

    def __init__(self, gen, customers, orders, lineitems, parts=None):
        self._gen = gen
        self.parts = parts if parts is not None else getattr(gen, "parts",
                                                             [])
        self.n_customer = len(customers)
        self.c_custkey = np.array([c[0] for c in customers], np.int64)
        self.c_mktsegment = np.array([_seg_code(c[1]) for c in customers],
                                     np.int64)
        self.n_orders = len(orders)
        self.o_orderkey = np.array([t[0] for t in orders], np.int64)
        self.o_custkey = np.array([t[1] for t in orders], np.int64)
        self.o_orderdate = np.array([_days(t[2]) for t in orders], np.int32)
        self.o_shippriority = np.zeros(self.n_orders, np.int32)
        self._pack_lineitems(lineitems)

    # ---- TpchGen surface used by Q3Dataflow.load ----
    def customer_updates(self):
        return self.c_custkey, self.c_mktsegment.reshape(-1, 1)

    def orders_vals(self, idx):
        n = len(idx)
        v = np.zeros((n, 16), np.uint8)
        v[:, 0:8] = self.o_custkey[idx].view(np.uint8).reshape(n, 8)
        v[:, 8:12] = self.o_orderdate[idx].view(np.uint8).reshape(n, 4)
        v[:, 12:16] = self.o_shippriority[idx].view(np.uint8).reshape(n, 4)
        return v

    def orders_bycust_vals(self, idx):
        n = len(idx)
        v = np.zeros((n, 16), np.uint8)
        v[:, 0:8] = self.o_orderkey[idx].view(np.uint8).reshape(n, 8)
        v[:, 8:12] = self.o_orderdate[idx].view(np.uint8).reshape(n, 4)
        v[:, 12:16] = self.o_shippriority[idx].view(np.uint8).reshape(n, 4)
        return v

    @staticmethod
    def lineitem_vals(extprice, discount, shipdate):
        n = len(extprice)
        v = np.zeros((n, 24), np.uint8)
        v[:, 0:8] = extprice.view(np.uint8).reshape(n, 8)
        v[:, 8:16] = discount.view(np.uint8).reshape(n, 8)
        v[:, 16:20] = shipdate.view(np.uint8).reshape(n, 4)
        return v

    def lineitem_updates(self):
        return self.l_orderkey, self.lineitem_vals(
            self.l_extendedprice, self.l_discount, self.l_shipdate)

    # ---- Q17 surfaces (lineitem by partkey, part) ----
    def _pack_lineitems(self, lineitems):
        self.l_orderkey = np.array([t[0] for t in lineitems], np.int64)
        self.l_partkey = np.array([t[1] for t in lineitems], np.int64)
        self.l_quantity = np.array([t[2] for t in lineitems], np.int64)
        self.l_extendedprice = np.array([t[3] * 100 for t in lineitems],
                                        np.int64)  # cents
        self.l_discount = np.array([t[4] * 100 for t in lineitems],
                                   np.int64)       # basis points
        self.l_shipdate = np.array([_days(t[5]) for t in lineitems],
                                   np.int32)

    @staticmethod
    def lineitem_bypart_vals(quantity, extprice):
        n = len(quantity)
        v = np.zeros((n, 16), np.uint8)
        v[:, 0:8] = quantity.view(np.uint8).reshape(n, 8)
        v[:, 8:16] = extprice.view(np.uint8).reshape(n, 8)
        return v

    def lineitem_bypart_updates(self):
        return self.l_partkey, self.lineitem_bypart_vals(
            self.l_quantity, self.l_extendedprice)

    def part_updates(self):
        n = len(self.parts)
        keys = np.array([p[0] for p in self.parts], np.int64)
        v = np.zeros((n, 16), np.uint8)
        v[:, 0:8] = np.array([p[1] for p in self.parts], np.int64) \
            .view(np.uint8).reshape(n, 8)
        v[:, 8:16] = np.array([p[2] for p in self.parts], np.int64) \
            .view(np.uint8).reshape(n, 8)
        return keys, v

    @staticmethod
    def churn_to_engine(batch):
        """Map one exact churn batch (old_order, old_lines, new_order,
        new_lines) to the engine churn dict shape (TpchGen.churn):
        {"lineitem": (keys, vals, diffs), "orders": ..., "orders_by_cust":
        ...} with retractions before insertions."""
        (okey, ck_o, od_o, _p_o), old_lines,             (_, ck_n, od_n, _p_n), new_lines = batch

        def pack_lines(lines, diff):
            n = len(lines)
            keys = np.array([t[0] for t in lines], np.int64)
            ep = np.array([t[3] * 100 for t in lines], np.int64)
            d = np.array([t[4] * 100 for t in lines], np.int64)
            sd = np.array([_days(t[5]) for t in lines], np.int32)
            return keys, ExactEngineData.lineitem_vals(ep, d, sd),                 np.full(n, diff, np.int64)

        def pack_bypart(lines, diff):
            n = len(lines)
            pk = np.array([t[1] for t in lines], np.int64)
            q = np.array([t[2] for t in lines], np.int64)
            ep = np.array([t[3] * 100 for t in lines], np.int64)
            return (pk, ExactEngineData.lineitem_bypart_vals(q, ep),
                    np.full(n, diff, np.int64))

        lk_o, lv_o, ld_o = pack_lines(old_lines, -1)
        lk_n, lv_n, ld_n = pack_lines(new_lines, 1)
        l_keys = np.concatenate([lk_o, lk_n])
        l_vals = np.concatenate([lv_o, lv_n])
        l_diffs = np.concatenate([ld_o, ld_n])
        bp_o = pack_bypart(old_lines, -1)
        bp_n = pack_bypart(new_lines, 1)

        def pack_order(ck, od, diff, by_cust):
            v = np.zeros((1, 16), np.uint8)
            first = np.array([okey if by_cust else ck], np.int64)
            v[:, 0:8] = first.view(np.uint8).reshape(1, 8)
            v[:, 8:12] = np.array([_days(od)], np.int32)                 .view(np.uint8).reshape(1, 4)
            key = np.array([ck if by_cust else okey], np.int64)
            return key, v, np.array([diff], np.int64)

        ok_o, ov_o, od_do = pack_order(ck_o, od_o, -1, False)
        ok_n, ov_n, od_dn = pack_order(ck_n, od_n, 1, False)
        ck_ko, cv_o, cd_o = pack_order(ck_o, od_o, -1, True)
        ck_kn, cv_n, cd_n = pack_order(ck_n, od_n, 1, True)
        return {
            "lineitem": (l_keys, l_vals, l_diffs),
            "lineitem_by_part": (np.concatenate([bp_o[0], bp_n[0]]),
                                 np.concatenate([bp_o[1], bp_n[1]]),
                                 np.concatenate([bp_o[2], bp_n[2]])),
            "orders": (np.concatenate([ok_o, ok_n]),
                       np.concatenate([ov_o, ov_n]),
                       np.concatenate([od_do, od_dn])),
            "orders_by_cust": (np.concatenate([ck_ko, ck_kn]),
                               np.concatenate([cv_o, cv_n]),
                               np.concatenate([cd_o, cd_n])),
        }


def q17_avg_yearly(parts, lineitems):
    """Reference Q17 over the current state, exactly: per selected part
    (Brand#23, MED BOX), lineitems with quantity < 0.2 * avg(quantity)
    contribute extendedprice; result = sum / 7.0 rendered in decNumber
    39-digit standard notation (cx_datum semantics emulated with Python
    decimal, the same General Decimal Arithmetic spec). Returns the
    rendered string, or None for an empty result (SQL NULL)."""
    from collections import defaultdict
    from decimal import ROUND_HALF_EVEN, Decimal, localcontext
    sel = {pk for (pk, b, cc) in parts if b == 23 and cc == 10}
    with localcontext() as ctx:
        ctx.prec = 39
        ctx.rounding = ROUND_HALF_EVEN
        bypk = defaultdict(list)
        for t in lineitems:
            if t[1] in bypk or t[1] in sel:
                bypk[t[1]].append(t)
        total = Decimal(0)
        nrows = 0
        for pk in sel:
            lines = bypk.get(pk, [])
            if not lines:
                continue
            avg = Decimal(sum(t[2] for t in lines)) / Decimal(len(lines))
            thr = Decimal("0.2") * avg
            for t in lines:
                if Decimal(t[2]) < thr:
                    total += Decimal(t[3])
                    nrows += 1
        if nrows == 0:
            return None
        return format(total / Decimal("7.0"), "f")
