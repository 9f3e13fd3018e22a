"""Seeded TPC-H-shaped data generator (synthetic, numpy).

Shapes follow the reference's TPCH load generator
(src/storage/src/source/generator/tpch.rs): scale mapping per
src/sql/src/plan/statement/ddl.rs:2140-2144 (customer 150k*sf, orders
1.5M*sf, lineitem 1-7 per order, tpch.rs:274), and the churn step
retracts an active order's lineitems + the order and inserts regenerated
ones under the same order key (tpch.rs:204-241).

Encodings (DESIGN.md §2.3 — exact integers for fixed-scale decimals):
  dates         int32 days since 1992-01-01
  extendedprice int64 cents  (= quantity * retailprice_cents)
  discount      int64 basis points (0..800 = 0.00..0.08)
  mktsegment    int64 code (0 = BUILDING, tpch.rs SEGMENTS)
Revenue l_extendedprice*(1-l_discount) = cents*(10000-bp), exact i64
1e-6-dollar units.
"""
import numpy as np

CUTOFF_19950315 = 1169  # days from 1992-01-01 to 1995-03-15
SEGMENTS = 5  # BUILDING, AUTOMOBILE, MACHINERY, HOUSEHOLD, FURNITURE


class TpchGen:
    def __init__(self, sf=1.0, seed=42):
        self.sf = sf
        self.rng = np.random.default_rng(seed)
        self.n_customer = int(150_000 * sf)
        self.n_orders = int(1_500_000 * sf)
        self.n_part = int(200_000 * sf)
        # base tables
        rng = self.rng
        self.c_custkey = np.arange(1, self.n_customer + 1, dtype=np.int64)
        self.c_mktsegment = rng.integers(0, SEGMENTS, self.n_customer
                                         ).astype(np.int64)
        self.o_orderkey = np.arange(1, self.n_orders + 1, dtype=np.int64)
        # custkey % 3 != 0 rule (tpch.rs:265-270)
        ck = rng.integers(1, max(self.n_customer, 2), self.n_orders
                          ).astype(np.int64)
        ck += (ck % 3 == 0)
        ck = np.minimum(ck, max(self.n_customer, 1))
        self.o_custkey = ck
        self.o_orderdate = rng.integers(1, 2252, self.n_orders
                                        ).astype(np.int32)
        self.o_shippriority = np.zeros(self.n_orders, np.int32)
        # lineitems: 1..7 per order (tpch.rs:274)
        self.l_count = rng.integers(1, 8, self.n_orders).astype(np.int32)
        self._gen_lineitems()

    def _lineitem_cols(self, order_idx, rng):
        """Generate lineitem columns for the given order indices (one row
        per entry of order_idx): extprice, discount, shipdate, partkey,
        quantity."""
        n = len(order_idx)
        quantity = rng.integers(1, 51, n).astype(np.int64)
        retail_cents = rng.integers(90_000, 200_001, n).astype(np.int64)
        extprice = quantity * retail_cents
        discount_bp = rng.integers(0, 9, n).astype(np.int64) * 100
        shipdate = (self.o_orderdate[order_idx].astype(np.int64) +
                    rng.integers(1, 122, n)).astype(np.int32)
        partkey = rng.integers(1, max(self.n_part, 2), n).astype(np.int64)
        return extprice, discount_bp, shipdate, partkey, quantity

    def _gen_part(self):
        """part table (Q17): brand code 10*m+n (Brand#23 = 23),
        container code 10*c1+c2 (MED BOX = 10) — the exact generator's
        encodings; synthetic draws cover the same ranges with ~1/1000
        joint selectivity like the reference's predicates."""
        rng = self.rng
        self.p_partkey = np.arange(1, self.n_part + 1, dtype=np.int64)
        self.p_brand = rng.integers(0, 25, self.n_part).astype(np.int64)
        self.p_container = rng.integers(0, 40, self.n_part).astype(np.int64)

    def _gen_lineitems(self):
        self._gen_part()
        order_idx = np.repeat(np.arange(self.n_orders), self.l_count)
        self.l_order_idx = order_idx
        self.l_orderkey = self.o_orderkey[order_idx]
        (self.l_extendedprice, self.l_discount, self.l_shipdate,
         self.l_partkey, self.l_quantity) = \
            self._lineitem_cols(order_idx, self.rng)
        # per-order start offsets into lineitem arrays
        self.l_offs = np.zeros(self.n_orders + 1, np.int64)
        np.cumsum(self.l_count, out=self.l_offs[1:])

    # ------------------------------------------------------------ views

    def customer_updates(self):
        """(key custkey, val [mktsegment i64]) insertions."""
        return self.c_custkey, self.c_mktsegment.reshape(-1, 1)

    def orders_vals(self, idx):
        """orders val layout: [o_custkey i64][o_orderdate i32][o_shippriority
        i32] = 16B."""
        n = len(idx)
        v = np.zeros((n, 16), np.uint8)
        v[:, 0:8] = self.o_custkey[idx].view(np.uint8).reshape(n, 8)
        v[:, 8:12] = self.o_orderdate[idx].view(np.uint8).reshape(n, 4)
        v[:, 12:16] = self.o_shippriority[idx].view(np.uint8).reshape(n, 4)
        return v

    def orders_bycust_vals(self, idx):
        """orders-by-custkey val: [o_orderkey i64][date i32][prio i32]."""
        n = len(idx)
        v = np.zeros((n, 16), np.uint8)
        v[:, 0:8] = self.o_orderkey[idx].view(np.uint8).reshape(n, 8)
        v[:, 8:12] = self.o_orderdate[idx].view(np.uint8).reshape(n, 4)
        v[:, 12:16] = self.o_shippriority[idx].view(np.uint8).reshape(n, 4)
        return v

    @staticmethod
    def lineitem_vals(extprice, discount, shipdate):
        """lineitem val (Q3, by orderkey): [extprice i64][discount i64]
        [shipdate i32][pad] = 24B."""
        n = len(extprice)
        v = np.zeros((n, 24), np.uint8)
        v[:, 0:8] = extprice.view(np.uint8).reshape(n, 8)
        v[:, 8:16] = discount.view(np.uint8).reshape(n, 8)
        v[:, 16:20] = shipdate.view(np.uint8).reshape(n, 4)
        return v

    def lineitem_updates(self):
        return self.l_orderkey, self.lineitem_vals(
            self.l_extendedprice, self.l_discount, self.l_shipdate)

    @staticmethod
    def lineitem_bypart_vals(quantity, extprice):
        """lineitem val (Q17, by partkey): [quantity i64][extprice i64]."""
        n = len(quantity)
        v = np.zeros((n, 16), np.uint8)
        v[:, 0:8] = quantity.view(np.uint8).reshape(n, 8)
        v[:, 8:16] = extprice.view(np.uint8).reshape(n, 8)
        return v

    def lineitem_bypart_updates(self):
        return self.l_partkey, self.lineitem_bypart_vals(
            self.l_quantity, self.l_extendedprice)

    def part_updates(self):
        """part val: [brand i64][container i64]."""
        n = self.n_part
        v = np.zeros((n, 16), np.uint8)
        v[:, 0:8] = self.p_brand.view(np.uint8).reshape(n, 8)
        v[:, 8:16] = self.p_container.view(np.uint8).reshape(n, 8)
        return self.p_partkey, v

    # ------------------------------------------------------------ churn

    def churn(self, target_rows):
        """One churn batch (tpch.rs:204-241 shape): pick k active orders;
        retract each order row + its lineitems; insert a regenerated order
        (same orderkey, new custkey/date) + new lineitems. Returns dicts of
        per-relation (keys, vals, diffs) update columns, totalling
        ~target_rows rows. Mutates the generator's state arrays.
        """
        rng = self.rng
        # expected rows per churned order: 2 (order) + 2 * avg lines (~4)
        k = max(1, int(target_rows / 10))
        idx = rng.choice(self.n_orders, k, replace=False)
        # flat positions of the chosen orders' lineitems (vectorized gather)
        counts = self.l_count[idx].astype(np.int64)
        starts = self.l_offs[idx]
        cum = np.concatenate([[0], np.cumsum(counts)])
        pos = np.repeat(starts - cum[:-1], counts) + np.arange(cum[-1])
        # --- retractions of current state
        r_l_keys = self.l_orderkey[pos].copy()
        r_l_vals = self.lineitem_vals(self.l_extendedprice[pos].copy(),
                                      self.l_discount[pos].copy(),
                                      self.l_shipdate[pos].copy())
        r_lp_keys = self.l_partkey[pos].copy()
        r_lp_vals = self.lineitem_bypart_vals(
            self.l_quantity[pos].copy(), self.l_extendedprice[pos].copy())
        o_retract_vals = self.orders_vals(idx)
        o_retract_bycust_vals = self.orders_bycust_vals(idx)
        o_retract_bycust_keys = self.o_custkey[idx].copy()
        # --- regenerate the orders in place
        ck = rng.integers(1, max(self.n_customer, 2), k).astype(np.int64)
        ck += (ck % 3 == 0)
        ck = np.minimum(ck, max(self.n_customer, 1))
        self.o_custkey[idx] = ck
        self.o_orderdate[idx] = rng.integers(1, 2252, k).astype(np.int32)
        # regenerate the lineitems in place (same count per order — keeps
        # the flat arrays stable; the reference redraws 1..7, a shape
        # detail that does not change the maintained row rate)
        order_idx_rep = np.repeat(idx, counts)
        ep, disc, sd, pk, qty = self._lineitem_cols(order_idx_rep, rng)
        self.l_extendedprice[pos] = ep
        self.l_discount[pos] = disc
        self.l_shipdate[pos] = sd
        self.l_partkey[pos] = pk
        self.l_quantity[pos] = qty
        n_keys = self.l_orderkey[pos].copy()
        n_vals = self.lineitem_vals(ep, disc, sd)
        l_keys = np.concatenate([r_l_keys, n_keys])
        l_vals = np.concatenate([r_l_vals, n_vals])
        nr = len(r_l_keys)
        l_diffs = np.concatenate([-np.ones(nr, np.int64),
                                  np.ones(len(l_keys) - nr, np.int64)])
        lp_keys = np.concatenate([r_lp_keys, pk])
        lp_vals = np.concatenate([r_lp_vals,
                                  self.lineitem_bypart_vals(qty, ep)])
        lp_diffs = l_diffs.copy()
        o_keys = np.concatenate([self.o_orderkey[idx], self.o_orderkey[idx]])
        o_vals = np.concatenate([o_retract_vals, self.orders_vals(idx)])
        o_diffs = np.concatenate([-np.ones(k, np.int64),
                                  np.ones(k, np.int64)])
        oc_keys = np.concatenate([o_retract_bycust_keys, self.o_custkey[idx]])
        oc_vals = np.concatenate([o_retract_bycust_vals,
                                  self.orders_bycust_vals(idx)])
        return {
            "lineitem": (l_keys, l_vals, l_diffs),
            "lineitem_by_part": (lp_keys, lp_vals, lp_diffs),
            "orders": (o_keys, o_vals, o_diffs),
            "orders_by_cust": (oc_keys, oc_vals, o_diffs.copy()),
        }

    def churn_customers(self, k):
        """Retract k customers and re-insert them with fresh segments
        (exercises the customer -> orders -> lineitem delta path under
        retractions; the reference's loadgen keeps customers static, so
        this is engine-coverage churn, not a reference-protocol shape).
        Mutates c_mktsegment. Returns (keys, vals, diffs)."""
        rng = self.rng
        idx = rng.choice(self.n_customer, k, replace=False)
        keys = np.concatenate([self.c_custkey[idx], self.c_custkey[idx]])
        old_vals = self.c_mktsegment[idx].copy()
        self.c_mktsegment[idx] = rng.integers(0, SEGMENTS, k)
        vals = np.concatenate([old_vals,
                               self.c_mktsegment[idx]]).reshape(-1, 1)
        diffs = np.concatenate([-np.ones(k, np.int64),
                                np.ones(k, np.int64)])
        return keys, vals, diffs
