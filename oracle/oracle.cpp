// oracle/oracle.cpp — CPU restatement of Materialize's incremental
// join/reduce hot path. *** TEST INFRASTRUCTURE ONLY ***
//
// This library is the parity oracle: only tests/, __graft_entry__.smoke()
// and bench.py's cpu_baseline leg may load it. The product path
// (materialize_amd/) never links or imports it.
//
// Restates, line-for-line in behavior (citations are file:line under
// /root/reference):
//  - linear join:  src/compute/src/render/join/mz_join_core.rs
//      per-key merge scan (:644-663), EditList load+consolidation
//      (:857-878), simple cross-product strategy (:755-767, chosen when
//      either side has <10 edits, :743), linear time scan (:770-834),
//      output consolidation (:604).
//  - delta join:   src/compute/src/render/join/delta_join.rs
//      half-join probe with le/lt time tie-break by relation order
//      (:356-399); output time = the stream tuple's promoted data-time.
//  - reduce:       src/compute/src/render/reduce.rs
//      accumulable reduce (:1357-1581), Accum semigroup with wrapping
//      i128 adds (:2102-2203), multiply by diff (:2205-2266), fixed-point
//      float sums with 24 fractional bits (:1641-1697), finalize_accum
//      (:1840-1997); corrections = new minus old output per changed key
//      (reduce_abelian contract, src/compute/src/extensions/reduce.rs:131).
//  - consolidation: DD consolidate_updates as used at mz_join_core.rs:604 —
//      sort by (data, time), sum diffs (wrapping i64), drop zeros; ordering
//      matches ColInternalMerger::merge
//      (src/timely-util/src/columnation.rs:653-713).
//  - arrangement:  sorted immutable columnar batches + merge with logical
//      compaction (times advance to the frontier so add/retract pairs
//      cancel) — RowRowSpine blueprint, src/row-spine/src/lib.rs:56-135.
//
// Diff = Overflowing<i64> wraps in release builds
// (src/ore/src/overflowing.rs:24-31): all diff arithmetic here is wrapping.
//
// Parity pinning: tests/golden/ fixtures extracted from the reference's own
// test/sqllogictest/{joins,aggregates}.slt literal results (see
// tests/golden/README.md). The DD third-party boundary (batch/cursor
// internals) is unpinned in-repo and covered by property
// tests instead.

#include "../include/mz_gpu.h"

#include <algorithm>
#include <cstdint>
#include <cstring>
#include <cstdio>
#include <cmath>
#include <map>
#include <memory>
#include <string>
#include <vector>

using u8 = uint8_t;
using u32 = uint32_t;
using u64 = uint64_t;
using i64 = int64_t;
using i128 = __int128;
using u128 = unsigned __int128;
static constexpr int MAX_KW = 2;

namespace {

// ------------------------------------------------------------------ rows

struct Schema {
  u32 kw;  // key words
  u32 vb;  // val bytes
};

// SoA update columns (mirrors mz_gpu_updates host layout).
struct Cols {
  std::vector<u64> keys;  // kw * n
  std::vector<u8> vals;   // vb * n
  std::vector<u64> times;
  std::vector<i64> diffs;
  size_t size() const { return times.size(); }
  void push(const u64 *k, u32 kw, const u8 *v, u32 vb, u64 t, i64 d) {
    keys.insert(keys.end(), k, k + kw);
    if (vb) vals.insert(vals.end(), v, v + vb);
    times.push_back(t);
    diffs.push_back(d);
  }
};

// Key order: i64-tuple ascending. Val order: the val bytes
// as a tuple of zero-padded little-endian u64 words, unsigned ascending —
// the engine's canonical order (an arbitrary-but-fixed total order in the
// role of the reference's Row byte-lex order; identical on oracle and GPU).
inline int cmp_key(const u64 *a, const u64 *b, u32 kw) {
  for (u32 i = 0; i < kw; i++) {
    i64 x = (i64)a[i], y = (i64)b[i];
    if (x < y) return -1;
    if (x > y) return 1;
  }
  return 0;
}
inline int cmp_val(const u8 *a, const u8 *b, u32 vb) {
  for (u32 off = 0; off < vb; off += 8) {
    u64 x = 0, y = 0;
    u32 m = vb - off < 8 ? vb - off : 8;
    std::memcpy(&x, a + off, m);
    std::memcpy(&y, b + off, m);
    if (x < y) return -1;
    if (x > y) return 1;
  }
  return 0;
}

// consolidate_updates semantics: sort by (key, val, time), sum diffs
// (wrapping), drop zeros. Cf. mz_join_core.rs:604;
// timely-util/src/columnation.rs:686-694 (equal keys sum via plus_equals,
// zeros dropped).
void consolidate(const Schema &s, Cols &c) {
  size_t n = c.size();
  if (n == 0) return;
  std::vector<u32> idx(n);
  for (size_t i = 0; i < n; i++) idx[i] = (u32)i;
  const u64 *K = c.keys.data();
  const u8 *V = c.vals.data();
  const u64 *T = c.times.data();
  u32 kw = s.kw, vb = s.vb;
  std::sort(idx.begin(), idx.end(), [&](u32 a, u32 b) {
    int ck = cmp_key(K + (size_t)a * kw, K + (size_t)b * kw, kw);
    if (ck) return ck < 0;
    int cv = cmp_val(V + (size_t)a * vb, V + (size_t)b * vb, vb);
    if (cv) return cv < 0;
    if (T[a] != T[b]) return T[a] < T[b];
    return a < b;  // stable
  });
  Cols out;
  out.keys.reserve(c.keys.size());
  out.vals.reserve(c.vals.size());
  out.times.reserve(n);
  out.diffs.reserve(n);
  size_t i = 0;
  while (i < n) {
    size_t j = i;
    i64 d = 0;
    while (j < n &&
           cmp_key(K + (size_t)idx[i] * kw, K + (size_t)idx[j] * kw, kw) == 0 &&
           cmp_val(V + (size_t)idx[i] * vb, V + (size_t)idx[j] * vb, vb) == 0 &&
           T[idx[i]] == T[idx[j]]) {
      d = (i64)((u64)d + (u64)c.diffs[idx[j]]);  // wrapping add
      j++;
    }
    if (d != 0)
      out.push(K + (size_t)idx[i] * kw, kw, V + (size_t)idx[i] * vb, vb,
               T[idx[i]], d);
    i = j;
  }
  c = std::move(out);
}

// ------------------------------------------------------------ arrangement

// A sealed batch: updates sorted by (key,val,time), consolidated.
struct Batch {
  Cols cols;
  u64 lower, upper;
  // key index: start offset of each distinct key run (plus end sentinel)
  std::vector<size_t> key_starts;
  void build_index(const Schema &s) {
    key_starts.clear();
    size_t n = cols.size();
    const u64 *K = cols.keys.data();
    for (size_t i = 0; i < n; i++)
      if (i == 0 || cmp_key(K + (i - 1) * s.kw, K + i * s.kw, s.kw) != 0)
        key_starts.push_back(i);
    key_starts.push_back(n);
  }
  // binary search for key; returns [lo,hi) update range or empty
  std::pair<size_t, size_t> seek(const Schema &s, const u64 *key) const {
    if (key_starts.size() <= 1) return {0, 0};
    size_t lo = 0, hi = key_starts.size() - 1;  // distinct-key count
    const u64 *K = cols.keys.data();
    while (lo < hi) {
      size_t mid = (lo + hi) / 2;
      if (cmp_key(K + key_starts[mid] * s.kw, key, s.kw) < 0)
        lo = mid + 1;
      else
        hi = mid;
    }
    if (lo == key_starts.size() - 1) return {0, 0};
    if (cmp_key(K + key_starts[lo] * s.kw, key, s.kw) != 0) return {0, 0};
    return {key_starts[lo], key_starts[lo + 1]};
  }
};

struct Arr {
  Schema schema;
  std::vector<std::unique_ptr<Batch>> batches;
  u64 logical_compaction = 0;  // times advance to this on merge
  u64 upper = 0;               // acknowledged frontier = max pushed upper

  // Merge batches [from, to) into one, advancing times to the logical
  // compaction frontier (add/retract pairs then cancel) — Spine merge +
  // logical compaction semantics (mz_join_core.rs:458-465; manager.rs:54).
  void merge_span(size_t from, size_t to) {
    Cols all;
    u64 lo = UINT64_MAX, hi = 0;
    for (size_t bi = from; bi < to; bi++) {
      auto &b = batches[bi];
      lo = std::min(lo, b->lower);
      hi = std::max(hi, b->upper);
      size_t n = b->cols.size();
      for (size_t i = 0; i < n; i++) {
        u64 t = std::max(b->cols.times[i], logical_compaction);
        all.push(b->cols.keys.data() + i * schema.kw, schema.kw,
                 b->cols.vals.data() + i * schema.vb, schema.vb, t,
                 b->cols.diffs[i]);
      }
    }
    consolidate(schema, all);
    auto nb = std::make_unique<Batch>();
    nb->cols = std::move(all);
    nb->lower = lo == UINT64_MAX ? 0 : lo;
    nb->upper = hi;
    nb->build_index(schema);
    batches.erase(batches.begin() + from, batches.begin() + to);
    batches.insert(batches.begin() + from, std::move(nb));
  }
  void merge_all() { merge_span(0, batches.size()); }
};

// ------------------------------------------------------------- closures

// JoinClosure restatement (plan/join.rs:60-86): filters then field map.
// For the linear join, VAL_STREAM = input 1's val and VAL_LOOKUP = input 2's
// val regardless of which side the delta arrived on (result_fn(key,v1,v2),
// mz_join_core.rs:69). For the half join, VAL_STREAM = the stream tuple's
// val, VAL_LOOKUP = the arrangement's val (delta_join.rs:511-528).
inline i64 read_int(const u8 *p, u8 width) {
  if (width == 4) {
    int32_t v;
    std::memcpy(&v, p, 4);
    return v;
  }
  i64 v;
  std::memcpy(&v, p, 8);
  return v;
}

inline const u8 *cl_src(const u64 *key, const u8 *v1, const u8 *v2, u8 src) {
  switch (src) {
    case MZ_SRC_KEY: return (const u8 *)key;
    case MZ_SRC_VAL_STREAM: return v1;
    default: return v2;
  }
}

// 0 = filtered out, 1 = ok, 2 = evaluation ERROR (division by zero) —
// the could_error ok/err split of linear_join.rs:495-541.
int closure_apply(const mz_gpu_closure *cl, const u64 *key, const u8 *v1,
                  const u8 *v2, u64 *out_key, u8 *out_val) {
  for (u32 i = 0; i < cl->n_filters; i++) {
    const auto &f = cl->filters[i];
    if (f.src == MZ_SRC_COMPUTE) {
      if (f.off == MZ_COMPUTE_Q17_QTYLT) {
        // quantity < 0.2*sum/count <=> 5*q*count < sum, count > 0
        // (reference: reduce output CASE WHEN count=0 THEN NULL;
        //  tpch_create_index.slt:1470)
        i64 q = read_int(cl_src(key, v1, v2, f.arg0_src) + f.arg0, 8);
        const u8 *slot = cl_src(key, v1, v2, f.arg1_src) + f.arg1;
        i128 S;
        std::memcpy(&S, slot, 16);
        i64 C = read_int(slot + 24, 8);
        if (!(C > 0 && (i128)5 * q * C < S)) return 0;
      }
      if (f.off == MZ_COMPUTE_CMP_FIELDS) {
        i64 x = read_int(cl_src(key, v1, v2, f.arg0_src) + f.arg0,
                           f.width);
        i64 y = read_int(cl_src(key, v1, v2, f.arg1_src) + f.arg1,
                           f.width);
        bool ok;
        switch (f.cmp) {
          case MZ_CMP_LT: ok = x < y; break;
          case MZ_CMP_LE: ok = x <= y; break;
          case MZ_CMP_GT: ok = x > y; break;
          case MZ_CMP_GE: ok = x >= y; break;
          case MZ_CMP_EQ: ok = x == y; break;
          default: ok = x != y; break;
        }
        if (!ok) return 0;
      }
      continue;
    }
    i64 x = read_int(cl_src(key, v1, v2, f.src) + f.off, f.width);
    bool ok;
    switch (f.cmp) {
      case MZ_CMP_LT: ok = x < f.imm; break;
      case MZ_CMP_LE: ok = x <= f.imm; break;
      case MZ_CMP_GT: ok = x > f.imm; break;
      case MZ_CMP_GE: ok = x >= f.imm; break;
      case MZ_CMP_EQ: ok = x == f.imm; break;
      default: ok = x != f.imm; break;
    }
    if (!ok) return 0;
  }
  u8 *outs[2] = {(u8 *)out_key, out_val};
  for (int which = 0; which < 2; which++) {
    u32 nf = which ? cl->n_val_fields : cl->n_key_fields;
    const mz_gpu_field *fs = which ? cl->val_fields : cl->key_fields;
    u8 *dst = outs[which];
    for (u32 i = 0; i < nf; i++) {
      const auto &f = fs[i];
      if (f.src == MZ_SRC_COMPUTE) {
        i64 v = 0;
        if (f.off == MZ_COMPUTE_REVENUE) {
          // extendedprice_cents * (10000 - discount_bp), exact i64 1e-4
          // currency units.
          i64 ep = read_int(cl_src(key, v1, v2, f.arg0_src) + f.arg0, 8);
          i64 disc = read_int(cl_src(key, v1, v2, f.arg1_src) + f.arg1, 8);
          v = ep * (10000 - disc);
        } else if (f.off == MZ_COMPUTE_DIV_I64) {
          i64 a = read_int(cl_src(key, v1, v2, f.arg0_src) + f.arg0, 8);
          i64 b = read_int(cl_src(key, v1, v2, f.arg1_src) + f.arg1, 8);
          if (b == 0) return 2;  // -> error stream
          v = a / b;
        } else if (f.off == MZ_COMPUTE_MUL_I64) {
          i64 a = read_int(cl_src(key, v1, v2, f.arg0_src) + f.arg0, 8);
          i64 b = read_int(cl_src(key, v1, v2, f.arg1_src) + f.arg1, 8);
          v = (i64)((u64)a * (u64)b);  // wrapping, overflowing.rs:24-31
        }  // MZ_COMPUTE_CONST0 leaves v = 0
        std::memcpy(dst, &v, 8);
        dst += 8;
      } else {
        std::memcpy(dst, cl_src(key, v1, v2, f.src) + f.off, f.width);
        dst += f.width;
      }
    }
  }
  return 1;
}

// ------------------------------------------------------ linear join (oracle)

// EditList (mz_join_core.rs:841-891): per distinct val, consolidated
// (time,diff) edits with times advanced by join(meet).
struct EditList {
  // values[i] = (val bytes offset into vals, end index into edits)
  std::vector<std::pair<size_t, size_t>> values;
  std::vector<u8> vals;  // vb-strided storage of distinct vals
  std::vector<std::pair<u64, i64>> edits;
  size_t len() const { return edits.size(); }
};

// Load a key's edits from one or more batches (the trace side presents a
// merged cursor over batches, CursorList; vals iterate in sorted order with
// per-val times from every batch — mz_join_core.rs:857-878).
void editlist_load(const Schema &s, EditList &el,
                   const std::vector<std::pair<const Batch *, std::pair<size_t, size_t>>> &ranges,
                   u64 meet) {
  el.values.clear();
  el.vals.clear();
  el.edits.clear();
  u32 vb = s.vb;
  // k-way merge of per-batch val runs in val order
  struct Cur {
    const Batch *b;
    size_t i, end;
  };
  std::vector<Cur> cur;
  for (auto &r : ranges)
    if (r.second.first < r.second.second)
      cur.push_back({r.first, r.second.first, r.second.second});
  size_t edit_idx = 0;
  while (!cur.empty()) {
    // find smallest current val
    const u8 *minv = nullptr;
    for (auto &c : cur) {
      const u8 *v = c.b->cols.vals.data() + c.i * vb;
      if (!minv || cmp_val(v, minv, vb) < 0) minv = v;
    }
    u8 minv_copy[64];
    if (vb) std::memcpy(minv_copy, minv, vb);
    // collect all edits for this val across batches; times join with meet
    for (auto &c : cur) {
      while (c.i < c.end &&
             cmp_val(c.b->cols.vals.data() + c.i * vb, minv_copy, vb) == 0) {
        u64 t = std::max(c.b->cols.times[c.i], meet);  // time.join_assign(meet)
        el.edits.emplace_back(t, c.b->cols.diffs[c.i]);
        c.i++;
      }
    }
    cur.erase(std::remove_if(cur.begin(), cur.end(),
                             [](const Cur &c) { return c.i >= c.end; }),
              cur.end());
    // consolidate_from(edits, edit_idx) — mz_join_core.rs:869
    std::sort(el.edits.begin() + edit_idx, el.edits.end());
    size_t w = edit_idx;
    for (size_t r = edit_idx; r < el.edits.size();) {
      size_t q = r;
      i64 d = 0;
      while (q < el.edits.size() && el.edits[q].first == el.edits[r].first) {
        d = (i64)((u64)d + (u64)el.edits[q].second);
        q++;
      }
      if (d != 0) el.edits[w++] = {el.edits[r].first, d};
      r = q;
    }
    el.edits.resize(w);
    if (el.edits.size() > edit_idx) {
      size_t off = el.vals.size();
      el.vals.insert(el.vals.end(), minv_copy, minv_copy + vb);
      edit_idx = el.edits.size();
      el.values.emplace_back(off, edit_idx);
    }
  }
}

struct JoinOp {
  Arr *arr1, *arr2;
  mz_gpu_closure cl;
};

// ValueHistory for the linear time scan (mz_join_core.rs:893-975).
struct ValueHistory {
  EditList *edits;
  // future entries: (time, meet, value_idx, diff), sorted DESCENDING;
  // popped from the back (ascending time).
  std::vector<std::tuple<u64, u64, size_t, i64>> future;
  // past entries: (value_idx, time, diff)
  std::vector<std::tuple<size_t, u64, i64>> past;

  void replay() {  // :940-959
    future.clear();
    past.clear();
    for (size_t idx = 0; idx < edits->values.size(); idx++) {
      size_t start = idx == 0 ? 0 : edits->values[idx - 1].second;
      size_t end = edits->values[idx].second;
      for (size_t e = start; e < end; e++)
        future.emplace_back(edits->edits[e].first, edits->edits[e].first, idx,
                            edits->edits[e].second);
    }
    std::sort(future.begin(), future.end(),
              [](const auto &x, const auto &y) { return y < x; });
    for (size_t i = 1; i < future.size(); i++)
      std::get<1>(future[i]) =
          std::min(std::get<1>(future[i]), std::get<1>(future[i - 1]));  // meet
  }
  bool empty() const { return future.empty(); }
  void step() {  // :962-966
    auto [t, m, v, r] = future.back();
    future.pop_back();
    past.emplace_back(v, t, r);
  }
  void advance_past_by(u64 meet) {  // :969-974
    for (auto &p : past) std::get<1>(p) = std::max(std::get<1>(p), meet);
    // consolidate_updates on (value_idx, time, diff)
    std::sort(past.begin(), past.end());
    size_t w = 0;
    for (size_t r = 0; r < past.size();) {
      size_t q = r;
      i64 d = 0;
      while (q < past.size() &&
             std::get<0>(past[q]) == std::get<0>(past[r]) &&
             std::get<1>(past[q]) == std::get<1>(past[r])) {
        d = (i64)((u64)d + (u64)std::get<2>(past[q]));
        q++;
      }
      if (d != 0) past[w++] = {std::get<0>(past[r]), std::get<1>(past[r]), d};
      r = q;
    }
    past.resize(w);
  }
};

// Produce matches for one key (mz_join_core.rs:727-834). `swap` = true when
// the delta came from input 2 (history1 is then the trace side) — result_fn
// argument order is always (key, val-of-input1, val-of-input2).
void join_key(const mz_gpu_closure *cl, const u64 *key, EditList &h1,
              EditList &h2, Cols &out, const Schema &os,
              std::vector<std::tuple<u64, u64, i64>> *errs = nullptr) {
  std::vector<u64> okey(os.kw);
  std::vector<u8> oval(os.vb ? os.vb : 1);
  auto emit = [&](const u8 *v1, const u8 *v2, u64 t, i64 r) {
    int cls = closure_apply(cl, key, v1, v2, okey.data(), oval.data());
    if (cls == 1)
      out.push(okey.data(), os.kw, oval.data(), os.vb, t, r);
    else if (cls == 2 && errs)
      errs->push_back({MZ_ERR_DIVISION_BY_ZERO, t, r});
  };
  if (h1.len() < 10 || h2.len() < 10) {
    // simple strategy: full cross product (:755-767)
    for (size_t i1 = 0; i1 < h1.values.size(); i1++) {
      size_t s1 = i1 == 0 ? 0 : h1.values[i1 - 1].second;
      for (size_t e1 = s1; e1 < h1.values[i1].second; e1++) {
        for (size_t i2 = 0; i2 < h2.values.size(); i2++) {
          size_t s2 = i2 == 0 ? 0 : h2.values[i2 - 1].second;
          for (size_t e2 = s2; e2 < h2.values[i2].second; e2++) {
            u64 t = std::max(h1.edits[e1].first, h2.edits[e2].first);
            i64 r = (i64)((u64)h1.edits[e1].second * (u64)h2.edits[e2].second);
            emit(h1.vals.data() + h1.values[i1].first,
                 h2.vals.data() + h2.values[i2].first, t, r);
          }
        }
      }
    }
  } else {
    // linear time scan (:770-834)
    ValueHistory vh1{&h1}, vh2{&h2};
    vh1.replay();
    vh2.replay();
    auto work1 = [&]() {
      auto [t1, meet, v1, r1] = vh1.future.back();
      vh2.advance_past_by(meet);
      for (auto &[v2, t2, r2] : vh2.past)
        emit(h1.vals.data() + h1.values[v1].first,
             h2.vals.data() + h2.values[v2].first, std::max(t1, t2),
             (i64)((u64)r1 * (u64)r2));
      vh1.step();
    };
    auto work2 = [&]() {
      auto [t2, meet, v2, r2] = vh2.future.back();
      vh1.advance_past_by(meet);
      for (auto &[v1, t1, r1] : vh1.past)
        emit(h1.vals.data() + h1.values[v1].first,
             h2.vals.data() + h2.values[v2].first, std::max(t1, t2),
             (i64)((u64)r1 * (u64)r2));
      vh2.step();
    };
    while (!vh1.empty() && !vh2.empty()) {
      if (std::get<0>(vh1.future.back()) < std::get<0>(vh2.future.back()))
        work1();
      else
        work2();
    }
    while (!vh1.empty()) work1();
    while (!vh2.empty()) work2();
  }
}

// ------------------------------------------------------------ reduce

// Accum restatement (reduce.rs:1611-2270). The benchmark path carries
// COUNT / SUM_I64 / SUM_F64; each slot keeps the full Float bookkeeping.
struct Accum {
  i128 accum = 0;  // wrapping
  i64 non_nulls = 0, pos_infs = 0, neg_infs = 0, nans = 0;
  bool is_zero() const {
    return accum == 0 && non_nulls == 0 && pos_infs == 0 && neg_infs == 0 &&
           nans == 0;
  }
  void plus(const Accum &o) {  // :2102-2149 (wrapping adds)
    accum = (i128)((u128)accum + (u128)o.accum);
    non_nulls = (i64)((u64)non_nulls + (u64)o.non_nulls);
    pos_infs = (i64)((u64)pos_infs + (u64)o.pos_infs);
    neg_infs = (i64)((u64)neg_infs + (u64)o.neg_infs);
    nans = (i64)((u64)nans + (u64)o.nans);
  }
  void mul(i64 f) {  // :2205-2236
    accum = (i128)((u128)accum * (u128)(i128)f);
    non_nulls = (i64)((u64)non_nulls * (u64)f);
    pos_infs = (i64)((u64)pos_infs * (u64)f);
    neg_infs = (i64)((u64)neg_infs * (u64)f);
    nans = (i64)((u64)nans * (u64)f);
  }
};

// float_to_fixed_point (reduce.rs:1663-1697): trunc(n * 2^24) mod 2^128,
// wrapping (group homomorphism), via integer_decode.
i128 float_to_fixed_point(double n) {
  u64 bits;
  std::memcpy(&bits, &n, 8);
  // Rust Float::integer_decode: mantissa incl. implicit bit, exponent-1075
  u64 mantissa = bits & ((1ULL << 52) - 1);
  int exp_bits = (int)((bits >> 52) & 0x7ff);
  int exponent;
  if (exp_bits == 0) {
    exponent = -1074;  // subnormal: no implicit bit
  } else {
    mantissa |= 1ULL << 52;
    exponent = exp_bits - 1075;
  }
  int sign = (bits >> 63) ? -1 : 1;
  long exp = (long)exponent + 24;
  u128 significand = (u128)mantissa;
  u128 magnitude;
  if (exp >= 0) {
    magnitude = exp < 128 ? (significand << exp) : (u128)0;
  } else {
    long sh = -exp;
    magnitude = sh < 128 ? (significand >> sh) : (u128)0;
  }
  i128 m = (i128)magnitude;
  return sign < 0 ? (i128)(~(u128)m + 1) : m;
}

// datum_to_accumulator (reduce.rs:1699-1838) for the three funcs.
Accum datum_to_accum(const mz_gpu_aggregate &a, const u8 *val) {
  Accum r;
  bool null = a.nullable && val[a.off + a.width] != 0;
  switch (a.func) {
    case MZ_AGG_COUNT:
      r.non_nulls = null ? 0 : 1;
      break;
    case MZ_AGG_SUM_I64:
      if (!null) {
        r.accum = (i128)read_int(val + a.off, a.width);
        r.non_nulls = 1;
      }
      break;
    case MZ_AGG_SUM_F64:
      if (!null) {
        double n;
        if (a.width == 4) {
          float f;
          std::memcpy(&f, val + a.off, 4);
          n = (double)f;
        } else {
          std::memcpy(&n, val + a.off, 8);
        }
        r.nans = std::isnan(n) ? 1 : 0;
        r.pos_infs = (n == HUGE_VAL) ? 1 : 0;
        r.neg_infs = (n == -HUGE_VAL) ? 1 : 0;
        r.non_nulls = 1;
        if (!r.nans && !r.pos_infs && !r.neg_infs)
          r.accum = float_to_fixed_point(n);
      }
      break;
  }
  return r;
}

// finalize_accum (reduce.rs:1840-1997). Output slot layout:
// 24 bytes per aggregate: { u8 null; u8 pad[7]; 16-byte value }.
//   COUNT    -> i64 non_nulls in low 8 bytes
//   SUM_I64  -> i128 little-endian (matches Datum::from(i128), :1882)
//   SUM_F64  -> f64 bits in low 8 bytes (fixed-point decode, :1952)
void finalize_accum(const mz_gpu_aggregate &a, const Accum &ac, i64 total,
                    u8 *slot) {
  std::memset(slot, 0, 24);
  // reduce.rs:1844: total positive, accum zero, func != COUNT => NULL
  if (total > 0 && ac.is_zero() && a.func != MZ_AGG_COUNT) {
    slot[0] = 1;
    return;
  }
  switch (a.func) {
    case MZ_AGG_COUNT: {
      i64 c = ac.non_nulls;
      std::memcpy(slot + 8, &c, 8);
      break;
    }
    case MZ_AGG_SUM_I64: {
      i128 v = ac.accum;
      std::memcpy(slot + 8, &v, 16);
      break;
    }
    case MZ_AGG_SUM_F64: {
      double v;
      if (ac.nans > 0 || (ac.pos_infs > 0 && ac.neg_infs > 0))
        v = NAN;
      else if (ac.pos_infs > 0)
        v = HUGE_VAL;
      else if (ac.neg_infs > 0)
        v = -HUGE_VAL;
      else
        v = (double)ac.accum / 16777216.0;  // / 2^24, :1952
      std::memcpy(slot + 8, &v, 8);
      break;
    }
  }
}

struct AccumRow {
  Accum a[MZ_GPU_MAX_AGGS];
  i64 total = 0;
  bool exists() const {
    if (total != 0) return true;
    for (auto &x : a)
      if (!x.is_zero()) return true;
    return false;
  }
};

struct RedOp {
  mz_gpu_reduce_spec spec;
  // deterministic iteration: ordered map on key words
  std::map<std::vector<u64>, AccumRow> state;
};

// ------------------------------------------------- hierarchical min/max
// Restates build_bucketed/build_bucketed_stage (reduce.rs:850-1224) +
// ReductionMonoid (:2273): a reduction tree over val-hash buckets; each
// stage recomputes the min/max of a changed (key, bucket) group from its
// input arrangement (retraction-safe), emitting corrections that feed the
// next stage. Buckets follow plan/reduce.rs:319-326 (val hash % buckets).
// Vals are single i64 datums (8 bytes); single-timestamp pushes.
struct MinMaxOp {
  u32 kw;        // user key words
  bool is_max;
  std::vector<u32> buckets;         // e.g. {4096, 256, 16, 1}
  std::vector<Arr> levels;          // level l keyed (key, bucket_l)
  // last emitted value per (key..., bucket) per level
  std::vector<std::map<std::vector<u64>, i64>> state;
};

// ------------------------------------------------------------ threshold
// Restates threshold_local (src/compute/src/render/threshold.rs:34-51):
// a reduce over the row-keyed arrangement that keeps each record whose
// accumulated count is positive, with that count as the output
// multiplicity. Maintained incrementally per time slice with
// new-minus-old corrections (reduce_abelian contract,
// src/compute/src/extensions/reduce.rs:131). State is the net count per
// (key, val) pair; output diff delta = pos(new) - pos(old), wrapping i64
// like every Diff (src/ore/src/overflowing.rs:24-31).
struct ThrOp {
  Schema s;
  std::map<std::vector<u8>, i64> state;  // key-words bytes || val bytes
};

// ----------------------------------------------------------------- topk
// Restates build_topk / build_topk_negated_stage
// (src/compute/src/render/top_k.rs:322-418,614-770): per group, order the
// records by the order columns (ColumnOrder asc/desc over signed
// little-endian integers; compare_columns, :733-739), break ties in the
// engine's canonical val order (the role of the reference's Row-order
// tie-break `left.cmp(right)`, :738 — deviation: our packed encoding's
// order, identical on oracle and GPU), then keep multiplicities clipped
// to [offset, offset+limit) of the running prefix (:743-766). The
// reference's bucketed stage hierarchy (:380-398) thins work but never
// changes the final modulus-1 stage's output, so we evaluate groups
// directly (policy deviation like the spine-merge schedule).
// Negative input multiplicities are an error (:476,"Negative
// multiplicities in TopK", :494).
using TopKOrderCol = mz_gpu_order_col;

struct TopKOp {
  Schema s;
  u64 offset;
  i64 limit;  // < 0 = none
  std::vector<TopKOrderCol> order;
  // group key -> (val bytes -> net count); inner map in canonical val
  // order via explicit sort at eval time
  std::map<std::vector<u64>, std::map<std::vector<u8>, i64>> state;
};

static i64 read_order_datum(const u8 *val, const TopKOrderCol &c) {
  return read_int(val + c.off, c.width);
}

// Current TopK output of one group: vector of (val, kept multiplicity).
static int topk_eval(const TopKOp *op,
                     const std::map<std::vector<u8>, i64> &grp,
                     std::vector<std::pair<std::vector<u8>, i64>> &out) {
  out.clear();
  std::vector<std::pair<const std::vector<u8> *, i64>> items;
  for (auto &[v, c] : grp) {
    if (c == 0) continue;
    if (c < 0) return -1;  // top_k.rs:690-697 validating stage
    items.push_back({&v, c});
  }
  std::stable_sort(items.begin(), items.end(), [&](auto &a, auto &b) {
    for (auto &c : op->order) {
      i64 x = read_order_datum(a.first->data(), c);
      i64 y = read_order_datum(b.first->data(), c);
      if (x != y) return c.desc ? x > y : x < y;
    }
    return cmp_val(a.first->data(), b.first->data(),
                   (u32)a.first->size()) < 0;
  });
  u64 O = op->offset;
  i64 L = op->limit;
  u64 running = 0;
  for (auto &[v, c] : items) {
    u64 lo = running, hi = running + (u64)c;
    running = hi;
    u64 wlo = lo > O ? lo : O;
    u64 whi = hi;
    if (L >= 0) {
      u64 cap = O + (u64)L;
      if (whi > cap) whi = cap;
    }
    if (whi > wlo) out.push_back({*v, (i64)(whi - wlo)});
  }
  return 0;
}

// ------------------------------------------------------------- context

struct Out {
  mz_gpu_out pub_;
  Cols cols;
  Schema schema;
  std::vector<u64> err_codes, err_times;
  std::vector<i64> err_diffs;
};

}  // namespace

struct orc_ctx {
  std::string err;
  std::vector<std::unique_ptr<Arr>> arrs;
  std::vector<std::unique_ptr<JoinOp>> joins;
  std::vector<std::unique_ptr<RedOp>> reds;
  std::vector<std::unique_ptr<ThrOp>> thrs;
  std::vector<std::unique_ptr<TopKOp>> topks;
};

extern "C" {

u64 orc_route_hash(const u64 *kw, u32 n);

orc_ctx *orc_init() { return new orc_ctx(); }
void orc_fini(orc_ctx *c) { delete c; }
const char *orc_last_error(orc_ctx *c) { return c->err.c_str(); }

Arr *orc_arr_create(orc_ctx *c, const mz_gpu_schema *s) {
  auto a = std::make_unique<Arr>();
  a->schema = {s->key_words, s->val_bytes};
  Arr *p = a.get();
  c->arrs.push_back(std::move(a));
  return p;
}

static void cols_from_updates(const mz_gpu_updates *u, const Schema &s,
                              Cols &c) {
  c.keys.assign(u->keys, u->keys + u->n * s.kw);
  if (s.vb)
    c.vals.assign(u->vals, u->vals + u->n * s.vb);
  c.times.assign(u->times, u->times + u->n);
  c.diffs.assign(u->diffs, u->diffs + u->n);
}

int orc_arr_push_batch(orc_ctx *c, Arr *a, const mz_gpu_updates *u) {
  auto b = std::make_unique<Batch>();
  cols_from_updates(u, a->schema, b->cols);
  b->lower = u->lower;
  b->upper = u->upper;
  b->build_index(a->schema);
  a->batches.push_back(std::move(b));
  a->upper = std::max(a->upper, u->upper);
  // geometric tail merging (same amortized policy as the GPU engine):
  // keeps probe fan-out logarithmic in steady state
  while (a->batches.size() >= 2) {
    size_t nb = a->batches.size();
    if (a->batches[nb - 2]->cols.size() <= 2 * a->batches[nb - 1]->cols.size())
      a->merge_span(nb - 2, nb);
    else
      break;
  }
  return 0;
}

// fused consolidate + push (mirror of mz_gpu_arr_insert)
int orc_arr_insert(orc_ctx *c, Arr *a, const mz_gpu_updates *u) {
  auto b = std::make_unique<Batch>();
  cols_from_updates(u, a->schema, b->cols);
  consolidate(a->schema, b->cols);
  b->lower = u->lower;
  b->upper = u->upper;
  b->build_index(a->schema);
  a->batches.push_back(std::move(b));
  a->upper = std::max(a->upper, u->upper);
  while (a->batches.size() >= 2) {
    size_t nb = a->batches.size();
    if (a->batches[nb - 2]->cols.size() <=
        2 * a->batches[nb - 1]->cols.size())
      a->merge_span(nb - 2, nb);
    else
      break;
  }
  return 0;
}

int orc_arr_set_logical_compaction(orc_ctx *c, Arr *a, u64 f) {
  a->logical_compaction = f;
  return 0;
}

int orc_arr_maintain(orc_ctx *c, Arr *a, u64 fuel) {
  (void)fuel;  // oracle merges eagerly (scheduling policy)
  a->merge_all();
  return 0;
}

int orc_arr_stats(orc_ctx *c, Arr *a, u64 *nb, u64 *nu, u64 *bytes) {
  *nb = a->batches.size();
  u64 n = 0, by = 0;
  for (auto &b : a->batches) {
    n += b->cols.size();
    by += b->cols.keys.size() * 8 + b->cols.vals.size() +
          b->cols.size() * 16;
  }
  *nu = n;
  *bytes = by;
  return 0;
}

JoinOp *orc_join_create(orc_ctx *c, Arr *a1, Arr *a2,
                        const mz_gpu_closure *cl) {
  auto j = std::make_unique<JoinOp>();
  j->arr1 = a1;
  j->arr2 = a2;
  j->cl = *cl;
  JoinOp *p = j.get();
  c->joins.push_back(std::move(j));
  return p;
}

using ErrRows = std::vector<std::tuple<u64, u64, i64>>;  // code,time,diff

static void consolidate_errs(ErrRows &e) {
  std::sort(e.begin(), e.end(),
            [](const auto &a, const auto &b) {
              if (std::get<0>(a) != std::get<0>(b))
                return std::get<0>(a) < std::get<0>(b);
              return std::get<1>(a) < std::get<1>(b);
            });
  ErrRows out;
  for (auto &r : e) {
    if (!out.empty() && std::get<0>(out.back()) == std::get<0>(r) &&
        std::get<1>(out.back()) == std::get<1>(r))
      std::get<2>(out.back()) =
          (i64)((u64)std::get<2>(out.back()) + (u64)std::get<2>(r));
    else
      out.push_back(r);
  }
  out.erase(std::remove_if(out.begin(), out.end(),
                           [](const auto &r) { return std::get<2>(r) == 0; }),
            out.end());
  e = std::move(out);
}

static Out *attach_errs(Out *o, ErrRows &&e) {
  consolidate_errs(e);
  for (auto &[code, t, d] : e) {
    o->err_codes.push_back(code);
    o->err_times.push_back(t);
    o->err_diffs.push_back(d);
  }
  o->pub_.err_n = o->err_codes.size();
  o->pub_.err_codes = o->err_codes.data();
  o->pub_.err_times = o->err_times.data();
  o->pub_.err_diffs = o->err_diffs.data();
  return o;
}

static Out *make_out(Cols &&c, const Schema &s) {
  Out *o = new Out();
  o->cols = std::move(c);
  o->schema = s;
  o->pub_ = mz_gpu_out{};
  o->pub_.keys = o->cols.keys.data();
  o->pub_.vals = o->cols.vals.data();
  o->pub_.times = o->cols.times.data();
  o->pub_.diffs = o->cols.diffs.data();
  o->pub_.n = o->cols.size();
  o->pub_.on_device = 0;
  o->pub_.schema = {s.kw, s.vb};
  return o;
}

// mz_join_core's per-batch work: join `delta` (side 1 or 2) against the
// opposing arrangement's batches as of now (= cursor_through(acknowledged),
// mz_join_core.rs:237-368). meet = delta batch's capability time (lower).
int orc_join_push(orc_ctx *c, JoinOp *op, int side, const mz_gpu_updates *u,
                  mz_gpu_out **out) {
  Arr *own = side == 1 ? op->arr1 : op->arr2;
  Arr *opp = side == 1 ? op->arr2 : op->arr1;
  const Schema &ds = own->schema;  // delta rows use their own side's schema
  const Schema &ts = opp->schema;  // trace rows use the opposing schema
  Batch delta;
  cols_from_updates(u, ds, delta.cols);
  delta.lower = u->lower;
  delta.upper = u->upper;
  delta.build_index(ds);
  u64 meet = u->lower;
  Schema os = {op->cl.out.key_words, op->cl.out.val_bytes};
  Cols result;
  ErrRows errs;
  EditList el_delta, el_trace;
  // merge scan over delta's distinct keys (start_work, :644-663): the delta
  // cursor drives; trace cursors seek.
  size_t ndk = delta.key_starts.size() ? delta.key_starts.size() - 1 : 0;
  for (size_t ki = 0; ki < ndk; ki++) {
    const u64 *key = delta.cols.keys.data() + delta.key_starts[ki] * ds.kw;
    std::vector<std::pair<const Batch *, std::pair<size_t, size_t>>> tr;
    bool any = false;
    for (auto &b : opp->batches) {
      auto r = b->seek(ts, key);
      if (r.first < r.second) {
        tr.push_back({b.get(), r});
        any = true;
      }
    }
    if (!any) continue;
    editlist_load(ds, el_delta,
                  {{&delta, {delta.key_starts[ki], delta.key_starts[ki + 1]}}},
                  meet);
    editlist_load(ts, el_trace, tr, meet);
    if (el_delta.len() == 0 || el_trace.len() == 0) continue;
    if (side == 1)
      join_key(&op->cl, key, el_delta, el_trace, result, os, &errs);
    else
      join_key(&op->cl, key, el_trace, el_delta, result, os, &errs);
  }
  consolidate(os, result);
  *out = &attach_errs(make_out(std::move(result), os),
                      std::move(errs))->pub_;
  return 0;
}

// half_join probe (delta_join.rs:500-583 / half_join2 contract): for each
// stream update (key, val, t) probe `lookup`; trace updates at t' match iff
// le ? t' <= t : t' < t; emit (closure(key, stream_val, lookup_val), t,
// d_stream * d_lookup). Output consolidated. `stream_vb` = the stream
// updates' val stride (may differ from the lookup arrangement's).
int orc_halfjoin(orc_ctx *c, Arr *lookup, const mz_gpu_updates *u,
                 u32 stream_vb, int le, const mz_gpu_closure *cl,
                 mz_gpu_out **out) {
  const Schema &ls = lookup->schema;
  Schema os = {cl->out.key_words, cl->out.val_bytes};
  Cols result;
  ErrRows errs;
  std::vector<u64> okey(os.kw);
  std::vector<u8> oval(os.vb ? os.vb : 1);
  for (u64 i = 0; i < u->n; i++) {
    const u64 *key = u->keys + i * ls.kw;
    const u8 *sval = stream_vb ? u->vals + i * stream_vb : nullptr;
    u64 t = u->times[i];
    i64 d1 = u->diffs[i];
    for (auto &b : lookup->batches) {
      auto r = b->seek(ls, key);
      for (size_t j = r.first; j < r.second; j++) {
        u64 t2 = b->cols.times[j];
        if (le ? (t2 <= t) : (t2 < t)) {
          const u8 *lval = ls.vb ? b->cols.vals.data() + j * ls.vb : nullptr;
          int cls = closure_apply(cl, key, sval, lval, okey.data(),
                                  oval.data());
          i64 d = (i64)((u64)d1 * (u64)b->cols.diffs[j]);
          if (cls == 1)
            result.push(okey.data(), os.kw, oval.data(), os.vb, t, d);
          else if (cls == 2)
            errs.push_back({MZ_ERR_DIVISION_BY_ZERO, t, d});
        }
      }
    }
  }
  consolidate(os, result);
  *out = &attach_errs(make_out(std::move(result), os),
                      std::move(errs))->pub_;
  return 0;
}

// flat-map mirror: closure over a stream, no lookup
int orc_map(orc_ctx *c, const mz_gpu_schema *in, const mz_gpu_updates *u,
            const mz_gpu_closure *cl, mz_gpu_out **out) {
  Schema s = {in->key_words, in->val_bytes};
  Schema os = {cl->out.key_words, cl->out.val_bytes};
  Cols result;
  std::vector<u64> okey(os.kw);
  std::vector<u8> oval(os.vb ? os.vb : 1);
  for (u64 i = 0; i < u->n; i++) {
    const u64 *key = u->keys + i * s.kw;
    const u8 *val = s.vb ? u->vals + i * s.vb : nullptr;
    if (closure_apply(cl, key, val, nullptr, okey.data(),
                      oval.data()) == 1)
      result.push(okey.data(), os.kw, oval.data(), os.vb, u->times[i],
                  u->diffs[i]);
  }
  *out = &make_out(std::move(result), os)->pub_;
  return 0;
}

// peek mirror (handle_peek analog): per requested key, (val, summed
// diff) as of `time` via the le half-join with the identity closure.
int orc_peek(orc_ctx *c, Arr *arr, const u64 *keys, u64 n_keys, u64 time,
             mz_gpu_out **out) {
  u32 kw = arr->schema.kw, vb = arr->schema.vb;
  std::vector<u64> times(n_keys, time);
  std::vector<i64> diffs(n_keys, 1);
  mz_gpu_updates u{};
  u.keys = keys;
  u.vals = nullptr;
  u.times = times.data();
  u.diffs = diffs.data();
  u.n = n_keys;
  u.lower = time;
  u.upper = time + 1;
  u.on_device = 0;
  mz_gpu_closure cl{};
  cl.n_filters = 0;
  cl.n_key_fields = 1;
  cl.key_fields[0] = mz_gpu_field{MZ_SRC_KEY, 0, (u8)(8 * kw), 0, 0, 0, 0};
  cl.n_val_fields = vb ? 1u : 0u;
  if (vb)
    cl.val_fields[0] =
        mz_gpu_field{MZ_SRC_VAL_LOOKUP, 0, (u8)vb, 0, 0, 0, 0};
  cl.out.key_words = kw;
  cl.out.val_bytes = vb;
  return orc_halfjoin(c, arr, &u, 0, 1, &cl, out);
}

RedOp *orc_reduce_create(orc_ctx *c, const mz_gpu_reduce_spec *spec) {
  auto r = std::make_unique<RedOp>();
  r->spec = *spec;
  RedOp *p = r.get();
  c->reds.push_back(std::move(r));
  return p;
}

ThrOp *orc_threshold_create(orc_ctx *c, const mz_gpu_schema *s) {
  auto r = std::make_unique<ThrOp>();
  r->s = {s->key_words, s->val_bytes};
  ThrOp *p = r.get();
  c->thrs.push_back(std::move(r));
  return p;
}

int orc_threshold_push(orc_ctx *c, ThrOp *op, const mz_gpu_updates *u,
                       mz_gpu_out **out) {
  const Schema s = op->s;
  Cols result;
  std::vector<u64> order(u->n);
  for (u64 i = 0; i < u->n; i++) order[i] = i;
  std::stable_sort(order.begin(), order.end(),
                   [&](u64 a, u64 b) { return u->times[a] < u->times[b]; });
  std::vector<u8> ck(s.kw * 8 + s.vb);
  size_t p = 0;
  while (p < order.size()) {
    u64 t = u->times[order[p]];
    std::map<std::vector<u8>, i64> olds;
    while (p < order.size() && u->times[order[p]] == t) {
      u64 i = order[p++];
      std::memcpy(ck.data(), u->keys + i * s.kw, s.kw * 8);
      if (s.vb) std::memcpy(ck.data() + s.kw * 8, u->vals + i * s.vb, s.vb);
      auto it = op->state.find(ck);
      i64 cur = it == op->state.end() ? 0 : it->second;
      if (olds.find(ck) == olds.end()) olds[ck] = cur;
      op->state[ck] = (i64)((u64)cur + (u64)u->diffs[i]);
    }
    for (auto &[k, old] : olds) {
      i64 nw = op->state[k];
      i64 po = old > 0 ? old : 0, pn = nw > 0 ? nw : 0;
      i64 delta = (i64)((u64)pn - (u64)po);
      if (delta) {
        u64 kws[MAX_KW];
        std::memcpy(kws, k.data(), s.kw * 8);
        result.push(kws, s.kw, k.data() + s.kw * 8, s.vb, t, delta);
      }
      if (nw == 0) op->state.erase(k);
    }
  }
  consolidate(s, result);
  *out = &make_out(std::move(result), s)->pub_;
  return 0;
}

// build_accumulable push: move datums into accumulators scaled by diff
// (explode_one, reduce.rs:1409-1431 + Multiply :2205), merge into resident
// state per key (Semigroup :2102), emit corrections new-minus-old per
// changed key at each timestamp in order (reduce_abelian contract).
int orc_reduce_push(orc_ctx *c, RedOp *op, const mz_gpu_updates *u,
                    mz_gpu_out **out) {
  const Schema in = {op->spec.in.key_words, op->spec.in.val_bytes};
  Schema os = {op->spec.out.key_words, op->spec.out.val_bytes};
  u32 na = op->spec.n_aggs;
  Cols result;
  // group updates by time (ascending), then process per time
  std::vector<u64> order(u->n);
  for (u64 i = 0; i < u->n; i++) order[i] = i;
  std::stable_sort(order.begin(), order.end(),
                   [&](u64 a, u64 b) { return u->times[a] < u->times[b]; });
  std::vector<u8> oldrow(os.vb), newrow(os.vb);
  size_t p = 0;
  while (p < order.size()) {
    u64 t = u->times[order[p]];
    // apply all updates at time t, tracking changed keys
    std::map<std::vector<u64>, AccumRow> olds;
    while (p < order.size() && u->times[order[p]] == t) {
      u64 i = order[p++];
      std::vector<u64> key(u->keys + i * in.kw, u->keys + (i + 1) * in.kw);
      auto it = op->state.find(key);
      if (olds.find(key) == olds.end())
        olds[key] = it != op->state.end() ? it->second : AccumRow();
      AccumRow &st = op->state[key];
      const u8 *val = in.vb ? u->vals + i * in.vb : nullptr;
      i64 d = u->diffs[i];
      for (u32 a = 0; a < na; a++) {
        Accum ac = datum_to_accum(op->spec.aggs[a], val);
        ac.mul(d);
        st.a[a].plus(ac);
      }
      st.total = (i64)((u64)st.total + (u64)d);
    }
    // emit corrections per changed key
    for (auto &[key, old] : olds) {
      AccumRow &nw = op->state[key];
      bool oe = old.exists(), ne = nw.exists();
      if (oe) {
        for (u32 a = 0; a < na; a++)
          finalize_accum(op->spec.aggs[a], old.a[a], old.total,
                         oldrow.data() + 24 * a);
      }
      if (ne) {
        for (u32 a = 0; a < na; a++)
          finalize_accum(op->spec.aggs[a], nw.a[a], nw.total,
                         newrow.data() + 24 * a);
      }
      if (oe && ne && std::memcmp(oldrow.data(), newrow.data(), os.vb) == 0)
        continue;
      if (oe) result.push(key.data(), os.kw, oldrow.data(), os.vb, t, -1);
      if (ne) result.push(key.data(), os.kw, newrow.data(), os.vb, t, 1);
      if (!ne) op->state.erase(key);
    }
  }
  consolidate(os, result);
  *out = &make_out(std::move(result), os)->pub_;
  return 0;
}

TopKOp *orc_topk_create(orc_ctx *c, const mz_gpu_topk_spec *spec) {
  auto r = std::make_unique<TopKOp>();
  r->s = {spec->in.key_words, spec->in.val_bytes};
  r->offset = spec->offset;
  r->limit = spec->limit;
  for (u32 i = 0; i < spec->n_order; i++) r->order.push_back(spec->order[i]);
  TopKOp *p = r.get();
  c->topks.push_back(std::move(r));
  return p;
}

int orc_topk_push(orc_ctx *c, TopKOp *op, const mz_gpu_updates *u,
                  mz_gpu_out **out) {
  const Schema s = op->s;
  Cols result;
  std::vector<u64> order(u->n);
  for (u64 i = 0; i < u->n; i++) order[i] = i;
  std::stable_sort(order.begin(), order.end(),
                   [&](u64 a, u64 b) { return u->times[a] < u->times[b]; });
  std::vector<std::pair<std::vector<u8>, i64>> oldo, newo;
  size_t p = 0;
  while (p < order.size()) {
    u64 t = u->times[order[p]];
    // group contents before this slice, for changed groups
    std::map<std::vector<u64>, std::map<std::vector<u8>, i64>> olds;
    while (p < order.size() && u->times[order[p]] == t) {
      u64 i = order[p++];
      std::vector<u64> key(u->keys + i * s.kw, u->keys + (i + 1) * s.kw);
      std::vector<u8> val(u->vals + i * s.vb, u->vals + (i + 1) * s.vb);
      auto it = op->state.find(key);
      if (olds.find(key) == olds.end())
        olds[key] = it != op->state.end()
                        ? it->second
                        : std::map<std::vector<u8>, i64>();
      auto &grp = op->state[key];
      grp[val] = (i64)((u64)grp[val] + (u64)u->diffs[i]);
      if (grp[val] == 0) grp.erase(val);
    }
    for (auto &[key, old] : olds) {
      auto it = op->state.find(key);
      static const std::map<std::vector<u8>, i64> kEmpty;
      const auto &nw = it != op->state.end() ? it->second : kEmpty;
      if (topk_eval(op, old, oldo) || topk_eval(op, nw, newo)) {
        c->err = "negative multiplicities in TopK";
        return -1;
      }
      for (auto &[v, d] : oldo)
        result.push(key.data(), s.kw, v.data(), s.vb, t, -d);
      for (auto &[v, d] : newo)
        result.push(key.data(), s.kw, v.data(), s.vb, t, d);
      if (it != op->state.end() && it->second.empty()) op->state.erase(it);
    }
  }
  consolidate(s, result);
  *out = &make_out(std::move(result), s)->pub_;
  return 0;
}

int orc_consolidate(orc_ctx *c, const mz_gpu_schema *sc,
                    const mz_gpu_updates *u, mz_gpu_out **out) {
  Schema s = {sc->key_words, sc->val_bytes};
  Cols cols;
  cols_from_updates(u, s, cols);
  consolidate(s, cols);
  Out *o = make_out(std::move(cols), s);
  o->pub_.schema = *sc;
  *out = &o->pub_;
  return 0;
}

void orc_out_release(orc_ctx *c, mz_gpu_out *o) {
  // Out::pub_ is the first member; the public pointer IS the Out pointer.
  delete reinterpret_cast<Out *>(o);
}

// Routing hash: splitmix64 over the key words (substitute for fixed-seed
// ahash, used identically by oracle and GPU — DESIGN.md §2.2).
MinMaxOp *orc_minmax_create(orc_ctx *c, const mz_gpu_schema *in, int is_max,
                            const u32 *buckets, u32 n_levels) {
  auto *op = new MinMaxOp();
  op->kw = in->key_words;
  op->is_max = is_max;
  op->buckets.assign(buckets, buckets + n_levels);
  op->levels.resize(n_levels);
  op->state.resize(n_levels);
  for (u32 l = 0; l < n_levels; l++)
    op->levels[l].schema = {op->kw + 1, 8};  // (key,bucket) -> val i64
  return op;
}

// One push of (key, val i64) updates at a single timestamp; returns the
// top-level corrections (key, minmax val, +/-1).
int orc_minmax_push(orc_ctx *c, MinMaxOp *op, const mz_gpu_updates *u,
                    mz_gpu_out **out) {
  u32 kw = op->kw;
  if (u->upper > u->lower + 1) return -1;  // single-timestamp contract
  u64 t = u->lower;
  u32 L = (u32)op->buckets.size();
  // build level-0 input: (key, b0(val)) -> val
  Cols cur;
  for (u64 i = 0; i < u->n; i++) {
    u64 kb[MAX_KW + 1];
    for (u32 w = 0; w < kw; w++) kb[w] = u->keys[i * kw + w];
    u64 vword;
    std::memcpy(&vword, u->vals + i * 8, 8);
    kb[kw] = orc_route_hash(&vword, 1) % op->buckets[0];
    cur.push(kb, kw + 1, u->vals + i * 8, 8, t, u->diffs[i]);
  }
  Cols final_out;
  for (u32 l = 0; l < L; l++) {
    Arr &A = op->levels[l];
    Schema ls = A.schema;
    consolidate(ls, cur);
    if (cur.size() == 0) break;
    // insert into the level arrangement
    auto b = std::make_unique<Batch>();
    b->cols = cur;  // copy
    b->lower = t;
    b->upper = t + 1;
    b->build_index(ls);
    A.batches.push_back(std::move(b));
    while (A.batches.size() >= 2) {
      size_t nb = A.batches.size();
      if (A.batches[nb - 2]->cols.size() <= 2 * A.batches[nb - 1]->cols.size())
        A.merge_span(nb - 2, nb);
      else
        break;
    }
    // changed groups = distinct keys of cur; recompute each from A
    Cols next;
    std::map<std::vector<u64>, i64> &st = op->state[l];
    size_t n = cur.size();
    for (size_t i = 0; i < n;) {
      size_t j = i;
      while (j < n && cmp_key(cur.keys.data() + i * ls.kw,
                              cur.keys.data() + j * ls.kw, ls.kw) == 0)
        j++;
      const u64 *gkey = cur.keys.data() + i * ls.kw;
      // group scan over all batches: per val net diff
      std::map<i64, i64> valsum;
      for (auto &bb : A.batches) {
        auto r = bb->seek(ls, gkey);
        for (size_t q = r.first; q < r.second; q++) {
          i64 v;
          std::memcpy(&v, bb->cols.vals.data() + q * 8, 8);
          valsum[v] = (i64)((u64)valsum[v] + (u64)bb->cols.diffs[q]);
        }
      }
      bool exists = false;
      i64 m = 0;
      for (auto &[v, d] : valsum) {
        if (d == 0) continue;
        if (!exists || (op->is_max ? v > m : v < m)) m = v;
        exists = true;
      }
      std::vector<u64> skey(gkey, gkey + ls.kw);
      auto it = st.find(skey);
      bool old_exists = it != st.end();
      i64 old_m = old_exists ? it->second : 0;
      if (old_exists != exists || (exists && old_m != m)) {
        // corrections keyed for the next level (or the user key at top)
        u64 okb[MAX_KW + 1];
        for (u32 w = 0; w < kw; w++) okb[w] = gkey[w];
        u32 okw = (l + 1 < L) ? kw + 1 : kw;
        if (l + 1 < L) okb[kw] = gkey[kw] % op->buckets[l + 1];
        if (old_exists)
          next.push(okb, okw, (const u8 *)&old_m, 8, t, -1);
        if (exists) {
          next.push(okb, okw, (const u8 *)&m, 8, t, 1);
          st[skey] = m;
        } else {
          st.erase(skey);
        }
      }
      i = j;
    }
    if (l + 1 < L) {
      cur = std::move(next);
    } else {
      final_out = std::move(next);
    }
  }
  Schema os = {kw, 8};
  consolidate(os, final_out);
  *out = &make_out(std::move(final_out), os)->pub_;
  return 0;
}

u64 orc_route_hash(const u64 *kw, u32 n) {
  u64 h = 0x9E3779B97F4A7C15ULL;
  for (u32 i = 0; i < n; i++) {
    u64 x = kw[i] + h;
    x ^= x >> 30;
    x *= 0xBF58476D1CE4E5B9ULL;
    x ^= x >> 27;
    x *= 0x94D049BB133111EBULL;
    x ^= x >> 31;
    h = x;
  }
  return h;
}

}  // extern "C"
