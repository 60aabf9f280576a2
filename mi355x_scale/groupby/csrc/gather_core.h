// Core of the multithreaded group-gather engine (N1), free of any
// Python/pybind dependency so sanitizer harnesses (TSAN) can exercise
// the concurrent factorize directly (tests/native/gather_tsan.cpp,
// SURVEY §5.2). Included by gather.cpp (the pybind module).
#pragma once

#include <algorithm>
#include <atomic>
#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string_view>
#include <thread>
#include <vector>

namespace gather_core {

static inline uint64_t splitmix64(uint64_t x) {
  x += 0x9e3779b97f4a7c15ULL;
  x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ULL;
  x = (x ^ (x >> 27)) * 0x94d049bb133111ebULL;
  return x ^ (x >> 31);
}

static inline uint64_t fnv1a(const char* p, size_t n) {
  uint64_t h = 1469598103934665603ULL;
  for (size_t i = 0; i < n; ++i) {
    h ^= (unsigned char)p[i];
    h *= 1099511628211ULL;
  }
  return h;
}

static int n_threads(int64_t n) {
  unsigned hw = std::thread::hardware_concurrency();
  int t = hw ? (int)hw : 8;
  if (t > 16) t = 16;
  int64_t per = 64 * 1024;  // don't spin threads for tiny inputs
  if (n / per < t) t = (int)std::max<int64_t>(1, n / per);
  return std::max(1, t);
}

template <typename F>
static void parallel_rows(int64_t n, F body) {
  int t = n_threads(n);
  if (t == 1) {
    body((int64_t)0, n);
    return;
  }
  std::vector<std::thread> ths;
  int64_t chunk = (n + t - 1) / t;
  for (int i = 0; i < t; ++i) {
    int64_t lo = i * chunk, hi = std::min<int64_t>(n, lo + chunk);
    if (lo >= hi) break;
    ths.emplace_back([=] { body(lo, hi); });
  }
  for (auto& th : ths) th.join();
}

struct FactorizeResult {
  std::vector<int32_t> codes;
  std::vector<int64_t> first_rows;  // uniques as first-row indices, sorted
};

// Per-thread open-addressing table: grows by rehash, stays L2-resident
// for group-key cardinalities (a shared 2n-slot table thrashes cache/TLB
// — measured 25x super-linear scaling from 1.57M to 15.7M rows).
template <typename KeyAccess>
struct LocalTable {
  const KeyAccess& key;
  std::vector<int64_t> slot_row;    // first row holding the key, -1 empty
  std::vector<int32_t> slot_code;   // local code of that key
  std::vector<int64_t> uniq_rows;   // local code -> first row
  size_t mask;

  explicit LocalTable(const KeyAccess& k, size_t initial = 1 << 14)
      : key(k), slot_row(initial, -1), slot_code(initial, -1),
        mask(initial - 1) {}

  void rehash() {
    size_t nsize = slot_row.size() * 2;
    std::vector<int64_t> nrow(nsize, -1);
    std::vector<int32_t> ncode(nsize, -1);
    size_t nmask = nsize - 1;
    for (size_t s = 0; s < slot_row.size(); ++s) {
      if (slot_row[s] < 0) continue;
      size_t t = key.hash(slot_row[s]) & nmask;
      while (nrow[t] >= 0) t = (t + 1) & nmask;
      nrow[t] = slot_row[s];
      ncode[t] = slot_code[s];
    }
    slot_row.swap(nrow);
    slot_code.swap(ncode);
    mask = nmask;
  }

  int32_t insert(int64_t r) {
    size_t s = key.hash(r) & mask;
    while (slot_row[s] >= 0) {
      if (key.eq(slot_row[s], r)) return slot_code[s];
      s = (s + 1) & mask;
    }
    if ((uniq_rows.size() + 1) * 10 > slot_row.size() * 7) {
      rehash();
      s = key.hash(r) & mask;
      while (slot_row[s] >= 0) s = (s + 1) & mask;
    }
    int32_t code = (int32_t)uniq_rows.size();
    slot_row[s] = r;
    slot_code[s] = code;
    uniq_rows.push_back(r);
    return code;
  }
};

// KeyAccess: hash(row), eq(rowA, rowB), less(rowA, rowB), valid(row).
// Runs WITHOUT the GIL — no Python objects may be touched here, and
// worker-thread bodies must not throw (std::terminate).
//
// Three phases: (1) each thread factorizes its row slice against its own
// local table; (2) the threads' unique sets are merged serially into one
// global table (cost ~ threads x cardinality, not rows) and sorted by
// key value (pd.factorize(sort=True) semantics); (3) rows remap
// local code -> sorted global code in parallel.
template <typename KeyAccess>
static FactorizeResult factorize_impl(int64_t n, const KeyAccess& key) {
  int nt = n_threads(n);
  std::vector<int32_t> local_codes(n);
  std::vector<LocalTable<KeyAccess>> tables;
  tables.reserve(nt);
  for (int i = 0; i < nt; ++i) tables.emplace_back(key);

  int64_t chunk = (n + nt - 1) / nt;
  {
    std::vector<std::thread> ths;
    for (int t = 0; t < nt; ++t) {
      int64_t lo = t * chunk, hi = std::min<int64_t>(n, lo + chunk);
      if (lo >= hi) break;
      ths.emplace_back([&, t, lo, hi] {
        auto& tab = tables[t];
        for (int64_t r = lo; r < hi; ++r)
          local_codes[r] = key.valid(r) ? tab.insert(r) : -1;
      });
    }
    for (auto& th : ths) th.join();
  }

  // merge local uniques into a global table; map[t][local] -> global
  LocalTable<KeyAccess> global(key, 1 << 15);
  std::vector<std::vector<int32_t>> to_global(nt);
  for (int t = 0; t < nt; ++t) {
    to_global[t].resize(tables[t].uniq_rows.size());
    for (size_t i = 0; i < tables[t].uniq_rows.size(); ++i)
      to_global[t][i] = global.insert(tables[t].uniq_rows[i]);
  }

  // sort uniques by key value; rank[global code] -> sorted code
  size_t G = global.uniq_rows.size();
  std::vector<int32_t> order(G);
  for (size_t i = 0; i < G; ++i) order[i] = (int32_t)i;
  std::sort(order.begin(), order.end(), [&](int32_t a, int32_t b) {
    return key.less(global.uniq_rows[a], global.uniq_rows[b]);
  });
  std::vector<int32_t> rank(G);
  FactorizeResult res;
  res.first_rows.resize(G);
  for (size_t i = 0; i < G; ++i) {
    rank[order[i]] = (int32_t)i;
    res.first_rows[i] = global.uniq_rows[order[i]];
  }
  // fold the sort rank into the per-thread maps
  for (int t = 0; t < nt; ++t)
    for (auto& c : to_global[t]) c = rank[c];

  res.codes.resize(n);
  int32_t* cp = res.codes.data();
  {
    std::vector<std::thread> ths;
    for (int t = 0; t < nt; ++t) {
      int64_t lo = t * chunk, hi = std::min<int64_t>(n, lo + chunk);
      if (lo >= hi) break;
      ths.emplace_back([&, t, lo, hi] {
        const int32_t* m = to_global[t].data();
        for (int64_t r = lo; r < hi; ++r)
          cp[r] = local_codes[r] < 0 ? -1 : m[local_codes[r]];
      });
    }
    for (auto& th : ths) th.join();
  }
  return res;
}


struct I64Key {
  const int64_t* v;
  uint64_t hash(int64_t r) const { return splitmix64((uint64_t)v[r]); }
  bool eq(int64_t a, int64_t b) const { return v[a] == v[b]; }
  bool less(int64_t a, int64_t b) const { return v[a] < v[b]; }
  bool valid(int64_t) const { return true; }
};

template <typename OffT>
struct StrKey {
  const OffT* off;
  const char* data;
  const uint8_t* validity;  // arrow bitmap, may be null
  int64_t voffset;          // arrow array offset into the bitmap
  std::string_view sv(int64_t r) const {
    return {data + off[r], (size_t)(off[r + 1] - off[r])};
  }
  uint64_t hash(int64_t r) const {
    auto s = sv(r);
    return fnv1a(s.data(), s.size());
  }
  bool eq(int64_t a, int64_t b) const { return sv(a) == sv(b); }
  bool less(int64_t a, int64_t b) const { return sv(a) < sv(b); }
  bool valid(int64_t r) const {
    if (!validity) return true;
    int64_t i = r + voffset;
    return (validity[i >> 3] >> (i & 7)) & 1;
  }
};

}  // namespace gather_core
