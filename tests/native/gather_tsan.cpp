// ThreadSanitizer harness for the group-gather engine core
// (sanitizer passes on the native components).
//
// Build & run (no Python, no GPU):
//   g++ -std=c++17 -O1 -g -fsanitize=thread -pthread \
//       -I mi355x_scale/groupby/csrc \
//       tests/native/gather_tsan.cpp -o /tmp/gather_tsan && /tmp/gather_tsan
//
// Exercises the concurrent phases (per-thread local tables, shared
// local_codes writes, parallel remap) under TSAN, and checks the result
// against a single-threaded oracle.

#include <cassert>
#include <cstdio>
#include <map>
#include <random>
#include <string>
#include <vector>

#include "gather_core.h"

using namespace gather_core;

int main() {
  std::mt19937_64 rng(123);
  const int64_t N = 2'000'000;
  const int64_t G = 50'000;

  // int64 keys with duplicates
  std::vector<int64_t> vals(N);
  for (auto& v : vals) v = (int64_t)(rng() % G) * 1'000'003 - 7;
  I64Key key{vals.data()};
  FactorizeResult res = factorize_impl(N, key);

  // oracle: sorted-unique ranks
  std::map<int64_t, int32_t> uniq;
  for (auto v : vals) uniq.emplace(v, 0);
  int32_t c = 0;
  for (auto& kv : uniq) kv.second = c++;
  assert((int64_t)res.first_rows.size() == (int64_t)uniq.size());
  for (int64_t r = 0; r < N; ++r) {
    if (res.codes[r] != uniq[vals[r]]) {
      std::fprintf(stderr, "MISMATCH row %lld: %d vs %d\n", (long long)r,
                   res.codes[r], uniq[vals[r]]);
      return 1;
    }
  }

  // string keys through the Arrow offsets/data layout
  std::vector<std::string> pool;
  for (int i = 0; i < 1000; ++i) pool.push_back("KEY" + std::to_string(i));
  std::string data;
  std::vector<int32_t> off{0};
  const int64_t NS = 500'000;
  for (int64_t i = 0; i < NS; ++i) {
    const std::string& s = pool[rng() % pool.size()];
    data += s;
    off.push_back((int32_t)data.size());
  }
  StrKey<int32_t> skey{off.data(), data.data(), nullptr, 0};
  FactorizeResult sres = factorize_impl(NS, skey);
  assert((int64_t)sres.first_rows.size() == 1000);
  // codes of equal strings must match
  std::map<std::string, int32_t> first_code;
  for (int64_t r = 0; r < NS; ++r) {
    std::string s(data.data() + off[r], (size_t)(off[r + 1] - off[r]));
    auto it = first_code.find(s);
    if (it == first_code.end())
      first_code.emplace(s, sres.codes[r]);
    else if (it->second != sres.codes[r]) {
      std::fprintf(stderr, "STR MISMATCH at %lld\n", (long long)r);
      return 1;
    }
  }

  std::printf("gather_tsan OK: %lld i64 rows (%lld uniques), %lld str rows\n",
              (long long)N, (long long)uniq.size(), (long long)NS);
  return 0;
}
