// CHWBL hash ring in C++ — the control plane's hottest loop.
//
// Reference analog: internal/loadbalancer/balance_chwbl.go (xxhash ring,
// vnode replication, binary-search successor, bounded-load walk). The
// gateway calls lookup() once per routed request; the Python fallback in
// controlplane/loadbalancer.py implements identical semantics (the
// equivalence is property-tested in tests/test_chwbl_native.py).
#include <torch/extension.h>

#include <algorithm>
#include <cstdint>
#include <string>
#include <tuple>
#include <vector>

namespace {

// ---- xxHash64 (same spec as kubeai_amd/utils/xxhash64.py) ----
constexpr uint64_t P1 = 0x9E3779B185EBCA87ull;
constexpr uint64_t P2 = 0xC2B2AE3D27D4EB4Full;
constexpr uint64_t P3 = 0x165667B19E3779F9ull;
constexpr uint64_t P4 = 0x85EBCA77C2B2AE63ull;
constexpr uint64_t P5 = 0x27D4EB2F165667C5ull;

inline uint64_t rotl(uint64_t x, int r) { return (x << r) | (x >> (64 - r)); }

inline uint64_t round_(uint64_t acc, uint64_t lane) {
  return rotl(acc + lane * P2, 31) * P1;
}

inline uint64_t merge_round(uint64_t acc, uint64_t val) {
  return (acc ^ round_(0, val)) * P1 + P4;
}

inline uint64_t read64(const char* p) {
  uint64_t v;
  memcpy(&v, p, 8);
  return v;
}

inline uint32_t read32(const char* p) {
  uint32_t v;
  memcpy(&v, p, 4);
  return v;
}

uint64_t xxh64(const char* data, size_t n, uint64_t seed = 0) {
  const char* p = data;
  const char* end = data + n;
  uint64_t h;
  if (n >= 32) {
    uint64_t v1 = seed + P1 + P2, v2 = seed + P2, v3 = seed, v4 = seed - P1;
    const char* limit = end - 32;
    do {
      v1 = round_(v1, read64(p));
      v2 = round_(v2, read64(p + 8));
      v3 = round_(v3, read64(p + 16));
      v4 = round_(v4, read64(p + 24));
      p += 32;
    } while (p <= limit);
    h = rotl(v1, 1) + rotl(v2, 7) + rotl(v3, 12) + rotl(v4, 18);
    h = merge_round(h, v1);
    h = merge_round(h, v2);
    h = merge_round(h, v3);
    h = merge_round(h, v4);
  } else {
    h = seed + P5;
  }
  h += (uint64_t)n;
  while (p + 8 <= end) {
    h = rotl(h ^ round_(0, read64(p)), 27) * P1 + P4;
    p += 8;
  }
  if (p + 4 <= end) {
    h = rotl(h ^ (uint64_t)read32(p) * P1, 23) * P2 + P3;
    p += 4;
  }
  while (p < end) {
    h = rotl(h ^ (uint8_t)(*p) * P5, 11) * P1;
    ++p;
  }
  h ^= h >> 33;
  h *= P2;
  h ^= h >> 29;
  h *= P3;
  h ^= h >> 32;
  return h;
}

struct ChwblRing {
  std::vector<uint64_t> hashes;  // sorted vnode hashes
  std::vector<int32_t> owner;    // endpoint index per vnode
  int n_endpoints = 0;

  void rebuild(const std::vector<std::string>& endpoints, int64_t replication) {
    n_endpoints = (int)endpoints.size();
    std::vector<std::pair<uint64_t, int32_t>> ring;
    ring.reserve(endpoints.size() * replication);
    for (int32_t e = 0; e < (int32_t)endpoints.size(); ++e) {
      for (int64_t i = 0; i < replication; ++i) {
        // vnode key = addr + str(i) (balance_chwbl.go:140-150 semantics)
        std::string key = endpoints[e] + std::to_string(i);
        ring.emplace_back(xxh64(key.data(), key.size()), e);
      }
    }
    // tie-break matches Python's (hash, addr) sort: compare endpoint name
    std::sort(ring.begin(), ring.end(),
              [&](const auto& a, const auto& b) {
                if (a.first != b.first) return a.first < b.first;
                return endpoints[a.second] < endpoints[b.second];
              });
    hashes.resize(ring.size());
    owner.resize(ring.size());
    for (size_t i = 0; i < ring.size(); ++i) {
      hashes[i] = ring[i].first;
      owner[i] = ring[i].second;
    }
  }

  // returns (endpoint_idx, iterations, defaulted). allowed[e]: adapter
  // filter; loads[e]: in-flight; load_ok: load+1 <= (total+1)/n * factor.
  std::tuple<int64_t, int64_t, bool> lookup(
      const std::string& key, const std::vector<int64_t>& loads,
      int64_t total_load, double load_factor,
      const std::vector<bool>& allowed) const {
    if (hashes.empty() || n_endpoints == 0) return {-1, 0, true};
    const uint64_t h = xxh64(key.data(), key.size());
    size_t i =
        std::lower_bound(hashes.begin(), hashes.end(), h) - hashes.begin();
    if (i == hashes.size()) i = 0;
    const size_t i0 = i;
    // reference chwblLoadOK: totalLoad==0 always OK; else
    // load <= (total+1)/n * loadFactor
    const double bound =
        (double)(total_load + 1) / (double)n_endpoints * load_factor;
    int64_t first = -1;
    int64_t iters = 0;
    while (true) {
      ++iters;
      const int32_t e = owner[i];
      if (allowed[e]) {
        if (first < 0) first = e;
        if (total_load == 0 || (double)loads[e] <= bound) return {e, iters, false};
      }
      i = (i + 1) % hashes.size();
      if (i == i0) return {first, iters, true};
    }
  }
};

}  // namespace

void register_chwbl(pybind11::module_& m) {
  namespace py = pybind11;
  m.def("xxh64", [](py::bytes data, uint64_t seed) {
    std::string s = data;
    return xxh64(s.data(), s.size(), seed);
  }, py::arg("data"), py::arg("seed") = 0);
  py::class_<ChwblRing>(m, "ChwblRing")
      .def(py::init<>())
      .def("rebuild", &ChwblRing::rebuild)
      .def("lookup", &ChwblRing::lookup);
}
