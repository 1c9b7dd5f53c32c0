// rng.hpp — host RNG (xorshift1024*, matching the reference's generator
// family, sboxgates.c:246-268) plus a small stateless device-friendly hash
// used by kernels to randomize LUT don't-cares and coloring choices.
//
// Unlike the reference, the generator is an object (no global state) and can
// be deterministically seeded for reproducible tests (`--seed`). By default
// it is seeded from /dev/urandom, matching reference behavior
// (options.randomize is always on, sboxgates.c:1070).
#pragma once

#include <cstdio>
#include "sbg/common.hpp"

namespace sbg {

class Xorshift1024 {
 public:
  Xorshift1024() { seed_urandom(); }
  explicit Xorshift1024(u64 seed) { seed_splitmix(seed); }

  void seed_splitmix(u64 seed) {
    // Fill the 16-word pool with splitmix64 output; guarantees nonzero state.
    u64 x = seed + 0x9E3779B97F4A7C15ULL;
    for (int i = 0; i < 16; i++) {
      u64 z = (x += 0x9E3779B97F4A7C15ULL);
      z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
      z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
      s_[i] = z ^ (z >> 31);
    }
    p_ = 0;
  }

  void seed_urandom() {
    FILE* fp = std::fopen("/dev/urandom", "rb");
    bool ok = false;
    if (fp != nullptr) {
      ok = std::fread(s_, sizeof(s_), 1, fp) == 1;
      std::fclose(fp);
    }
    if (!ok) seed_splitmix(0x5bd1e995u);
    p_ = 0;
  }

  u64 next() {
    u64 s0 = s_[p_];
    p_ = (p_ + 1) & 15;
    u64 s1 = s_[p_];
    s1 ^= s1 << 31;
    s_[p_] = s1 ^ s0 ^ (s1 >> 11) ^ (s0 >> 30);
    return s_[p_] * 1181783497276652981ULL;
  }

  // Uniform in [0, n). Used for Fisher-Yates shuffles; modulo bias is
  // irrelevant for search-order randomization.
  u64 below(u64 n) { return next() % n; }

 private:
  u64 s_[16] = {};
  int p_ = 0;
};

// Stateless 64-bit mix (splitmix64 finalizer). Device kernels call this on
// (seed ^ candidate_index) to get per-candidate pseudo-random bits — the
// counter-based analog of the reference's shared xorshift stream; it only
// influences WHICH valid solution is produced, never validity.
SBG_HD inline u64 hash_mix64(u64 z) {
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
  return z ^ (z >> 31);
}

}  // namespace sbg
