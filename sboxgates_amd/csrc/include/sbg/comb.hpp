// comb.hpp — combinatorics: binomial coefficients, combination
// ranking/unranking and lexicographic successor. Host + device.
//
// Behavioral parity points with the reference: lut.c:761-770 (n_choose_k),
// lut.c:635-662 (get_nth_combination), lut.c:743-758 (next_combination).
// The implementations here are fresh; unranking is iterative (device-safe,
// no recursion) and there are closed-form pair/triple decoders used by the
// HIP kernels to map flat candidate indices to (d,e) / (e,f,g) tuples.
#pragma once

#include "sbg/common.hpp"

namespace sbg {

// C(n, k); exact for every intermediate (prefix products are binomials).
// Safe for n <= 500, k <= 7 (max intermediate ~1e16 < 2^63).
SBG_HD inline i64 n_choose_k(int n, int k) {
  if (k < 0 || k > n) return 0;
  i64 ret = 1;
  for (int i = 1; i <= k; i++) {
    ret *= (n - i + 1);
    ret /= i;
  }
  return ret;
}

// Writes the n-th (0-based, lexicographic) k-combination of
// {first, first+1, ..., first+num-1} into ret[0..k-1].
SBG_HD inline void nth_combination(i64 n, int num, int k, int first, gatenum* ret) {
  int base = first;
  int remaining = num;
  for (int slot = 0; slot < k; slot++) {
    // Choose the value for this slot: advance while the block of
    // combinations starting with `base` is entirely below n.
    for (;;) {
      i64 block = n_choose_k(remaining - 1, k - slot - 1);
      if (n < block) break;
      n -= block;
      base += 1;
      remaining -= 1;
    }
    ret[slot] = static_cast<gatenum>(base);
    base += 1;
    remaining -= 1;
  }
}

// Lexicographic rank of combination c[0..k-1] (ascending values in [0,num)).
SBG_HD inline i64 combination_rank(const gatenum* c, int k, int num) {
  i64 rank = 0;
  int prev = -1;
  for (int slot = 0; slot < k; slot++) {
    for (int v = prev + 1; v < c[slot]; v++) {
      rank += n_choose_k(num - v - 1, k - slot - 1);
    }
    prev = c[slot];
  }
  return rank;
}

// Advances combination[0..t-1] (values in [0, max)) to its lexicographic
// successor; leaves it unchanged if it is the last combination.
SBG_HD inline void next_combination(gatenum* combination, int t, int max) {
  int i = t - 1;
  while (i >= 0 && combination[i] + t - i >= max) i--;
  if (i < 0) return;
  combination[i] += 1;
  for (int k = i + 1; k < t; k++) {
    combination[k] = static_cast<gatenum>(combination[k - 1] + 1);
  }
}

// Decodes flat pair index q in [0, C(m,2)) into 0 <= d < e < m,
// lexicographic order (d major): q = sum_{j<d} (m-1-j) + (e-d-1).
// Uses the triangular-number inverse with an exact integer fix-up.
SBG_HD inline void decode_pair(i64 q, int m, int* d, int* e) {
  // Solve for d: largest d with S(d) <= q where S(d) = d*m - d*(d+1)/2.
  // Double-precision sqrt gets within 1; fix up exactly.
  double mm = static_cast<double>(m) - 0.5;
  int dd = static_cast<int>(mm - 0.5 -
      __builtin_sqrt(mm * mm - 2.0 * static_cast<double>(q) - 0.75));
  if (dd < 0) dd = 0;
  // S(d) with 64-bit math.
  auto S = [m](i64 d) { return d * static_cast<i64>(m) - d * (d + 1) / 2; };
  while (dd > 0 && S(dd) > q) dd--;
  while (S(dd + 1) <= q) dd++;
  *d = dd;
  *e = static_cast<int>(q - S(dd)) + dd + 1;
}

}  // namespace sbg
