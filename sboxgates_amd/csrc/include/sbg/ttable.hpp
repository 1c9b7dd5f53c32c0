// ttable.hpp — 256-bit truth-table vector type, shared by host engine and
// CDNA4 device kernels.
//
// Bit layout parity with the reference (state.h:65-68, state.c:232-250):
// bit i of the table (i = S-box input pattern, 0..255) lives at bit (i % 64)
// of word (i / 64). The host type is a plain struct of 4 u64 words, 32-byte
// aligned so the `state` fingerprint (see state.hpp) is layout-compatible
// with the reference's GCC-vector-based struct.
#pragma once

#include "sbg/common.hpp"

namespace sbg {

struct alignas(32) ttable {
  u64 w[4];
};

SBG_HD inline ttable tt_make(u64 a, u64 b, u64 c, u64 d) {
  ttable t;
  t.w[0] = a; t.w[1] = b; t.w[2] = c; t.w[3] = d;
  return t;
}

SBG_HD inline ttable tt_zero_table() { return tt_make(0, 0, 0, 0); }
SBG_HD inline ttable tt_ones_table() {
  return tt_make(~0ULL, ~0ULL, ~0ULL, ~0ULL);
}

SBG_HD inline ttable operator&(const ttable& a, const ttable& b) {
  return tt_make(a.w[0] & b.w[0], a.w[1] & b.w[1], a.w[2] & b.w[2], a.w[3] & b.w[3]);
}
SBG_HD inline ttable operator|(const ttable& a, const ttable& b) {
  return tt_make(a.w[0] | b.w[0], a.w[1] | b.w[1], a.w[2] | b.w[2], a.w[3] | b.w[3]);
}
SBG_HD inline ttable operator^(const ttable& a, const ttable& b) {
  return tt_make(a.w[0] ^ b.w[0], a.w[1] ^ b.w[1], a.w[2] ^ b.w[2], a.w[3] ^ b.w[3]);
}
SBG_HD inline ttable operator~(const ttable& a) {
  return tt_make(~a.w[0], ~a.w[1], ~a.w[2], ~a.w[3]);
}
SBG_HD inline ttable& operator&=(ttable& a, const ttable& b) { a = a & b; return a; }
SBG_HD inline ttable& operator|=(ttable& a, const ttable& b) { a = a | b; return a; }
SBG_HD inline ttable& operator^=(ttable& a, const ttable& b) { a = a ^ b; return a; }

// True if all 256 bits are zero (reference parity: sboxgates.c:76-83).
SBG_HD inline bool tt_zero(const ttable& t) {
  return (t.w[0] | t.w[1] | t.w[2] | t.w[3]) == 0;
}
// True if at least one bit is set.
SBG_HD inline bool tt_any(const ttable& t) { return !tt_zero(t); }

SBG_HD inline bool tt_eq(const ttable& a, const ttable& b) {
  return tt_zero(a ^ b);
}
// Masked equality: only bit positions set in mask are compared
// (reference parity: sboxgates.c:91-93).
SBG_HD inline bool tt_eq_mask(const ttable& a, const ttable& b, const ttable& mask) {
  return tt_zero((a ^ b) & mask);
}

SBG_HD inline int tt_get_bit(const ttable& t, int i) {
  return static_cast<int>((t.w[i >> 6] >> (i & 63)) & 1);
}
SBG_HD inline void tt_set_bit(ttable& t, int i, int v) {
  const u64 m = 1ULL << (i & 63);
  if (v) t.w[i >> 6] |= m; else t.w[i >> 6] &= ~m;
}

SBG_HD inline int tt_popcount(const ttable& t) {
  return __builtin_popcountll(t.w[0]) + __builtin_popcountll(t.w[1]) +
         __builtin_popcountll(t.w[2]) + __builtin_popcountll(t.w[3]);
}

// Search mask for an n-input S-box: the first 2^n positions are valid
// (reference parity: sboxgates.c:644-659).
SBG_HD inline ttable tt_mask_for_inputs(int num_inputs) {
  ttable t = tt_ones_table();
  if (num_inputs < 8) t.w[2] = t.w[3] = 0;
  if (num_inputs < 7) t.w[1] = 0;
  if (num_inputs < 6) t.w[0] = (1ULL << (1u << num_inputs)) - 1;
  return t;
}

}  // namespace sbg
