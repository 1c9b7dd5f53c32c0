// lutcover.hpp — cell algebra for LUT feasibility and function derivation.
// Shared by the host engine and the CDNA4 HIP kernels.
//
// Core idea (new in this implementation; the reference brute-forces these):
// for k input truth tables, the 256-bit position space partitions into 2^k
// *cells* (one per input-value pattern). Under target+mask, each cell is
// "forced 1" (contains a masked position with target=1), "forced 0", free,
// or contradictory (both). Then:
//
//   * k-LUT feasibility (reference check_n_lut_possible, lut.c:34-66)
//     <=> no cell is contradictory.
//   * 3-LUT function derivation (reference get_lut_function, lut.c:79-109,
//     a 256-step bit-serial loop) <=> read the forced bits off the cells
//     directly; don't-cares filled randomly.
//   * 5-LUT decomposition LUT(LUT(a,b,c),d,e) for a given 3+2 split
//     (reference: 256 outer functions x bit-serial derive, lut.c:174-246)
//     <=> a proper 2-coloring of an 8-node conflict graph over the outer
//     triple's cells: cells u,u' conflict iff merging them (same outer-LUT
//     output) would mix a forced-1 with a forced-0 in some (d,e) layer.
//     Valid outer functions are exactly the proper 2-colorings, so one
//     bipartiteness check replaces the 256-function scan.
//   * 7-LUT LUT(LUT(a,b,c),LUT(d,e,f),g) for a given (3,3,1) ordering
//     (reference: 256x256 function pairs, lut.c:416-484) <=> for each
//     middle function (256), a 4-layer 2-coloring for the outer function.
//
// Randomization (which valid solution is returned) comes from a caller-
// supplied 64-bit random word, matching the reference's "randomize
// don't-cares / shuffle function order" behavior in effect: any valid
// solution may be produced, never an invalid one.
#pragma once

#include "sbg/comb.hpp"
#include "sbg/common.hpp"
#include "sbg/ttable.hpp"

namespace sbg {

// ---------------------------------------------------------------------------
// Cell-mask construction. p1 bit c = "cell c contains a masked target-1
// position"; p0 bit c = same for target-0. T1 = target & mask,
// T0 = ~target & mask. Cell index convention: bit (k-1-j) of c is the value
// of input j — so for k=3, c = a<<2 | b<<1 | c, matching the LUT function
// byte (gen_lut_ttable).
// ---------------------------------------------------------------------------

// k=3. Returns false on contradiction (some cell forced both ways).
SBG_HD inline bool lut3_p_masks(const ttable& ta, const ttable& tb, const ttable& tc,
                                const ttable& T1, const ttable& T0,
                                u32* p1_out, u32* p0_out) {
  u32 p1 = 0, p0 = 0;
  for (int c = 0; c < 8; c++) {
    ttable cell = (c & 4 ? ta : ~ta) & (c & 2 ? tb : ~tb) & (c & 1 ? tc : ~tc);
    bool has1 = tt_any(cell & T1);
    bool has0 = tt_any(cell & T0);
    if (has1 && has0) return false;
    if (has1) p1 |= 1u << c;
    if (has0) p0 |= 1u << c;
  }
  *p1_out = p1;
  *p0_out = p0;
  return true;
}

// k=5 over tables t[0..4]; c bit 4 = t0's value, ..., bit 0 = t4's value.
SBG_HD inline bool lut5_p_masks(const ttable* t, const ttable& T1, const ttable& T0,
                                u32* p1_out, u32* p0_out) {
  u32 p1 = 0, p0 = 0;
  for (int c = 0; c < 32; c++) {
    ttable cell = (c & 16 ? t[0] : ~t[0]) & (c & 8 ? t[1] : ~t[1]) &
                  (c & 4 ? t[2] : ~t[2]) & (c & 2 ? t[3] : ~t[3]) &
                  (c & 1 ? t[4] : ~t[4]);
    bool has1 = tt_any(cell & T1);
    bool has0 = tt_any(cell & T0);
    if (has1 && has0) return false;
    if (has1) p1 |= 1u << c;
    if (has0) p0 |= 1u << c;
  }
  *p1_out = p1;
  *p0_out = p0;
  return true;
}

// k=7 over tables t[0..6]; cell bit 6 = t0's value, ..., bit 0 = t6's value.
// p masks are 128 bits: pX[0] holds cells 0-63, pX[1] cells 64-127.
SBG_HD inline bool lut7_p_masks(const ttable* t, const ttable& T1, const ttable& T0,
                                u64 p1_out[2], u64 p0_out[2]) {
  u64 p1[2] = {0, 0}, p0[2] = {0, 0};
  for (int c = 0; c < 128; c++) {
    ttable cell = (c & 64 ? t[0] : ~t[0]) & (c & 32 ? t[1] : ~t[1]) &
                  (c & 16 ? t[2] : ~t[2]) & (c & 8 ? t[3] : ~t[3]) &
                  (c & 4 ? t[4] : ~t[4]) & (c & 2 ? t[5] : ~t[5]) &
                  (c & 1 ? t[6] : ~t[6]);
    bool has1 = tt_any(cell & T1);
    bool has0 = tt_any(cell & T0);
    if (has1 && has0) return false;
    if (has1) p1[c >> 6] |= 1ULL << (c & 63);
    if (has0) p0[c >> 6] |= 1ULL << (c & 63);
  }
  p1_out[0] = p1[0]; p1_out[1] = p1[1];
  p0_out[0] = p0[0]; p0_out[1] = p0[1];
  return true;
}

// 3-LUT function from cell masks: forced-1 bits, don't-cares randomized.
// Guarantees a NONZERO byte whenever any don't-care exists: function 00
// does not round-trip through the gates.xsd loader (the reference's own
// validator rejects func <= 0, state.c:300 — an upstream latent bug its
// randomizer can also trigger). A forced-all-zero function returns 0 and
// callers treat the candidate as a miss (a constant-FALSE LUT is never a
// useful solution).
SBG_HD inline u8 lut3_function_from_p(u32 p1, u32 p0, u64 rnd) {
  u8 dontcare = static_cast<u8>(~(p1 | p0) & 0xff);
  u8 f = static_cast<u8>((p1 & 0xff) | (dontcare & static_cast<u8>(rnd)));
  if (f == 0 && dontcare != 0) f = static_cast<u8>(dontcare & (-dontcare));
  return f;
}

// ---------------------------------------------------------------------------
// 2-coloring solver over <=8 nodes with "must differ" edges given as
// adjacency bitmasks. Returns false if not bipartite; otherwise writes an
// 8-bit coloring. Nodes in `touched` with no edges keep a random color;
// per-component side choice is randomized from rnd.
// ---------------------------------------------------------------------------
SBG_HD inline bool two_color_8(const u8 adj[8], u64 rnd, u8* coloring_out) {
  u8 color1 = 0;     // nodes colored 1
  u8 visited = 0;
  int rbit = 0;
  for (int s = 0; s < 8; s++) {
    if (visited & (1u << s)) continue;
    if (adj[s] == 0) {
      // Isolated node: free choice.
      visited |= 1u << s;
      if ((rnd >> (rbit++ & 63)) & 1) color1 |= 1u << s;
      continue;
    }
    // BFS this component; seed color randomized.
    u8 frontier = 1u << s;
    u8 comp_seen = 1u << s;
    u8 comp_color1 = ((rnd >> (rbit++ & 63)) & 1) ? (1u << s) : 0;
    while (frontier != 0) {
      u8 next = 0;
      for (int u = 0; u < 8; u++) {
        if (!(frontier & (1u << u))) continue;
        bool u_is1 = (comp_color1 >> u) & 1;
        u8 nbrs = adj[u];
        // Conflict check: a neighbor already colored like u -> not bipartite.
        u8 same = u_is1 ? comp_color1 : static_cast<u8>(comp_seen & ~comp_color1);
        if (nbrs & same) return false;
        u8 fresh = nbrs & static_cast<u8>(~comp_seen);
        comp_seen |= fresh;
        next |= fresh;
        if (!u_is1) comp_color1 |= fresh;  // neighbors get the opposite color
      }
      frontier = next;
    }
    visited |= comp_seen;
    color1 |= comp_color1;
  }
  *coloring_out = color1;
  return true;
}

// ---------------------------------------------------------------------------
// 5-LUT decomposition solve. Given p1/p0 over the 32 cells (no
// contradictions), finds a split s (0..9: which 3 of the 5 inputs feed the
// outer LUT), an outer function fo and an inner function fi such that
// LUT(fi; LUT(fo; t[o0],t[o1],t[o2]), t[r0], t[r1]) matches target under
// mask. Returns true on success; outputs the split index and functions.
// SPLITS5[s] = {o0,o1,o2,r0,r1}, ascending triples/pairs — matches the
// reference's enumeration (lut.c:186-246) so any reference-findable
// decomposition is found.
// ---------------------------------------------------------------------------
constexpr u8 SPLITS5[10][5] = {
    {0, 1, 2, 3, 4}, {0, 1, 3, 2, 4}, {0, 1, 4, 2, 3}, {0, 2, 3, 1, 4},
    {0, 2, 4, 1, 3}, {0, 3, 4, 1, 2}, {1, 2, 3, 0, 4}, {1, 2, 4, 0, 3},
    {1, 3, 4, 0, 2}, {2, 3, 4, 0, 1}};

SBG_HD inline bool lut5_solve_from_p(u32 p1, u32 p0, u64 rnd, u8* fo_out,
                                     u8* fi_out, int* split_out) {
  for (int s = 0; s < 10; s++) {
    const u8* sp = SPLITS5[s];
    // Per (u = outer-triple cell, v = rest-pair cell): layer masks.
    // A1[v] bit u = p1 of 5-cell (u,v); likewise A0.
    u8 A1[4] = {0, 0, 0, 0};
    u8 A0[4] = {0, 0, 0, 0};
    for (int c = 0; c < 32; c++) {
      if (!((p1 >> c) & 1) && !((p0 >> c) & 1)) continue;
      int u = (((c >> (4 - sp[0])) & 1) << 2) | (((c >> (4 - sp[1])) & 1) << 1) |
              ((c >> (4 - sp[2])) & 1);
      int v = (((c >> (4 - sp[3])) & 1) << 1) | ((c >> (4 - sp[4])) & 1);
      if ((p1 >> c) & 1) A1[v] |= 1u << u;
      if ((p0 >> c) & 1) A0[v] |= 1u << u;
    }
    // Conflict graph: u and u' must differ if some layer v has p1 at u and
    // p0 at u' (or vice versa).
    u8 adj[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int v = 0; v < 4; v++) {
      for (int u = 0; u < 8; u++) {
        u8 m = 0;
        if ((A1[v] >> u) & 1) m |= A0[v];
        if ((A0[v] >> u) & 1) m |= A1[v];
        adj[u] |= m;
      }
    }
    for (int u = 0; u < 8; u++) adj[u] &= static_cast<u8>(~(1u << u));
    u8 fo;
    if (!two_color_8(adj, rnd, &fo)) continue;
    // function 00 does not round-trip (see lut3_function_from_p): the
    // complement of a proper coloring is proper, so an all-zero outer
    // byte can always be flipped.
    if (fo == 0) fo = 0xff;
    // Inner function over cells (o, v): inner pattern = o<<2 | v.
    u8 fi = 0, forced = 0;
    for (int v = 0; v < 4; v++) {
      u8 sel1 = fo;                      // cells colored 1
      u8 sel0 = static_cast<u8>(~fo);    // cells colored 0
      // o = 1 class:
      if (A1[v] & sel1) { fi |= 1u << (4 | v); forced |= 1u << (4 | v); }
      if (A0[v] & sel1) { forced |= 1u << (4 | v); }
      // o = 0 class:
      if (A1[v] & sel0) { fi |= 1u << v; forced |= 1u << v; }
      if (A0[v] & sel0) { forced |= 1u << v; }
    }
    fi |= static_cast<u8>(~forced) & static_cast<u8>(rnd >> 24);
    if (fi == 0) {
      if (forced != 0xff) {
        u8 dc = static_cast<u8>(~forced);
        fi = static_cast<u8>(dc & (-dc));
      } else {
        continue;  // inner forced identically 0: degenerate, try next split
      }
    }
    *fo_out = fo;
    *fi_out = fi;
    *split_out = s;
    return true;
  }
  return false;
}

// ---------------------------------------------------------------------------
// 7-LUT ordering table: all ways to pick the outer triple and middle triple
// from 7 inputs (g = the one left over), with the outer/middle symmetry
// deduped by requiring outer < middle lexicographically: 70 orderings, the
// same family the reference hard-codes (lut.c:396-415).
// Each row: {a,b,c, d,e,f, g} — indices into the 7-combination.
// ---------------------------------------------------------------------------
SBG_HD inline void lut7_ordering(int idx, u8 ord[7]) {
  // Enumerate ascending triples T1 < T2 (lexicographic on sorted triples)
  // from {0..6} with T1 ∩ T2 = ∅. idx in [0, 70).
  int count = 0;
  for (int a = 0; a < 7; a++)
  for (int b = a + 1; b < 7; b++)
  for (int c = b + 1; c < 7; c++) {
    u8 used = static_cast<u8>((1 << a) | (1 << b) | (1 << c));
    for (int d = 0; d < 7; d++) {
      if (used & (1 << d)) continue;
      for (int e = d + 1; e < 7; e++) {
        if (used & (1 << e)) continue;
        for (int f = e + 1; f < 7; f++) {
          if (used & (1 << f)) continue;
          // Dedup outer/middle swap: keep only outer < middle (first
          // element comparison suffices since triples are disjoint).
          if (a > d) continue;
          if (count == idx) {
            int g = 0;
            u8 all = static_cast<u8>(used | (1 << d) | (1 << e) | (1 << f));
            while (all & (1 << g)) g++;
            ord[0] = static_cast<u8>(a); ord[1] = static_cast<u8>(b);
            ord[2] = static_cast<u8>(c); ord[3] = static_cast<u8>(d);
            ord[4] = static_cast<u8>(e); ord[5] = static_cast<u8>(f);
            ord[6] = static_cast<u8>(g);
            return;
          }
          count++;
        }
      }
    }
  }
  // idx out of range: leave identity.
  for (int i = 0; i < 7; i++) ord[i] = static_cast<u8>(i);
}

constexpr int LUT7_NUM_ORDERINGS = 70;

// 7-LUT solve for ONE ordering. p1/p0 are the 128-cell masks; ord indexes
// into the 7 inputs: outer = (ord[0..2]), middle = (ord[3..5]), g = ord[6].
// Tries all 256 middle functions (starting at a randomized offset); for
// each, solves the outer function by 4-layer 2-coloring. On success writes
// fo (outer), fm (middle), fi (inner).
SBG_HD inline bool lut7_solve_ordering(const u64 p1[2], const u64 p0[2],
                                       const u8 ord[7], u64 rnd, u8* fo_out,
                                       u8* fm_out, u8* fi_out,
                                       int fm_offset = 0, int fm_count = 256) {
  // Aggregate to (u = outer cell, w = middle cell, g): bit w of B1[u][g].
  u8 B1[8][2] = {};
  u8 B0[8][2] = {};
  for (int c = 0; c < 128; c++) {
    bool b1 = (p1[c >> 6] >> (c & 63)) & 1;
    bool b0 = (p0[c >> 6] >> (c & 63)) & 1;
    if (!b1 && !b0) continue;
    int u = (((c >> (6 - ord[0])) & 1) << 2) | (((c >> (6 - ord[1])) & 1) << 1) |
            ((c >> (6 - ord[2])) & 1);
    int w = (((c >> (6 - ord[3])) & 1) << 2) | (((c >> (6 - ord[4])) & 1) << 1) |
            ((c >> (6 - ord[5])) & 1);
    int gg = (c >> (6 - ord[6])) & 1;
    if (b1) B1[u][gg] |= 1u << w;
    if (b0) B0[u][gg] |= 1u << w;
  }

  for (int fmi = fm_offset; fmi < fm_offset + fm_count; fmi++) {
    u8 fm = static_cast<u8>((fmi + (rnd >> 32)) & 0xff);
    // fm = 00 does not round-trip; a constant-0 middle input is always
    // expressible as fm = ff with the inner function adapting, which the
    // sweep reaches anyway.
    if (fm == 0) continue;
    // Layers for the outer coloring: (m, g) in {0,1}^2.
    // L1[layer] bit u = exists middle-cell w with fm-class m and p1 set.
    u8 L1[4] = {0, 0, 0, 0};
    u8 L0[4] = {0, 0, 0, 0};
    for (int u = 0; u < 8; u++) {
      for (int gg = 0; gg < 2; gg++) {
        if (B1[u][gg] & fm)  L1[2 | gg] |= 1u << u;
        if (B0[u][gg] & fm)  L0[2 | gg] |= 1u << u;
        if (B1[u][gg] & static_cast<u8>(~fm)) L1[gg] |= 1u << u;
        if (B0[u][gg] & static_cast<u8>(~fm)) L0[gg] |= 1u << u;
      }
    }
    // Self-contradiction: a (u, m, g) bucket forced both ways can never be
    // separated by the outer function (u fixed) -> this fm fails.
    bool self_bad = false;
    u8 adj[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int layer = 0; layer < 4 && !self_bad; layer++) {
      if (L1[layer] & L0[layer]) { self_bad = true; break; }
      for (int u = 0; u < 8; u++) {
        u8 m = 0;
        if ((L1[layer] >> u) & 1) m |= L0[layer];
        if ((L0[layer] >> u) & 1) m |= L1[layer];
        adj[u] |= m;
      }
    }
    if (self_bad) continue;
    for (int u = 0; u < 8; u++) adj[u] &= static_cast<u8>(~(1u << u));
    u8 fo;
    if (!two_color_8(adj, rnd ^ (static_cast<u64>(fm) * 0x9E3779B97F4A7C15ULL), &fo)) {
      continue;
    }
    if (fo == 0) fo = 0xff;  // complement coloring: function 00 never emitted
    // Inner function over (o, m, g): pattern = o<<2 | m<<1 | g.
    u8 fi = 0, forced = 0;
    for (int m = 0; m < 2; m++) {
      for (int gg = 0; gg < 2; gg++) {
        int layer = (m << 1) | gg;
        u8 sel1 = fo, sel0 = static_cast<u8>(~fo);
        int pat1 = 4 | (m << 1) | gg;
        int pat0 = (m << 1) | gg;
        if (L1[layer] & sel1) { fi |= 1u << pat1; forced |= 1u << pat1; }
        if (L0[layer] & sel1) { forced |= 1u << pat1; }
        if (L1[layer] & sel0) { fi |= 1u << pat0; forced |= 1u << pat0; }
        if (L0[layer] & sel0) { forced |= 1u << pat0; }
      }
    }
    fi |= static_cast<u8>(~forced) & static_cast<u8>(rnd >> 16);
    if (fi == 0) {
      if (forced != 0xff) {
        u8 dc = static_cast<u8>(~forced);
        fi = static_cast<u8>(dc & (-dc));
      } else {
        continue;  // inner forced identically 0: degenerate, next fm
      }
    }
    *fo_out = fo;
    *fm_out = fm;
    *fi_out = fi;
    return true;
  }
  return false;
}

}  // namespace sbg
