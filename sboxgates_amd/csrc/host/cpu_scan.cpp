// cpu_scan.cpp — CPU implementations of the 3/5/7-LUT combination scans
// (no-GPU execution path + oracle for the HIP kernels), plus the naive
// reference-style checkers used as independent test oracles.

#include "sbg/scan.hpp"

#include <cstring>

#include "sbg/comb.hpp"
#include "sbg/lutcover.hpp"
#include "sbg/rng.hpp"

namespace sbg {

// Pair-prefix cells for the k=3-shaped scans: lexicographic combination
// order increments the third gate fastest, so the four (a,b) pair cells
// (intersected with T1/T0, with liveness flags) are computed once per
// prefix and each c costs only the final split — the same prefix-sharing
// the gfx950 kernels use, on the CPU path.
struct PairCells {
  ttable H1[4], H0[4];
  u8 nz1 = 0, nz0 = 0;

  void fill(const ttable& ta, const ttable& tb, const ttable& T1,
            const ttable& T0) {
    nz1 = nz0 = 0;
    for (int u = 0; u < 4; u++) {
      ttable pc = (u & 2 ? ta : ~ta) & (u & 1 ? tb : ~tb);
      H1[u] = pc & T1;
      H0[u] = pc & T0;
      if (tt_any(H1[u])) nz1 |= static_cast<u8>(1u << u);
      if (tt_any(H0[u])) nz0 |= static_cast<u8>(1u << u);
    }
  }

  // Equivalent of lut3_p_masks(ta, tb, tc, ...) under this prefix.
  bool p_masks(const ttable& tc, u32* p1_out, u32* p0_out) const {
    u32 p1 = 0, p0 = 0;
    for (int u = 0; u < 4; u++) {
      const bool l1 = (nz1 >> u) & 1, l0 = (nz0 >> u) & 1;
      if (!l1 && !l0) continue;
      u32 c1 = 0, c0 = 0;  // cell bits (u<<1)|vc with content
      if (l1) {
        ttable x = H1[u] & tc;
        if (tt_any(x)) c1 |= 2;
        if (tt_any(H1[u] ^ x)) c1 |= 1;
      }
      if (l0) {
        ttable x = H0[u] & tc;
        if (tt_any(x)) c0 |= 2;
        if (tt_any(H0[u] ^ x)) c0 |= 1;
      }
      if (c1 & c0) return false;  // a cell forced both ways
      p1 |= c1 << (u << 1);
      p0 |= c0 << (u << 1);
    }
    *p1_out = p1;
    *p0_out = p0;
    return true;
  }
};

static inline bool excluded(const ScanRequest& rq, const gatenum* nums, int k) {
  if (rq.excl_low64 == 0) return false;
  for (int i = 0; i < k; i++) {
    if (nums[i] < 64 && (rq.excl_low64 >> nums[i]) & 1) return true;
  }
  return false;
}

ScanResult cpu_scan3(const ScanRequest& rq, i64 begin, i64 end) {
  ScanResult out;
  const i64 total = n_choose_k(rq.n, 3);
  if (begin >= total) return out;
  if (end > total) end = total;

  const ttable T1 = rq.target & rq.mask;
  const ttable T0 = ~rq.target & rq.mask;

  gatenum nums[3];
  nth_combination(begin, rq.n, 3, 0, nums);
  PairCells pc;
  int pa = -1, pb = -1;
  for (i64 i = begin; i < end; i++) {
    out.evaluated++;
    // NOTE: the reference's 3-LUT scan does not reject inbits combinations
    // (lut.c:501-523) — parity kept: no exclusion here.
    if (nums[0] != pa || nums[1] != pb) {
      pa = nums[0];
      pb = nums[1];
      pc.fill(rq.tables[pa], rq.tables[pb], T1, T0);
    }
    u32 p1, p0;
    if (pc.p_masks(rq.tables[nums[2]], &p1, &p0)) {
      u8 func = lut3_function_from_p(p1, p0, hash_mix64(rq.seed ^ static_cast<u64>(i)));
      if (func != 0 && !rq.count_all) {
        out.found = true;
        out.res[0] = func;
        out.res[1] = nums[0];
        out.res[2] = nums[1];
        out.res[3] = nums[2];
        return out;
      }
    }
    next_combination(nums, 3, rq.n);
  }
  return out;
}

ScanResult cpu_scan5(const ScanRequest& rq, i64 begin, i64 end) {
  ScanResult out;
  const i64 total = n_choose_k(rq.n, 5);
  if (begin >= total) return out;
  if (end > total) end = total;

  const ttable T1 = rq.target & rq.mask;
  const ttable T0 = ~rq.target & rq.mask;

  gatenum nums[5];
  nth_combination(begin, rq.n, 5, 0, nums);
  ttable tt[5];
  for (int j = 0; j < 5; j++) tt[j] = rq.tables[nums[j]];

  for (i64 i = begin; i < end; i++) {
    if (!excluded(rq, nums, 5)) {
      out.evaluated++;
      u32 p1, p0;
      if (lut5_p_masks(tt, T1, T0, &p1, &p0)) {
        u8 fo, fi;
        int split;
        u64 rnd = hash_mix64(rq.seed ^ static_cast<u64>(i));
        if (lut5_solve_from_p(p1, p0, rnd, &fo, &fi, &split) && !rq.count_all) {
          const u8* sp = SPLITS5[split];
          out.found = true;
          out.res[0] = fo;
          out.res[1] = fi;
          for (int j = 0; j < 3; j++) out.res[2 + j] = nums[sp[j]];
          out.res[5] = nums[sp[3]];
          out.res[6] = nums[sp[4]];
          return out;
        }
      }
    }
    next_combination(nums, 5, rq.n);
    for (int j = 0; j < 5; j++) tt[j] = rq.tables[nums[j]];
  }
  return out;
}

ScanResult cpu_scan7(const ScanRequest& rq, i64 begin, i64 end) {
  ScanResult out;
  const i64 total = n_choose_k(rq.n, 7);
  if (begin >= total) return out;
  if (end > total) end = total;

  const ttable T1 = rq.target & rq.mask;
  const ttable T0 = ~rq.target & rq.mask;

  gatenum nums[7];
  nth_combination(begin, rq.n, 7, 0, nums);
  ttable tt[7];
  for (int j = 0; j < 7; j++) tt[j] = rq.tables[nums[j]];

  for (i64 i = begin; i < end; i++) {
    if (!excluded(rq, nums, 7)) {
      out.evaluated++;
      u64 p1[2], p0[2];
      if (lut7_p_masks(tt, T1, T0, p1, p0)) {
        u64 rnd = hash_mix64(rq.seed ^ static_cast<u64>(i));
        for (int o = 0; o < LUT7_NUM_ORDERINGS; o++) {
          u8 ord[7];
          lut7_ordering(o, ord);
          u8 fo, fm, fi;
          if (lut7_solve_ordering(p1, p0, ord, rnd, &fo, &fm, &fi)) {
            if (!rq.count_all) {
              out.found = true;
              out.res[0] = fo;
              out.res[1] = fm;
              out.res[2] = fi;
              for (int j = 0; j < 7; j++) out.res[3 + j] = nums[ord[j]];
              return out;
            }
            break;
          }
        }
      }
    }
    next_combination(nums, 7, rq.n);
    for (int j = 0; j < 7; j++) tt[j] = rq.tables[nums[j]];
  }
  return out;
}

// --- Gate-mode step-4 triple scan (k = 4) ---

const int TRIPLE_PERMS6[6][3] = {{0, 1, 2}, {0, 2, 1}, {1, 0, 2},
                                 {1, 2, 0}, {2, 0, 1}, {2, 1, 0}};

u8 permute_cells8_host(u8 m, const int* sel) {
  u8 out = 0;
  for (int c = 0; c < 8; c++) {
    if (!((m >> c) & 1)) continue;
    int v[3] = {(c >> 2) & 1, (c >> 1) & 1, c & 1};
    out |= static_cast<u8>(1u << ((v[sel[0]] << 2) | (v[sel[1]] << 1) | v[sel[2]]));
  }
  return out;
}

void build_avail3_matcher(const u8* funs, const u8* costs, int count,
                          Avail3Matcher* out) {
  std::memset(out->bitmap, 0, sizeof(out->bitmap));
  out->count = count;
  for (int i = 0; i < count; i++) {
    out->funs[i] = funs[i];
    out->cost[i] = costs != nullptr ? costs[i] : 1;
  }
  // bitmap[care][req1] = exists f with (f & care) == req1. Enumerate per
  // function: for each care, the satisfied req1 is f & care.
  for (int i = 0; i < count; i++) {
    for (int care = 0; care < 256; care++) {
      int req1 = funs[i] & care;
      int idx = care * 256 + req1;
      out->bitmap[idx >> 3] |= static_cast<u8>(1u << (idx & 7));
    }
  }
}

ScanResult cpu_scan4(const ScanRequest& rq, i64 begin, i64 end) {
  ScanResult out;
  const i64 total = n_choose_k(rq.n, 3);
  if (begin >= total) return out;
  if (end > total) end = total;
  const Avail3Matcher* M = rq.matcher;

  const ttable T1 = rq.target & rq.mask;
  const ttable T0 = ~rq.target & rq.mask;

  gatenum nums[3];
  nth_combination(begin, rq.n, 3, 0, nums);
  PairCells pc;
  int pa = -1, pb = -1;
  for (i64 i = begin; i < end; i++) {
    out.evaluated++;
    if (nums[0] != pa || nums[1] != pb) {
      pa = nums[0];
      pb = nums[1];
      pc.fill(rq.tables[pa], rq.tables[pb], T1, T0);
    }
    u32 p1, p0;
    if (pc.p_masks(rq.tables[nums[2]], &p1, &p0)) {
      const u8 req1 = static_cast<u8>(p1);
      const u8 care = static_cast<u8>(p1 | p0);
      for (int perm = 0; perm < 6 && !rq.count_all; perm++) {
        const u8 r = permute_cells8_host(req1, TRIPLE_PERMS6[perm]);
        const u8 c = permute_cells8_host(care, TRIPLE_PERMS6[perm]);
        const int idx = c * 256 + r;
        if (!((M->bitmap[idx >> 3] >> (idx & 7)) & 1)) continue;
        for (int f = 0; f < M->count; f++) {
          if ((M->funs[f] & c) == r) {
            out.found = true;
            out.res[0] = static_cast<u16>(f);
            out.res[1] = static_cast<u16>(perm);
            out.res[2] = nums[0];
            out.res[3] = nums[1];
            out.res[4] = nums[2];
            return out;
          }
        }
      }
    }
    next_combination(nums, 3, rq.n);
  }
  return out;
}

// --- Naive oracles (fresh implementations of the reference semantics) ---

bool naive_check_n_lut_possible(int num, const ttable& target, const ttable& mask,
                                const ttable* tables) {
  // A k-LUT exists iff no input-pattern cell mixes masked target-1 and
  // target-0 positions.
  for (int c = 0; c < (1 << num); c++) {
    ttable cell = tt_ones_table();
    for (int j = 0; j < num; j++) {
      cell &= ((c >> (num - 1 - j)) & 1) ? tables[j] : ~tables[j];
    }
    bool has1 = tt_any(cell & target & mask);
    bool has0 = tt_any(cell & ~target & mask);
    if (has1 && has0) return false;
  }
  return true;
}

bool naive_get_lut_function(const ttable& a, const ttable& b, const ttable& c,
                            const ttable& target, const ttable& mask, u8* func) {
  // Per-position constraint propagation over all 256 positions.
  u8 f = 0, set = 0;
  for (int pos = 0; pos < 256; pos++) {
    if (!tt_get_bit(mask, pos)) continue;
    int p = (tt_get_bit(a, pos) << 2) | (tt_get_bit(b, pos) << 1) | tt_get_bit(c, pos);
    u8 want = static_cast<u8>(tt_get_bit(target, pos));
    if (set & (1u << p)) {
      if (((f >> p) & 1) != want) return false;
    } else {
      set |= 1u << p;
      f |= static_cast<u8>(want << p);
    }
  }
  *func = f;
  return true;
}

}  // namespace sbg
