// scan.hpp — candidate-scan primitives over a gate pool.
//
// Each scan walks a contiguous lexicographic range of k-combinations of the
// pool's gate truth tables and reports the first combination that admits a
// LUT (or LUT composition) realizing `target` under `mask`. The CPU
// implementations here are both the no-GPU execution path and the oracle
// for the CDNA4 kernels (gpu.hpp) in tests. Result layout matches the
// reference's res[10] wire format (lut.h:40-55):
//   3-LUT: res[0]=func,                res[1..3]=gate ids
//   5-LUT: res[0]=fo, res[1]=fi,       res[2..6]=gate ids (outer a,b,c; d,e)
//   7-LUT: res[0]=fo, res[1]=fm, res[2]=fi, res[3..9]=gate ids
#pragma once

#include <vector>

#include "sbg/common.hpp"
#include "sbg/state.hpp"
#include "sbg/ttable.hpp"

namespace sbg {

struct ScanResult {
  bool found = false;
  u16 res[10] = {};
  u64 evaluated = 0;  // combinations examined (for candidates/sec reporting)
};

// Precomputed matcher for the gate-mode step-4 triple scan: for every
// (care, req1) cell-requirement pair, whether some available composed
// 3-input function satisfies it. bitmap bit index = care*256 + req1.
// Built once per option set from avail_3 (8.2 KB; O(256^2 * |avail|) host
// preprocessing turns the reference's per-candidate 256-function x 4-order
// truth-table loop, sboxgates.c:406-432, into one bit probe per argument
// order).
struct Avail3Matcher {
  u8 bitmap[256 * 256 / 8];
  u8 funs[256];     // available function bytes, search order
  u8 cost[256];     // gates added when realized (1 + #input NOTs + not_out)
  int count;
};

// Packed scan request shared by CPU and GPU paths.
struct ScanRequest {
  const ttable* tables;  // gate pool truth tables, ids 0..n-1
  int n;                 // pool size (state.num_gates)
  ttable target;
  ttable mask;
  u64 excl_low64;        // bitmask of excluded gate ids < 64 (the inbits)
  u64 seed;              // randomization source
  bool count_all;        // true: never early-exit (bench mode)
  const Avail3Matcher* matcher = nullptr;  // k=4 scans only
};

// 3-LUT: scan combinations [begin, end) of C(n,3).
ScanResult cpu_scan3(const ScanRequest& rq, i64 begin, i64 end);

// 5-LUT: scan combinations [begin, end) of C(n,5).
ScanResult cpu_scan5(const ScanRequest& rq, i64 begin, i64 end);

// 7-LUT: scan combinations [begin, end) of C(n,7); feasibility filter and
// (3,3,1) function assignment are fused per-range (no global frontier cap,
// unlike the reference's 100k-per-rank truncation, lut.c:291-318).
ScanResult cpu_scan7(const ScanRequest& rq, i64 begin, i64 end);

// Gate-mode step-4 triple scan (k=4): find a triple realized by an
// available composed 3-input function in one of the 6 argument orders.
// Result: res[0]=avail index, res[1]=argument order (TRIPLE_PERMS index),
// res[2..4]=gate ids. Requires rq.matcher.
ScanResult cpu_scan4(const ScanRequest& rq, i64 begin, i64 end);

// Builds the matcher from an avail_3-style list (funs/cost arrays).
void build_avail3_matcher(const u8* funs, const u8* costs, int count,
                          Avail3Matcher* out);

// The 6 argument orders for a triple; shared by host and device code.
// TRIPLE_PERMS[p] selects which canonical input feeds A, B, C.
extern const int TRIPLE_PERMS6[6][3];

// Permute an 8-bit cell mask: out bit (v[s0]<<2|v[s1]<<1|v[s2]) = in bit
// (v0<<2|v1<<1|v2).
u8 permute_cells8_host(u8 m, const int* sel);

// --- Naive reference-style checkers (test oracles; see lut.c:34-109) ---

// True if some k-input LUT over `tables` can realize target under mask.
bool naive_check_n_lut_possible(int num, const ttable& target, const ttable& mask,
                                const ttable* tables);

// Derives a 3-LUT function by per-position constraint propagation.
bool naive_get_lut_function(const ttable& a, const ttable& b, const ttable& c,
                            const ttable& target, const ttable& mask, u8* func);

}  // namespace sbg
