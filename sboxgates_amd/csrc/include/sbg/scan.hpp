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

// Packed scan request shared by CPU and GPU paths.
struct ScanRequest {
  const ttable* tables;  // gate pool truth tables, ids 0..n-1
  int n;                 // pool size (state.num_gates)
  ttable target;
  ttable mask;
  u64 excl_low64;        // bitmask of excluded gate ids < 64 (the inbits)
  u64 seed;              // randomization source
  bool count_all;        // true: never early-exit (bench mode)
};

// 3-LUT: scan combinations [begin, end) of C(n,3).
ScanResult cpu_scan3(const ScanRequest& rq, i64 begin, i64 end);

// 5-LUT: scan combinations [begin, end) of C(n,5).
ScanResult cpu_scan5(const ScanRequest& rq, i64 begin, i64 end);

// 7-LUT: scan combinations [begin, end) of C(n,7); feasibility filter and
// (3,3,1) function assignment are fused per-range (no global frontier cap,
// unlike the reference's 100k-per-rank truncation, lut.c:291-318).
ScanResult cpu_scan7(const ScanRequest& rq, i64 begin, i64 end);

// --- Naive reference-style checkers (test oracles; see lut.c:34-109) ---

// True if some k-input LUT over `tables` can realize target under mask.
bool naive_check_n_lut_possible(int num, const ttable& target, const ttable& mask,
                                const ttable* tables);

// Derives a 3-LUT function by per-position constraint propagation.
bool naive_get_lut_function(const ttable& a, const ttable& b, const ttable& c,
                            const ttable& target, const ttable& mask, u8* func);

}  // namespace sbg
