// kernels.hip — CDNA4 (gfx950) kernels for the 3/5/7-LUT candidate scans,
// plus the GpuEngine host wrapper.
//
// Design (MI355X-first; see SURVEY.md §2.3 and the repo README):
//  * One thread evaluates one candidate combination on 256-bit truth
//    tables held as 4 u64 words. The reference runs these scans as CPU
//    loops across MPI ranks (lut.c:116-487); here the per-candidate
//    feasibility test is restructured as incremental cell algebra
//    (sbg/lutcover.hpp) with a per-u early exit.
//  * The live gate pool (<= 500 x 32 B = 16 KB) is staged in LDS once per
//    workgroup; the shared outer-prefix cells (8 for 5-LUT, 16 for 7-LUT)
//    are computed once per prefix and broadcast-read from LDS, so the
//    inner loop touches only the candidate's own 2-3 tables.
//  * Work distribution: workgroups pull combination *prefixes* (triples
//    for K5, quadruples for K7) from a device-scope atomic queue — the
//    per-prefix work varies by orders of magnitude, and the dequeue
//    primitive costs ~0.25-1 us (MI355X_MICROARch price list), negligible
//    against per-prefix work. Grids are sized >> 256 CUs.
//  * Early exit: winner election by atomicCAS on a device-scope lock; an
//    agent-scope abort flag (sc1, L1-bypassing load) is polled on a coarse
//    cadence. Any valid winner is acceptable (the reference is equally
//    nondeterministic across ranks, lut.c:213-218).
//  * No i64 division in device code (it is software-emulated): binomials
//    use closed forms with constant divisors (compiled to multiply-high).

#include <hip/hip_runtime.h>

#include <algorithm>
#include <chrono>
#include <cstddef>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <map>
#include <memory>
#include <mutex>
#include <stdexcept>
#include <string>
#include <thread>
#include <vector>

#include "sbg/comb.hpp"
#include "sbg/gpu.hpp"
#include "sbg/lutcover.hpp"
#include "sbg/rng.hpp"

namespace sbg {

#define SBG_HIP_CHECK(expr)                                                    \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) {                                                    \
      throw std::runtime_error(std::string("HIP error: ") +                    \
                               hipGetErrorString(_e) + " at " #expr);          \
    }                                                                          \
  } while (0)

namespace {

// ---------------------------------------------------------------------------
// Device-side control block.
// ---------------------------------------------------------------------------
struct DevCtl {
  u32 abort;
  u32 lock;
  u32 found;
  u32 pad;
  u16 res[10];
  u16 pad2[3];
  unsigned long long evaluated;
  unsigned long long queue;      // prefix dequeue counter
  unsigned long long hit_count;  // K7 filter
  u32 overflow;                  // K7 filter: hit buffer overflowed
  u32 pad3;
};

struct Hit7 {
  u64 p1[2];
  u64 p0[2];
  u16 nums[7];
  u16 pad;
};

// Closed-form binomials; constant divisors compile to multiply-high.
__device__ __forceinline__ i64 cf2(i64 x) { return x < 2 ? 0 : x * (x - 1) / 2; }
__device__ __forceinline__ i64 cf3(i64 x) {
  return x < 3 ? 0 : x * (x - 1) * (x - 2) / 6;
}
__device__ __forceinline__ i64 cf4(i64 x) {
  return x < 4 ? 0 : x * (x - 1) * (x - 2) * (x - 3) / 24;
}
__device__ __forceinline__ i64 cf5(i64 x) {
  return x < 5 ? 0 : x * (x - 1) * (x - 2) * (x - 3) * (x - 4) / 120;
}
__device__ __forceinline__ i64 cf6(i64 x) {
  return x < 6 ? 0 : x * (x - 1) * (x - 2) * (x - 3) * (x - 4) * (x - 5) / 720;
}
__device__ __forceinline__ i64 cf7(i64 x) {
  if (x < 7) return 0;
  u64 p = static_cast<u64>(x) * (x - 1) * (x - 2) * (x - 3);
  p *= static_cast<u64>(x - 4) * (x - 5) / 2;  // keep the product in range
  p *= static_cast<u64>(x - 6);
  return static_cast<i64>(p / 2520);
}

// Largest a in [0, n-k] with C(n,k) - C(n-a,k) <= r, i.e. the first element
// of the rank-r k-combination of [0,n). Binary search over a monotone
// closed form (~log2(n) probes).
template <i64 (*CF)(i64)>
__device__ __forceinline__ int first_of_rank(i64 r, int n, i64 total) {
  int lo = 0, hi = n;  // invariant: prefix(lo) <= r < prefix(hi)
  while (hi - lo > 1) {
    int mid = (lo + hi) >> 1;
    i64 prefix = total - CF(n - mid);
    if (prefix <= r) lo = mid; else hi = mid;
  }
  return lo;
}

// Flat pair index -> (d, e), 0 <= d < e < m (lexicographic d-major).
__device__ __forceinline__ void dev_decode_pair(i64 q, int m, int* d, int* e) {
  double mm = static_cast<double>(m) - 0.5;
  double disc = mm * mm - 2.0 * static_cast<double>(q) - 0.75;
  int dd = static_cast<int>(mm - 0.5 - __builtin_sqrt(disc > 0.0 ? disc : 0.0));
  if (dd < 0) dd = 0;
  if (dd > m - 2) dd = m - 2;
  // S(d) = d*m - d*(d+1)/2; fix up to the exact row.
  i64 S = static_cast<i64>(dd) * m - static_cast<i64>(dd) * (dd + 1) / 2;
  while (dd > 0 && S > q) { dd--; S -= m - 1 - dd; }
  while (S + (m - 1 - dd) <= q) { S += m - 1 - dd; dd++; }
  *d = dd;
  *e = static_cast<int>(q - S) + dd + 1;
}

// Flat triple index -> (d, e, f), 0 <= d < e < f < m.
__device__ __forceinline__ void dev_decode_triple(i64 q, int m, int* d, int* e,
                                                  int* f) {
  int dd = first_of_rank<cf3>(q, m, cf3(m));
  i64 rem = q - (cf3(m) - cf3(m - dd));
  int e2, f2;
  dev_decode_pair(rem, m - dd - 1, &e2, &f2);
  *d = dd;
  *e = dd + 1 + e2;
  *f = dd + 1 + f2;
}

__device__ __forceinline__ u64 dev_rnd(u64 seed, u64 idx) {
  return hash_mix64(seed ^ (idx * 0x9E3779B97F4A7C15ULL));
}

// Global-address-space pointer casts for cross-workgroup / host-visible
// words. Per the CDNA4 inter-workgroup rules (cdna_hip_programming.md
// Guideline 16): every shared word must be a GLOBAL agent/system-scope
// access — flat atomics miss the cache-bypass lowering and an XCD's
// private L2 then serves stale values to its readers forever.
typedef __attribute__((address_space(1))) unsigned svc_gu32;
typedef __attribute__((address_space(1))) unsigned long long svc_gu64;
__device__ __forceinline__ svc_gu32* as_gu32(u32* p) { return (svc_gu32*)p; }
__device__ __forceinline__ svc_gu64* as_gu64(u64* p) { return (svc_gu64*)p; }
__device__ __forceinline__ svc_gu64* as_gu64(unsigned long long* p) {
  return (svc_gu64*)p;
}

__device__ __forceinline__ bool dev_abort(const DevCtl* ctl) {
  return __hip_atomic_load(as_gu32(const_cast<u32*>(&ctl->abort)),
                           __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT) != 0;
}

// Publish a winner: single writer wins the lock; the rest skip. Payload
// (res) is plain-stored, then drained and released before the found/abort
// flag stores (Guideline 16 R1: drain -> release fence -> asm wait ->
// relaxed agent flag store), so an agent-scope reader that acquires after
// seeing found != 0 reads a complete payload on any XCD. Host readers via
// stream-sync + memcpy are ordered regardless.
__device__ __forceinline__ void dev_publish(DevCtl* ctl, const u16 res[10]) {
  if (atomicCAS(&ctl->lock, 0u, 1u) == 0u) {
    for (int i = 0; i < 10; i++) ctl->res[i] = res[i];
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __hip_atomic_store(as_gu32(&ctl->found), 1u, __ATOMIC_RELAXED,
                       __HIP_MEMORY_SCOPE_AGENT);
    __hip_atomic_store(as_gu32(&ctl->abort), 1u, __ATOMIC_RELAXED,
                       __HIP_MEMORY_SCOPE_AGENT);
  }
}

struct ScanArgs {
  const ttable* pool;  // device pool (n tables)
  DevCtl* ctl;
  Hit7* hits;          // K7 only
  u64 hit_cap;         // K7 only
  const Avail3Matcher* matcher;  // k=4 only
  ttable T1, T0;
  int n;
  i64 begin, end;
  u64 excl;  // excluded gate ids < 64
  u64 seed;
  int count_all;
  int slices7;  // K7: sub-quad slicing factor (dequeue unit = quad x slice)
};

constexpr int SCAN_BLOCK = 256;

// LDS pool stride: 6 u64 (48 B) per gate instead of 4 (32 B). A 32 B
// stride puts ds_read_b128 lane groups on 8-way-conflicting banks when
// lanes read consecutive gate ids; 48 B (12 dwords, 16 B aligned) spreads
// a 16-lane group over 16 distinct banks (measured: SQ_LDS_BANK_CONFLICT
// was 21% of LDS cycles at stride 32).
constexpr int PSTR = 6;

__device__ __forceinline__ void load_pool_lds(u64* s_pool, const ttable* pool,
                                              int n) {
  const u64* src = reinterpret_cast<const u64*>(pool);
  for (int i = threadIdx.x; i < n * 4; i += blockDim.x) {
    s_pool[(i >> 2) * PSTR + (i & 3)] = src[i];
  }
}

// ---------------------------------------------------------------------------
// K2 — 3-LUT scan. Grid-stride over combination ranks; per-thread decode by
// binary search on closed-form binomials; cells evaluated directly (the
// reference's serial rank-0 loop, lut.c:501-523).
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(SCAN_BLOCK) k_scan3(ScanArgs args) {
  __shared__ alignas(16) u64 s_pool[MAX_GATES * PSTR];
  const int n = args.n;
  load_pool_lds(s_pool, args.pool, n);
  __syncthreads();

  DevCtl* ctl = args.ctl;
  const i64 total3 = cf3(n);
  u64 local_eval = 0;
  const i64 stride = static_cast<i64>(gridDim.x) * blockDim.x;
  i64 idx = args.begin + blockIdx.x * static_cast<i64>(blockDim.x) + threadIdx.x;
  int tick = 0;

  for (; idx < args.end; idx += stride) {
    if (((tick++) & 255) == 0 && !args.count_all && dev_abort(ctl)) break;
    // Decode rank -> (a, b, c).
    int a = first_of_rank<cf3>(idx, n, total3);
    i64 rem = idx - (total3 - cf3(n - a));
    int b2, c2;
    dev_decode_pair(rem, n - a - 1, &b2, &c2);
    int b = a + 1 + b2, c = a + 1 + c2;

    local_eval++;
    // Copy out of LDS: the padded 48 B stride is not ttable-aligned, so a
    // reinterpret_cast would be UB.
    ttable ta, tb, tc;
#pragma unroll
    for (int w = 0; w < 4; w++) {
      ta.w[w] = s_pool[a * PSTR + w];
      tb.w[w] = s_pool[b * PSTR + w];
      tc.w[w] = s_pool[c * PSTR + w];
    }
    u32 p1, p0;
    if (lut3_p_masks(ta, tb, tc, args.T1, args.T0, &p1, &p0)) {
      u8 func = lut3_function_from_p(p1, p0, dev_rnd(args.seed, idx));
      if (!args.count_all && func != 0) {
        u16 res[10] = {};
        res[0] = func;
        res[1] = static_cast<u16>(a);
        res[2] = static_cast<u16>(b);
        res[3] = static_cast<u16>(c);
        dev_publish(ctl, res);
        break;
      }
    }
  }
  // Per-block evaluated reduction.
  __shared__ unsigned long long s_eval;
  if (threadIdx.x == 0) s_eval = 0;
  __syncthreads();
  atomicAdd(&s_eval, static_cast<unsigned long long>(local_eval));
  __syncthreads();
  if (threadIdx.x == 0) atomicAdd(&ctl->evaluated, s_eval);
}

// ---------------------------------------------------------------------------
// K1c — gate-mode step-4 triple scan: triples realized by an available
// composed 3-input function in one of 6 argument orders. The per-candidate
// function search is one bitmap probe per order (matcher precomputed on the
// host), replacing the reference's 256-function x 4-order truth-table loop
// (sboxgates.c:392-435).
// ---------------------------------------------------------------------------
__device__ __forceinline__ u8 dev_permute_cells8(u8 m, int s0, int s1, int s2) {
  u8 out = 0;
#pragma unroll
  for (int c = 0; c < 8; c++) {
    if (!((m >> c) & 1)) continue;
    int v[3] = {(c >> 2) & 1, (c >> 1) & 1, c & 1};
    out |= static_cast<u8>(1u << ((v[s0] << 2) | (v[s1] << 1) | v[s2]));
  }
  return out;
}

__constant__ int c_perm6[6][3] = {{0, 1, 2}, {0, 2, 1}, {1, 0, 2},
                                  {1, 2, 0}, {2, 0, 1}, {2, 1, 0}};

// Shared k=4 scan loop: grid-strided walk of triple ranks [idx0, end) with
// stride `stride`, pool/matcher staged in LDS by the caller. Returns this
// thread's evaluated count. Used by both the one-shot k_scan4 kernel and
// the persistent scan service below.
__device__ __forceinline__ u64 scan4_loop(const u64* s_pool, const u8* s_bitmap,
                                          const u8* s_funs, int s_count, int n,
                                          const ttable& T1, const ttable& T0,
                                          int count_all, DevCtl* ctl, i64 idx0,
                                          i64 end, i64 stride) {
  const i64 total3 = cf3(n);
  u64 local_eval = 0;
  int tick = 0;
  for (i64 idx = idx0; idx < end; idx += stride) {
    if (((tick++) & 255) == 0 && !count_all && dev_abort(ctl)) break;
    int a = first_of_rank<cf3>(idx, n, total3);
    i64 rem = idx - (total3 - cf3(n - a));
    int b2, c2;
    dev_decode_pair(rem, n - a - 1, &b2, &c2);
    int b = a + 1 + b2, c = a + 1 + c2;

    local_eval++;
    ttable ta, tb, tc;
#pragma unroll
    for (int w = 0; w < 4; w++) {
      ta.w[w] = s_pool[a * PSTR + w];
      tb.w[w] = s_pool[b * PSTR + w];
      tc.w[w] = s_pool[c * PSTR + w];
    }
    u32 p1, p0;
    if (!lut3_p_masks(ta, tb, tc, T1, T0, &p1, &p0)) continue;
    if (count_all) continue;
    const u8 req1 = static_cast<u8>(p1);
    const u8 care = static_cast<u8>(p1 | p0);
    for (int perm = 0; perm < 6; perm++) {
      const u8 r = dev_permute_cells8(req1, c_perm6[perm][0], c_perm6[perm][1],
                                      c_perm6[perm][2]);
      const u8 cr = dev_permute_cells8(care, c_perm6[perm][0], c_perm6[perm][1],
                                       c_perm6[perm][2]);
      const int bidx = cr * 256 + r;
      if (!((s_bitmap[bidx >> 3] >> (bidx & 7)) & 1)) continue;
      for (int f = 0; f < s_count; f++) {
        if ((s_funs[f] & cr) == r) {
          u16 res[10] = {};
          res[0] = static_cast<u16>(f);
          res[1] = static_cast<u16>(perm);
          res[2] = static_cast<u16>(a);
          res[3] = static_cast<u16>(b);
          res[4] = static_cast<u16>(c);
          dev_publish(ctl, res);
          break;
        }
      }
      break;  // bitmap said a function exists; it was found and published
    }
    if (dev_abort(ctl) && !count_all) break;
  }
  return local_eval;
}

__global__ void __launch_bounds__(SCAN_BLOCK) k_scan4(ScanArgs args) {
  __shared__ alignas(16) u64 s_pool[MAX_GATES * PSTR];
  __shared__ alignas(16) u8 s_bitmap[256 * 256 / 8];
  __shared__ u8 s_funs[256];
  __shared__ int s_count;
  const int n = args.n;
  load_pool_lds(s_pool, args.pool, n);
  for (int i = threadIdx.x; i < 256 * 256 / 8 / 8; i += blockDim.x) {
    reinterpret_cast<u64*>(s_bitmap)[i] =
        reinterpret_cast<const u64*>(args.matcher->bitmap)[i];
  }
  for (int i = threadIdx.x; i < 256 / 8; i += blockDim.x) {
    reinterpret_cast<u64*>(s_funs)[i] =
        reinterpret_cast<const u64*>(args.matcher->funs)[i];
  }
  if (threadIdx.x == 0) s_count = args.matcher->count;
  __syncthreads();

  DevCtl* ctl = args.ctl;
  const i64 stride = static_cast<i64>(gridDim.x) * blockDim.x;
  const i64 idx0 =
      args.begin + blockIdx.x * static_cast<i64>(blockDim.x) + threadIdx.x;
  u64 local_eval = scan4_loop(s_pool, s_bitmap, s_funs, s_count, n, args.T1,
                              args.T0, args.count_all, ctl, idx0, args.end,
                              stride);
  __shared__ unsigned long long s_eval;
  if (threadIdx.x == 0) s_eval = 0;
  __syncthreads();
  atomicAdd(&s_eval, static_cast<unsigned long long>(local_eval));
  __syncthreads();
  if (threadIdx.x == 0) atomicAdd(&ctl->evaluated, s_eval);
}

// ---------------------------------------------------------------------------
// K3 — 5-LUT scan. Workgroups dequeue triple prefixes; the 8 prefix cells
// (masked with target-1/target-0) are computed cooperatively into LDS; each
// thread then screens (d, e) pairs with a per-u early exit and runs the
// 2-coloring decomposition solver on survivors (replacing the reference's
// 10 x 256-function brute force, lut.c:174-246).
// ---------------------------------------------------------------------------

// Survivor handling for k_scan5 (rare path; noinline keeps its registers
// out of the hot loop's budget).
__device__ __attribute__((noinline)) void scan5_handle_survivor(
    const ScanArgs& args, DevCtl* ctl, const u64* s_pool, const u16* abc,
    i64 rank, int d, int e) {
  ttable tt5[5];
  const int ids[5] = {abc[0], abc[1], abc[2], d, e};
  for (int j = 0; j < 5; j++) {
    for (int w = 0; w < 4; w++) tt5[j].w[w] = s_pool[ids[j] * PSTR + w];
  }
  u32 p1, p0;
  if (!lut5_p_masks(tt5, args.T1, args.T0, &p1, &p0)) return;
  u8 fo, fi;
  int split;
  if (!lut5_solve_from_p(p1, p0, dev_rnd(args.seed, rank), &fo, &fi, &split)) {
    return;
  }
  if (args.count_all) return;
  const u16 nums[5] = {abc[0], abc[1], abc[2], static_cast<u16>(d),
                       static_cast<u16>(e)};
  const u8* sp = SPLITS5[split];
  u16 res[10] = {};
  res[0] = fo;
  res[1] = fi;
  for (int j = 0; j < 3; j++) res[2 + j] = nums[sp[j]];
  res[5] = nums[sp[3]];
  res[6] = nums[sp[4]];
  dev_publish(ctl, res);
}

__global__ void __launch_bounds__(SCAN_BLOCK, 4) k_scan5(ScanArgs args) {
  // Triple batching: one dequeue + one cooperative prefix-cell build + one
  // barrier per TB triples (measured at TB=1: 48% of wave time parked on
  // the per-triple barriers; batching amortizes them 8x).
  constexpr int TB = 8;
  __shared__ alignas(16) u64 s_pool[MAX_GATES * PSTR];
  __shared__ alignas(16) u64 s_H1[TB][8][4];
  __shared__ alignas(16) u64 s_H0[TB][8][4];
  __shared__ i64 s_base[TB];     // flat-rank base of each triple
  __shared__ i64 s_lo[TB], s_prefix[TB + 1];
  __shared__ int s_m[TB], s_c[TB];
  __shared__ u16 s_abc[TB][3];
  __shared__ u32 s_live[TB];     // cells with both forced-1 and forced-0
                                 // content; only these can contradict
  __shared__ int s_stop;         // all triples past range end
  __shared__ unsigned long long s_eval;

  const int n = args.n;
  load_pool_lds(s_pool, args.pool, n);
  if (threadIdx.x == 0) s_eval = 0;

  DevCtl* ctl = args.ctl;
  const i64 total3 = cf3(n);
  u64 local_eval = 0;

  for (;;) {
    __syncthreads();
    if (threadIdx.x == 0) {
      s_stop = (!args.count_all && dev_abort(ctl)) ? 1 : 0;
      if (!s_stop) {
        unsigned long long t0 = __hip_atomic_fetch_add(
            &ctl->queue, static_cast<unsigned long long>(TB), __ATOMIC_RELAXED,
            __HIP_MEMORY_SCOPE_AGENT);
        s_base[0] = static_cast<i64>(t0);  // reuse slot to pass the batch base
      }
    }
    __syncthreads();
    if (s_stop) break;
    const i64 batch_base = s_base[0];
    __syncthreads();

    // Threads 0..TB-1 decode one triple each and compute its pair window.
    if (threadIdx.x < TB) {
      const int t = threadIdx.x;
      const i64 tidx = batch_base + t;
      i64 base = -1, lo = 0, hi = 0;
      int m = 0, a = 0, b = 0, c = 0;
      if (tidx < total3) {
        a = first_of_rank<cf3>(tidx, n, total3);
        i64 rem = tidx - (total3 - cf3(n - a));
        int b2, c2;
        dev_decode_pair(rem, n - a - 1, &b2, &c2);
        b = a + 1 + b2;
        c = a + 1 + c2;
        base = (cf5(n) - cf5(n - a)) + (cf4(n - a - 1) - cf4(n - b)) +
               (cf3(n - b - 1) - cf3(n - c));
        m = n - 1 - c;
        i64 npairs = cf2(m);
        lo = args.begin > base ? args.begin - base : 0;
        hi = args.end - base < npairs ? args.end - base : npairs;
        bool excl_prefix =
            args.excl != 0 &&
            (((a < 64) && ((args.excl >> a) & 1)) ||
             ((b < 64) && ((args.excl >> b) & 1)) ||
             ((c < 64) && ((args.excl >> c) & 1)));
        if (base >= args.end || lo >= hi || excl_prefix) {
          lo = hi = 0;
        }
      }
      s_base[t] = base;
      s_lo[t] = lo;
      s_m[t] = m;
      s_c[t] = c;
      s_abc[t][0] = static_cast<u16>(a);
      s_abc[t][1] = static_cast<u16>(b);
      s_abc[t][2] = static_cast<u16>(c);
      s_prefix[t + 1] = hi - lo;  // count, turned into prefix sums below
      if (t == 0) s_prefix[0] = 0;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      // Prefix-sum the TB pair counts. Termination: triple flat-rank bases
      // increase with the triple index, so once the batch's first triple
      // is past the range end (or past the triple space), every later
      // batch is too.
      s_stop = (batch_base >= total3 || s_base[0] >= args.end) ? 1 : 0;
      for (int t = 0; t < TB; t++) s_prefix[t + 1] += s_prefix[t];
    }
    __syncthreads();
    if (s_stop) break;
    const i64 total_pairs = s_prefix[TB];
    if (total_pairs == 0) continue;

    // Cooperative prefix-cell build: TB x 8 cells x 4 words x {1,0} = 512
    // items, two per thread.
    {
      const int half = TB * 8 * 4;  // 256 at TB=8
      for (int item = threadIdx.x; item < 2 * half; item += blockDim.x) {
        const bool is1 = item < half;
        const int it = is1 ? item : item - half;
        const int t = it >> 5;          // /32
        const int u = (it >> 2) & 7;
        const int w = it & 3;
        if (s_base[t] < 0 || s_prefix[t + 1] == s_prefix[t]) continue;
        const int a = s_abc[t][0], b = s_abc[t][1], c = s_abc[t][2];
        u64 ca = (u & 4) ? s_pool[a * PSTR + w] : ~s_pool[a * PSTR + w];
        u64 cb = (u & 2) ? s_pool[b * PSTR + w] : ~s_pool[b * PSTR + w];
        u64 cc = (u & 1) ? s_pool[c * PSTR + w] : ~s_pool[c * PSTR + w];
        u64 cell = ca & cb & cc;
        if (is1) {
          s_H1[t][u][w] = cell & args.T1.w[w];
        } else {
          s_H0[t][u][w] = cell & args.T0.w[w];
        }
      }
    }
    if (threadIdx.x < TB) s_live[threadIdx.x] = 0;
    __syncthreads();
    // Liveness: a cell whose H1 or H0 side is empty can never contradict
    // (and survivors rebuild exact p-masks anyway). Sparse-mask scans in
    // deep recursion typically keep only a few of the 8 cells live.
    if (threadIdx.x < TB * 8) {
      const int t = threadIdx.x >> 3;
      const int u = threadIdx.x & 7;
      const u64 h1 = s_H1[t][u][0] | s_H1[t][u][1] | s_H1[t][u][2] | s_H1[t][u][3];
      const u64 h0 = s_H0[t][u][0] | s_H0[t][u][1] | s_H0[t][u][2] | s_H0[t][u][3];
      if (h1 != 0 && h0 != 0) atomicOr(&s_live[t], 1u << u);
    }
    __syncthreads();

    // Per-thread runs of K consecutive pairs: one decode per run, cheap
    // lexicographic increments inside (a decode costs ~40 ops, the
    // increment ~4). Runs rarely cross triple boundaries (re-locate
    // handles it).
    constexpr int K = 4;
    const i64 nruns = (total_pairs + K - 1) / K;
    int it = 0;
    for (i64 run = threadIdx.x; run < nruns; run += blockDim.x) {
      if (((it++) & 15) == 0 && !args.count_all && dev_abort(ctl)) break;
      i64 idx = run * K;
      const i64 idx_end = idx + K < total_pairs ? idx + K : total_pairs;
      int t = 0;
      while (s_prefix[t + 1] <= idx) t++;
      int m = s_m[t];
      int cgate = s_c[t];
      i64 q = s_lo[t] + (idx - s_prefix[t]);
      int d2, e2;
      dev_decode_pair(q, m, &d2, &e2);

      for (; idx < idx_end; idx++) {
        if (idx >= s_prefix[t + 1]) {
          // Crossed into the next non-empty triple.
          do { t++; } while (s_prefix[t + 1] <= idx);
          m = s_m[t];
          cgate = s_c[t];
          q = s_lo[t] + (idx - s_prefix[t]);
          dev_decode_pair(q, m, &d2, &e2);
        }
        const int d = cgate + 1 + d2, e = cgate + 1 + e2;
        bool skip = false;
        if (args.excl != 0) {
          skip = (d < 64 && ((args.excl >> d) & 1)) ||
                 (e < 64 && ((args.excl >> e) & 1));
        }
        if (!skip) {
          local_eval++;
          const u64* td = &s_pool[d * PSTR];
          const u64* te = &s_pool[e * PSTR];
          u64 td_[4], te_[4];
#pragma unroll
          for (int w = 0; w < 4; w++) {
            td_[w] = td[w];
            te_[w] = te[w];
          }

          bool ok = true;
          for (u32 lm = s_live[t]; lm != 0; lm &= lm - 1) {
            const int u = __ffs(lm) - 1;
            u64 r11_1 = 0, r10_1 = 0, r01_1 = 0, r00_1 = 0;
            u64 r11_0 = 0, r10_0 = 0, r01_0 = 0, r00_0 = 0;
#pragma unroll
            for (int w = 0; w < 4; w++) {
              const u64 h1 = s_H1[t][u][w];
              const u64 h0 = s_H0[t][u][w];
              // x & ~y == x ^ (x & y): avoids materializing ~td/~te.
              const u64 a1 = h1 & td_[w];
              const u64 na1 = h1 ^ a1;
              const u64 a0 = h0 & td_[w];
              const u64 na0 = h0 ^ a0;
              const u64 x11_1 = a1 & te_[w];
              const u64 x01_1 = na1 & te_[w];
              const u64 x11_0 = a0 & te_[w];
              const u64 x01_0 = na0 & te_[w];
              r11_1 |= x11_1;
              r10_1 |= a1 ^ x11_1;
              r01_1 |= x01_1;
              r00_1 |= na1 ^ x01_1;
              r11_0 |= x11_0;
              r10_0 |= a0 ^ x11_0;
              r01_0 |= x01_0;
              r00_0 |= na0 ^ x01_0;
            }
            if ((r11_1 && r11_0) || (r10_1 && r10_0) || (r01_1 && r01_0) ||
                (r00_1 && r00_0)) {
              ok = false;
              break;
            }
          }
          if (ok) {
            scan5_handle_survivor(args, ctl, s_pool, s_abc[t], s_base[t] + q,
                                  d, e);
          }
        }
        // Next pair in lexicographic order.
        q++;
        e2++;
        if (e2 >= m) {
          d2++;
          e2 = d2 + 1;
        }
      }
    }
  }

  atomicAdd(&s_eval, static_cast<unsigned long long>(local_eval));
  __syncthreads();
  if (threadIdx.x == 0) atomicAdd(&ctl->evaluated, s_eval);
}

// ---------------------------------------------------------------------------
// K4a — 7-LUT feasibility filter. Workgroups dequeue quadruple prefixes;
// 16 prefix cells in LDS; threads screen (e, f, g) triples; survivors'
// p-masks land in the hit buffer (no 100k cap — 288 GB HBM keeps the whole
// frontier resident; overflow of the per-chunk buffer is reported and the
// host re-scans a smaller range).
// ---------------------------------------------------------------------------
// Survivor handling for k_scan7_filter: exact p-mask rebuild + hit append.
__device__ __attribute__((noinline)) void scan7_handle_survivor(
    const ScanArgs& args, DevCtl* ctl, const u64* s_pool, const u16* abcd,
    int e, int f, int g) {
  ttable tt7[7];
  const int ids[7] = {abcd[0], abcd[1], abcd[2], abcd[3], e, f, g};
  for (int j = 0; j < 7; j++) {
    for (int w = 0; w < 4; w++) tt7[j].w[w] = s_pool[ids[j] * PSTR + w];
  }
  u64 p1[2], p0[2];
  if (!lut7_p_masks(tt7, args.T1, args.T0, p1, p0)) return;
  if (args.count_all) return;  // bench mode: no hit materialization
  unsigned long long slot = __hip_atomic_fetch_add(&ctl->hit_count, 1ULL,
                                                   __ATOMIC_RELAXED,
                                                   __HIP_MEMORY_SCOPE_AGENT);
  if (slot >= args.hit_cap) {
    __hip_atomic_store(&ctl->overflow, 1u, __ATOMIC_RELAXED,
                       __HIP_MEMORY_SCOPE_AGENT);
    return;
  }
  Hit7& h = args.hits[slot];
  h.p1[0] = p1[0];
  h.p1[1] = p1[1];
  h.p0[0] = p0[0];
  h.p0[1] = p0[1];
  for (int j = 0; j < 7; j++) h.nums[j] = static_cast<u16>(ids[j]);
}

__global__ void __launch_bounds__(SCAN_BLOCK, 4) k_scan7_filter(ScanArgs args) {
  // Quad batching: one dequeue + one cooperative 16-cell build per QB
  // quadruples (same barrier-amortization pattern as k_scan5).
  constexpr int QB = 4;
  __shared__ alignas(16) u64 s_pool[MAX_GATES * PSTR];
  __shared__ alignas(16) u64 s_H1[QB][16][4];
  __shared__ alignas(16) u64 s_H0[QB][16][4];
  __shared__ i64 s_lo[QB], s_prefix[QB + 1];
  __shared__ int s_m[QB], s_d[QB];
  __shared__ u16 s_abcd[QB][4];
  __shared__ u32 s_live[QB];
  __shared__ i64 s_batch[2];  // [0]=batch base quad index, [1]=first base
  __shared__ int s_stop;
  __shared__ unsigned long long s_eval;

  const int n = args.n;
  load_pool_lds(s_pool, args.pool, n);
  if (threadIdx.x == 0) s_eval = 0;

  DevCtl* ctl = args.ctl;
  const i64 total4 = cf4(n);
  u64 local_eval = 0;

  for (;;) {
    __syncthreads();
    if (threadIdx.x == 0) {
      bool halt = (!args.count_all && dev_abort(ctl)) ||
                  __hip_atomic_load(&ctl->overflow, __ATOMIC_RELAXED,
                                    __HIP_MEMORY_SCOPE_AGENT) != 0;
      s_stop = halt ? 1 : 0;
      if (!halt) {
        s_batch[0] = static_cast<i64>(__hip_atomic_fetch_add(
            &ctl->queue, static_cast<unsigned long long>(QB), __ATOMIC_RELAXED,
            __HIP_MEMORY_SCOPE_AGENT));
      }
    }
    __syncthreads();
    if (s_stop) break;
    const i64 batch_base = s_batch[0];
    __syncthreads();

    if (threadIdx.x < QB) {
      const int t = threadIdx.x;
      const int S = args.slices7;
      const i64 unit = batch_base + t;
      const i64 qidx = unit / S;
      const int slice = static_cast<int>(unit % S);
      i64 base = -1, lo = 0, hi = 0;
      int m = 0, a = 0, b = 0, c = 0, d = 0;
      if (qidx < total4) {
        a = first_of_rank<cf4>(qidx, n, total4);
        i64 rem = qidx - (total4 - cf4(n - a));
        int m1 = n - a - 1;
        int b0 = first_of_rank<cf3>(rem, m1, cf3(m1));
        i64 rem2 = rem - (cf3(m1) - cf3(m1 - b0));
        int c2, d2;
        dev_decode_pair(rem2, m1 - b0 - 1, &c2, &d2);
        b = a + 1 + b0;
        c = b + 1 + c2;
        d = b + 1 + d2;
        base = (cf7(n) - cf7(n - a)) + (cf6(n - a - 1) - cf6(n - b)) +
               (cf5(n - b - 1) - cf5(n - c)) + (cf4(n - c - 1) - cf4(n - d));
        m = n - 1 - d;
        i64 ntrips = cf3(m);
        lo = args.begin > base ? args.begin - base : 0;
        hi = args.end - base < ntrips ? args.end - base : ntrips;
        if (S > 1 && hi > lo) {
          // This unit covers one slice of the quad's triple window.
          const i64 span = hi - lo;
          const i64 per = (span + S - 1) / S;
          const i64 s_lo2 = lo + per * slice;
          const i64 s_hi2 = s_lo2 + per < hi ? s_lo2 + per : hi;
          lo = s_lo2 < hi ? s_lo2 : hi;
          hi = s_hi2;
        }
        bool excl_prefix =
            args.excl != 0 &&
            (((a < 64) && ((args.excl >> a) & 1)) ||
             ((b < 64) && ((args.excl >> b) & 1)) ||
             ((c < 64) && ((args.excl >> c) & 1)) ||
             ((d < 64) && ((args.excl >> d) & 1)));
        if (base >= args.end || lo >= hi || excl_prefix) lo = hi = 0;
      }
      if (t == 0) {
        s_batch[1] = base;
        s_prefix[0] = 0;
      }
      s_lo[t] = lo;
      s_m[t] = m;
      s_d[t] = d;
      s_abcd[t][0] = static_cast<u16>(a);
      s_abcd[t][1] = static_cast<u16>(b);
      s_abcd[t][2] = static_cast<u16>(c);
      s_abcd[t][3] = static_cast<u16>(d);
      s_prefix[t + 1] = hi - lo;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      s_stop = (batch_base / args.slices7 >= total4 || s_batch[1] >= args.end)
                   ? 1
                   : 0;
      for (int t = 0; t < QB; t++) s_prefix[t + 1] += s_prefix[t];
    }
    __syncthreads();
    if (s_stop) break;
    const i64 total_trips = s_prefix[QB];
    if (total_trips == 0) continue;

    // Cooperative prefix-cell build: QB x 16 cells x 4 words x {1,0} = 512.
    {
      const int half = QB * 16 * 4;  // 256
      for (int item = threadIdx.x; item < 2 * half; item += blockDim.x) {
        const bool is1 = item < half;
        const int iv = is1 ? item : item - half;
        const int t = iv >> 6;
        const int u = (iv >> 2) & 15;
        const int w = iv & 3;
        if (s_prefix[t + 1] == s_prefix[t]) continue;
        const int a = s_abcd[t][0], b = s_abcd[t][1], c = s_abcd[t][2],
                  d = s_abcd[t][3];
        u64 ca = (u & 8) ? s_pool[a * PSTR + w] : ~s_pool[a * PSTR + w];
        u64 cb = (u & 4) ? s_pool[b * PSTR + w] : ~s_pool[b * PSTR + w];
        u64 cc = (u & 2) ? s_pool[c * PSTR + w] : ~s_pool[c * PSTR + w];
        u64 cd = (u & 1) ? s_pool[d * PSTR + w] : ~s_pool[d * PSTR + w];
        u64 cell = ca & cb & cc & cd;
        if (is1) {
          s_H1[t][u][w] = cell & args.T1.w[w];
        } else {
          s_H0[t][u][w] = cell & args.T0.w[w];
        }
      }
    }
    if (threadIdx.x < QB) s_live[threadIdx.x] = 0;
    __syncthreads();
    if (threadIdx.x < QB * 16) {
      const int t = threadIdx.x >> 4;
      const int u = threadIdx.x & 15;
      const u64 h1 = s_H1[t][u][0] | s_H1[t][u][1] | s_H1[t][u][2] | s_H1[t][u][3];
      const u64 h0 = s_H0[t][u][0] | s_H0[t][u][1] | s_H0[t][u][2] | s_H0[t][u][3];
      if (h1 != 0 && h0 != 0) atomicOr(&s_live[t], 1u << u);
    }
    __syncthreads();

    int it = 0;
    for (i64 idx = threadIdx.x; idx < total_trips; idx += blockDim.x) {
      if (((it++) & 15) == 0) {
        if ((!args.count_all && dev_abort(ctl)) ||
            __hip_atomic_load(&ctl->overflow, __ATOMIC_RELAXED,
                              __HIP_MEMORY_SCOPE_AGENT) != 0) {
          break;
        }
      }
      int t = 0;
      while (s_prefix[t + 1] <= idx) t++;
      const i64 q = s_lo[t] + (idx - s_prefix[t]);
      const int m = s_m[t];
      const int dgate = s_d[t];
      int e2, f2, g2;
      dev_decode_triple(q, m, &e2, &f2, &g2);
      int e = dgate + 1 + e2, f = dgate + 1 + f2, g = dgate + 1 + g2;
      if (args.excl != 0) {
        if ((e < 64 && ((args.excl >> e) & 1)) || (f < 64 && ((args.excl >> f) & 1)) ||
            (g < 64 && ((args.excl >> g) & 1))) {
          continue;
        }
      }
      local_eval++;

      u64 te_[4], tf_[4], tg_[4];
#pragma unroll
      for (int w = 0; w < 4; w++) {
        te_[w] = s_pool[e * PSTR + w];
        tf_[w] = s_pool[f * PSTR + w];
        tg_[w] = s_pool[g * PSTR + w];
      }

      bool ok = true;
      for (u32 lm = s_live[t]; lm != 0 && ok; lm &= lm - 1) {
        const int u = __ffs(lm) - 1;
        u64 acc1[8] = {0, 0, 0, 0, 0, 0, 0, 0};
        u64 acc0[8] = {0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
        for (int w = 0; w < 4; w++) {
          const u64 h1 = s_H1[t][u][w];
          const u64 h0 = s_H0[t][u][w];
          const u64 e1 = h1 & te_[w];
          const u64 e0n = h1 ^ e1;
          const u64 z1 = h0 & te_[w];
          const u64 z0n = h0 ^ z1;
          const u64 ef11 = e1 & tf_[w];
          const u64 ef10 = e1 ^ ef11;
          const u64 ef01 = e0n & tf_[w];
          const u64 ef00 = e0n ^ ef01;
          const u64 zf11 = z1 & tf_[w];
          const u64 zf10 = z1 ^ zf11;
          const u64 zf01 = z0n & tf_[w];
          const u64 zf00 = z0n ^ zf01;
          u64 x;
          x = ef11 & tg_[w]; acc1[7] |= x; acc1[6] |= ef11 ^ x;
          x = ef10 & tg_[w]; acc1[5] |= x; acc1[4] |= ef10 ^ x;
          x = ef01 & tg_[w]; acc1[3] |= x; acc1[2] |= ef01 ^ x;
          x = ef00 & tg_[w]; acc1[1] |= x; acc1[0] |= ef00 ^ x;
          x = zf11 & tg_[w]; acc0[7] |= x; acc0[6] |= zf11 ^ x;
          x = zf10 & tg_[w]; acc0[5] |= x; acc0[4] |= zf10 ^ x;
          x = zf01 & tg_[w]; acc0[3] |= x; acc0[2] |= zf01 ^ x;
          x = zf00 & tg_[w]; acc0[1] |= x; acc0[0] |= zf00 ^ x;
        }
#pragma unroll
        for (int pp = 0; pp < 8; pp++) {
          if (acc1[pp] && acc0[pp]) {
            ok = false;
            break;
          }
        }
      }
      if (!ok) continue;
      scan7_handle_survivor(args, ctl, s_pool, s_abcd[t], e, f, g);
    }
  }

  atomicAdd(&s_eval, static_cast<unsigned long long>(local_eval));
  __syncthreads();
  if (threadIdx.x == 0) atomicAdd(&ctl->evaluated, s_eval);
}

// ---------------------------------------------------------------------------
// K4b — 7-LUT function assignment over filtered hits: one thread per
// (hit, ordering); each runs the middle-function sweep + outer 2-coloring
// (replacing the reference's 70 x 256 x 256 brute force, lut.c:416-484).
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(SCAN_BLOCK) k_scan7_assign(
    const Hit7* hits, unsigned long long nhits, DevCtl* ctl, u64 seed) {
  // One thread per (hit, ordering, fm-octant): the 256-middle-function
  // sweep splits across 8 threads (disjoint 32-function slices of the
  // same shuffled order), which bounds the divergent tail of exhaustive
  // items at ~32 colorings and feeds the GPU 8x more parallel slack.
  constexpr u64 FM_SPLIT = 8;
  const u64 nitems = nhits * LUT7_NUM_ORDERINGS * FM_SPLIT;
  const u64 stride = static_cast<u64>(gridDim.x) * blockDim.x;
  int tick = 0;
  for (u64 item = blockIdx.x * static_cast<u64>(blockDim.x) + threadIdx.x;
       item < nitems; item += stride) {
    if (((tick++) & 3) == 0 && dev_abort(ctl)) return;
    const u64 base_item = item / FM_SPLIT;
    const int oct = static_cast<int>(item % FM_SPLIT);
    const Hit7& h = hits[base_item / LUT7_NUM_ORDERINGS];
    int oidx = static_cast<int>(base_item % LUT7_NUM_ORDERINGS);
    u8 ord[7];
    lut7_ordering(oidx, ord);
    u8 fo, fm, fi;
    // rnd keyed by (hit, ordering) so the 8 octants partition the same
    // shuffled middle-function order disjointly.
    if (lut7_solve_ordering(h.p1, h.p0, ord, dev_rnd(seed, base_item), &fo,
                            &fm, &fi, oct * 32, 32)) {
      u16 res[10];
      res[0] = fo;
      res[1] = fm;
      res[2] = fi;
      for (int j = 0; j < 7; j++) res[3 + j] = h.nums[ord[j]];
      dev_publish(ctl, res);
      return;
    }
  }
}

// ---------------------------------------------------------------------------
// Persistent k=4 scan service.
//
// Gate-mode searches issue one step-4 triple scan per recursion node
// (~600k nodes for AES bit 0); through the one-shot launch path each scan
// pays a ~30 us API floor (pool H2D + ctl round trips + launch + sync) that
// dwarfs the kernel time of small scans. The service keeps a grid of
// resident workgroups parked on a mailbox in fine-grained pinned host
// memory: the host publishes {request header, pool delta} with one release
// store, workgroup 0 copies the delta into the device pool and re-publishes
// device-side, and the last workgroup to finish writes the response
// directly back to pinned memory — no launches, no stream ops, ~5-8 us per
// scan. (Price anchors: MI355X_MICROARCH.md persistent-kernel price list —
// host-paired flag round trips are single-digit us; relaxed polls +
// s_sleep from one lane, one acquire fence on the hit.)
//
// Liveness/safety design (every wait is bounded):
//  * The kernel retires itself after ~2 ms idle (svc_state -> RETIRED) so
//    an idle service never starves other kernels of CUs; the host reaps
//    and relaunches on the next submit (and resolves the publish/retire
//    race by relaunching when it observes RETIRED while waiting).
//  * Residency self-check: workgroups count themselves in at startup; if
//    the grid is not fully resident within a deadline the host quits the
//    kernel and permanently falls back to the one-shot path.
//  * The host can always stop the kernel without its cooperation: leader
//    workgroup polls quit_req (host memory); the other workgroups poll
//    dev_seq (device memory) which the host can overwrite via an async
//    copy on a separate stream, and the scan loops poll ctl.abort which
//    the host can set the same way.
// ---------------------------------------------------------------------------

enum : u32 { SVC_RUNNING = 1u, SVC_RETIRED = 2u };
enum : u32 { SVC_CMD_WORK = 0u, SVC_CMD_QUIT = 1u };

// Request header word layout (u64[16] block, read cooperatively).
enum {
  HDR_N_KEEP = 0,     // n (lo32) | pool_keep (hi32)
  HDR_EPOCH_CALL = 1, // matcher_epoch (lo32) | count_all (hi32)
  HDR_BEGIN = 2,
  HDR_END = 3,
  HDR_SEED = 4,
  HDR_T1 = 8,         // 8..11
  HDR_T0 = 12,        // 12..15
  HDR_WORDS = 16,
};

struct SvcMailbox {  // pinned fine-grained host memory
  // host -> device control
  u64 req_seq;       // release-stored by the host to publish a request
  u32 quit_req;
  u32 svc_state;     // SVC_RUNNING / SVC_RETIRED
  // device -> host
  u64 resp_seq;      // flag-stored by the last workgroup (after a drain)
  u64 r_evaluated;
  u32 r_found;
  u32 alive;         // residency self-check counter
  u64 r_res_w[3];    // res[10] packed 4 x u16 per word
  // request payload
  alignas(64) u64 hdr[HDR_WORDS];
  alignas(64) ttable pool_staging[MAX_GATES];  // full pool image (host shadow)
  alignas(64) Avail3Matcher matcher_staging;
};

struct SvcDev {  // device memory
  u64 dev_seq;   // published seq for non-leader workgroups
  u32 dev_cmd;   // vestigial (quit now travels in dev_seq's low bit);
                 // kept so SVC_DEV_CTL_BYTES spans a stable layout
  u32 matcher_epoch;
  unsigned long long done;
  DevCtl ctl;
  alignas(64) u64 hdr[HDR_WORDS];
  alignas(64) ttable pool[MAX_GATES];
  alignas(64) Avail3Matcher matcher;
};

// Leader poll iterations before idle retirement (each ~1-2 us: one PCIe
// read + s_sleep) — about a second. Gate-mode searches interleave CPU
// scans between service requests, so a short timeout causes relaunch
// storms; in-process k=5/7 launches park the service explicitly instead.
inline int svc_idle_polls() {
  static const int v = [] {
    const char* s = std::getenv("SBOXGATES_SVC_IDLE");
    return s != nullptr ? std::atoi(s) : 500000;
  }();
  return v;
}

__global__ void __launch_bounds__(SCAN_BLOCK) k_scan4_service(SvcMailbox* mb,
                                                              SvcDev* dev,
                                                              int c_idle_polls) {
  __shared__ alignas(16) u64 s_pool[MAX_GATES * PSTR];
  __shared__ alignas(16) u8 s_bitmap[256 * 256 / 8];
  __shared__ u8 s_funs[256];
  __shared__ alignas(16) u64 s_hdr[HDR_WORDS];
  __shared__ u64 s_word;
  __shared__ int s_count;
  __shared__ unsigned long long s_eval;

  if (threadIdx.x == 0) {
    __hip_atomic_fetch_add(as_gu32(&mb->alive), 1u, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_SYSTEM);
  }

  const bool leader = blockIdx.x == 0;
  // Protocol word: (req_seq << 1) | quit. Monotonic; one word carries both
  // the sequence and the command, so publish/consume is a single-flag
  // hand-off (Guideline 16 R2 style).
  u64 served = 0;

  for (;;) {
    // ---- wait for a request (or quit) ----
    if (threadIdx.x == 0) {
      u64 word;
      if (leader) {
        int idle = 0;
        for (;;) {
          u64 rs = __hip_atomic_load(as_gu64(&mb->req_seq), __ATOMIC_RELAXED,
                                     __HIP_MEMORY_SCOPE_SYSTEM);
          if ((rs << 1) != served) {
            word = rs << 1;
            break;
          }
          if (__hip_atomic_load(as_gu32(&mb->quit_req), __ATOMIC_RELAXED,
                                __HIP_MEMORY_SCOPE_SYSTEM) != 0) {
            word = served + 3;  // bump seq, set quit bit
            break;
          }
          if (++idle > c_idle_polls) {
            // Retire. After this store the kernel never touches the
            // mailbox again; the host reaps and relaunches on demand.
            __hip_atomic_store(as_gu32(&mb->svc_state), SVC_RETIRED,
                               __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM);
            word = served + 3;
            break;
          }
          __builtin_amdgcn_s_sleep(32);
        }
      } else {
        for (;;) {
          u64 ds = __hip_atomic_load(as_gu64(&dev->dev_seq), __ATOMIC_RELAXED,
                                     __HIP_MEMORY_SCOPE_AGENT);
          if (ds != served) {
            word = ds;
            break;
          }
          __builtin_amdgcn_s_sleep(16);
        }
      }
      // ONE acquire after the poll match (covers the workgroup: a
      // __syncthreads() follows). System scope: the leader's payload
      // source is host memory.
      __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");
      s_word = word;
    }
    __syncthreads();
    const u64 word = s_word;
    const bool quit = (word & 1) != 0;
    const u64 seq = word >> 1;

    if (leader) {
      if (!quit) {
        // Pull the request: header block, pool delta, matcher delta.
        if (threadIdx.x < HDR_WORDS) s_hdr[threadIdx.x] = mb->hdr[threadIdx.x];
        __syncthreads();
        // Defensive clamps: a corrupt header must never walk past the
        // fixed-size pool buffers (a GPU page fault can wedge the node).
        const int n = std::min(
            static_cast<int>(s_hdr[HDR_N_KEEP] & 0xFFFFFFFFu), MAX_GATES);
        const int keep = std::max(
            0, std::min(static_cast<int>(s_hdr[HDR_N_KEEP] >> 32), n));
        const u32 epoch = static_cast<u32>(s_hdr[HDR_EPOCH_CALL] & 0xFFFFFFFFu);
        if (threadIdx.x < HDR_WORDS) dev->hdr[threadIdx.x] = s_hdr[threadIdx.x];
        {
          const u64* src = reinterpret_cast<const u64*>(mb->pool_staging);
          u64* dst = reinterpret_cast<u64*>(dev->pool);
          for (int i = keep * 4 + threadIdx.x; i < n * 4; i += blockDim.x) {
            dst[i] = src[i];
          }
        }
        if (epoch != dev->matcher_epoch) {
          const u64* ms = reinterpret_cast<const u64*>(&mb->matcher_staging);
          u64* md = reinterpret_cast<u64*>(&dev->matcher);
          // Round UP: sizeof(Avail3Matcher) is not a multiple of 8 and the
          // last partial word holds `count`; truncating it left s_count = 0
          // and made every early-exit probe silently miss. Both structs sit
          // in 64-aligned tails, so the overread is in-allocation padding.
          for (size_t i = threadIdx.x; i < (sizeof(Avail3Matcher) + 7) / 8;
               i += blockDim.x) {
            md[i] = ms[i];
          }
          __syncthreads();
          if (threadIdx.x == 0) dev->matcher_epoch = epoch;
        }
        if (threadIdx.x == 0) {
          dev->done = 0;
          dev->ctl.abort = 0;
          dev->ctl.lock = 0;
          dev->ctl.found = 0;
          dev->ctl.evaluated = 0;
          dev->ctl.queue = 0;
          dev->ctl.hit_count = 0;
          dev->ctl.overflow = 0;
        }
      }
      // Publish to the other workgroups (Guideline 16 counter form):
      // every wave drains its plain stores, then one lane releases and
      // stores the flag word relaxed.
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();
      if (threadIdx.x == 0) {
        __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __hip_atomic_store(as_gu64(&dev->dev_seq), word, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
      }
      __syncthreads();
    }
    if (quit) return;
    served = word;

    // ---- stage request + pool + matcher into LDS ----
    if (!leader) {
      if (threadIdx.x < HDR_WORDS) s_hdr[threadIdx.x] = dev->hdr[threadIdx.x];
    }
    __syncthreads();
    const int n = std::min(static_cast<int>(s_hdr[HDR_N_KEEP] & 0xFFFFFFFFu),
                           MAX_GATES);
    const int count_all = static_cast<int>(s_hdr[HDR_EPOCH_CALL] >> 32);
    const i64 begin = static_cast<i64>(s_hdr[HDR_BEGIN]);
    const i64 end = static_cast<i64>(s_hdr[HDR_END]);
    ttable T1, T0;
#pragma unroll
    for (int w = 0; w < 4; w++) {
      T1.w[w] = s_hdr[HDR_T1 + w];
      T0.w[w] = s_hdr[HDR_T0 + w];
    }
    load_pool_lds(s_pool, dev->pool, n);
    for (int i = threadIdx.x; i < 256 * 256 / 8 / 8; i += blockDim.x) {
      reinterpret_cast<u64*>(s_bitmap)[i] =
          reinterpret_cast<const u64*>(dev->matcher.bitmap)[i];
    }
    for (int i = threadIdx.x; i < 256 / 8; i += blockDim.x) {
      reinterpret_cast<u64*>(s_funs)[i] =
          reinterpret_cast<const u64*>(dev->matcher.funs)[i];
    }
    if (threadIdx.x == 0) {
      s_count = dev->matcher.count;
      s_eval = 0;
    }
    __syncthreads();

    // ---- scan ----
    const i64 stride = static_cast<i64>(gridDim.x) * blockDim.x;
    const i64 idx0 =
        begin + blockIdx.x * static_cast<i64>(blockDim.x) + threadIdx.x;
    u64 local_eval = scan4_loop(s_pool, s_bitmap, s_funs, s_count, n, T1, T0,
                                count_all, &dev->ctl, idx0, end, stride);

    // ---- completion: last workgroup publishes the response ----
    atomicAdd(&s_eval, static_cast<unsigned long long>(local_eval));
    __syncthreads();
    if (threadIdx.x == 0) {
      if (s_eval != 0) {
        __hip_atomic_fetch_add(as_gu64(&dev->ctl.evaluated),
                               static_cast<unsigned long long>(s_eval),
                               __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      }
      // Fence-then-ticket order (Guideline 16 split-K recipe).
      __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      unsigned long long d = __hip_atomic_fetch_add(
          as_gu64(&dev->done), 1ULL, __ATOMIC_RELAXED,
          __HIP_MEMORY_SCOPE_AGENT);
      if (d == gridDim.x - 1ULL) {
        __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
        const unsigned long long ev = __hip_atomic_load(
            as_gu64(&dev->ctl.evaluated), __ATOMIC_RELAXED,
            __HIP_MEMORY_SCOPE_AGENT);
        const u32 found = __hip_atomic_load(as_gu32(&dev->ctl.found),
                                            __ATOMIC_RELAXED,
                                            __HIP_MEMORY_SCOPE_AGENT);
        u64 rw[3] = {0, 0, 0};
        if (found != 0) {
          for (int i = 0; i < 10; i++) {
            rw[i >> 2] |= static_cast<u64>(dev->ctl.res[i]) << ((i & 3) * 16);
          }
        }
        // Response: relaxed system stores (write-through), drained before
        // the resp_seq flag store.
        __hip_atomic_store(as_gu64(&mb->r_evaluated), ev, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_SYSTEM);
        __hip_atomic_store(as_gu32(&mb->r_found), found, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_SYSTEM);
        for (int i = 0; i < 3; i++) {
          __hip_atomic_store(as_gu64(&mb->r_res_w[i]), rw[i], __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_SYSTEM);
        }
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __hip_atomic_store(as_gu64(&mb->resp_seq), seq, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_SYSTEM);
      }
    }
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// ScanService — host side of the persistent k=4 scan kernel. One instance
// per device per process (gate-mode engines and --jobs workers share it;
// requests serialize on a mutex, which is correct because every request is
// a whole-device scan anyway).
// ---------------------------------------------------------------------------
class ScanService {
 public:
  static std::shared_ptr<ScanService> acquire(int device, std::string* err);
  ~ScanService();

  // Blocking k=4 scan via the resident kernel. Thread-safe.
  ScanResult scan4(const ScanRequest& rq, i64 begin, i64 end);

  // Retire the resident kernel if it is running (used before launching
  // other kernels so an idle service never delays them). Cheap when the
  // kernel already retired itself.
  void park();

 private:
  explicit ScanService(int device);
  void ensure_running_locked();
  void quit_locked();
  bool quit_bounded();

  std::mutex mu_;
  int device_;
  hipStream_t stream_ = nullptr;      // service kernel lives here
  hipStream_t esc_stream_ = nullptr;  // escape-hatch async writes
  SvcMailbox* mb_ = nullptr;          // pinned fine-grained
  SvcDev* d_svc_ = nullptr;
  u64 seq_ = 0;
  u64 relaunches_ = 0;
  int grid_ = 0;
  bool running_ = false;
  int shadow_n_ = 0;                  // valid prefix of mb_->pool_staging
  u32 matcher_epoch_ = 0;
};

namespace {
std::mutex g_svc_mu;
std::map<int, std::weak_ptr<ScanService>> g_svc_registry;

// Size of the SvcDev control header (everything the host must reset before
// a relaunch; pool/matcher content survives, matcher_epoch reset forces a
// re-copy).
constexpr size_t SVC_DEV_CTL_BYTES = offsetof(SvcDev, hdr);
}  // namespace

static bool svc_debug() {
  static const bool on = [] {
    const char* s = std::getenv("SBOXGATES_SVC_DEBUG");
    return s != nullptr && s[0] != '\0' && s[0] != '0';
  }();
  return on;
}

ScanService::ScanService(int device) : device_(device) {
  SBG_HIP_CHECK(hipSetDevice(device_));
  SBG_HIP_CHECK(hipStreamCreate(&stream_));
  SBG_HIP_CHECK(hipStreamCreate(&esc_stream_));
  SBG_HIP_CHECK(hipHostMalloc(&mb_, sizeof(SvcMailbox), hipHostMallocCoherent));
  std::memset(mb_, 0, sizeof(SvcMailbox));
  SBG_HIP_CHECK(hipMalloc(&d_svc_, sizeof(SvcDev)));
  SBG_HIP_CHECK(hipMemset(d_svc_, 0, sizeof(SvcDev)));
  int per_cu = 0;
  SBG_HIP_CHECK(hipOccupancyMaxActiveBlocksPerMultiprocessor(
      &per_cu, reinterpret_cast<const void*>(k_scan4_service), SCAN_BLOCK, 0));
  hipDeviceProp_t prop;
  SBG_HIP_CHECK(hipGetDeviceProperties(&prop, device_));
  // Small grid by design: the scans the service exists for are <= ~3e5
  // candidates (deep-recursion step-4 scans), and the request/response
  // round trip scales with the completion barrier — the flat-counter
  // barrier price is ~7-26 us at 256-1024 workgroups
  // (MI355X_MICROARCH.md barrier-counter) but single-digit at 128. A
  // 128-WG grid also coexists with other kernels (half the CUs stay
  // empty), so an idle-resident service cannot starve them.
  grid_ = std::clamp(std::min(per_cu * prop.multiProcessorCount, 64), 8, 2048);
  if (const char* g = std::getenv("SBOXGATES_SVC_GRID")) {
    long v = std::strtol(g, nullptr, 10);
    if (v >= 1 && v <= 4096) grid_ = static_cast<int>(v);
  }
  if (svc_debug()) {
    std::fprintf(stderr, "[svc] device %d: occupancy %d/CU x %d CUs -> grid %d\n",
                 device_, per_cu, prop.multiProcessorCount, grid_);
  }
}

ScanService::~ScanService() {
  std::lock_guard<std::mutex> lk(mu_);
  bool freed_ok = true;
  if (running_) {
    (void)hipSetDevice(device_);
    freed_ok = quit_bounded();
    if (!freed_ok) {
      std::fprintf(stderr,
                   "sboxgates: scan service kernel did not exit; leaking its "
                   "resources\n");
    }
  }
  if (freed_ok) {
    if (d_svc_ != nullptr) (void)hipFree(d_svc_);
    if (mb_ != nullptr) (void)hipHostFree(mb_);
    if (stream_ != nullptr) (void)hipStreamDestroy(stream_);
    if (esc_stream_ != nullptr) (void)hipStreamDestroy(esc_stream_);
  }
}

// Ask the resident kernel to exit and wait for it, bounded. The primary
// channel is the pinned quit_req word (a plain host store — always lands):
// the leader workgroup polls it between requests and broadcasts QUIT to
// the other workgroups through device stores (which, being on-device,
// cannot be blocked by anything). The hipMemcpy escalation is best-effort
// only: small H2D copies may be executed as blit kernels, which cannot get
// CUs while the persistent grid holds them.
bool ScanService::quit_bounded() {
  __atomic_store_n(&mb_->quit_req, 1u, __ATOMIC_RELEASE);
  const auto t0 = std::chrono::steady_clock::now();
  bool escalated = false;
  for (;;) {
    hipError_t q = hipStreamQuery(stream_);
    if (q != hipErrorNotReady) {
      running_ = false;
      mb_->quit_req = 0;
      return true;
    }
    const auto dt = std::chrono::steady_clock::now() - t0;
    if (!escalated && dt > std::chrono::seconds(5)) {
      escalated = true;
      static const u32 one = 1;
      static const u64 big_seq = ~0ULL;
      (void)hipMemcpyAsync(&d_svc_->ctl.abort, &one, sizeof(u32),
                           hipMemcpyHostToDevice, esc_stream_);
      (void)hipMemcpyAsync(&d_svc_->dev_cmd, &one, sizeof(u32),
                           hipMemcpyHostToDevice, esc_stream_);
      (void)hipMemcpyAsync(&d_svc_->dev_seq, &big_seq, sizeof(u64),
                           hipMemcpyHostToDevice, esc_stream_);
    }
    if (dt > std::chrono::seconds(10)) return false;
    std::this_thread::yield();
  }
}

std::shared_ptr<ScanService> ScanService::acquire(int device, std::string* err) {
  std::lock_guard<std::mutex> lk(g_svc_mu);
  auto& slot = g_svc_registry[device];
  auto p = slot.lock();
  if (p != nullptr) return p;
  try {
    p = std::shared_ptr<ScanService>(new ScanService(device));
  } catch (const std::exception& e) {
    if (err != nullptr) *err = e.what();
    return nullptr;
  }
  slot = p;
  return p;
}

void ScanService::ensure_running_locked() {
  if (running_) {
    if (__atomic_load_n(&mb_->svc_state, __ATOMIC_ACQUIRE) == SVC_RUNNING) {
      return;
    }
    // The kernel retired itself; reap it.
    SBG_HIP_CHECK(hipStreamSynchronize(stream_));
    running_ = false;
  }
  SBG_HIP_CHECK(hipMemset(d_svc_, 0, SVC_DEV_CTL_BYTES));
  mb_->quit_req = 0;
  mb_->alive = 0;
  __atomic_store_n(&mb_->svc_state, SVC_RUNNING, __ATOMIC_RELEASE);
  hipLaunchKernelGGL(k_scan4_service, dim3(grid_), dim3(SCAN_BLOCK), 0, stream_,
                     mb_, d_svc_, svc_idle_polls());
  SBG_HIP_CHECK(hipGetLastError());
  running_ = true;
  relaunches_ += 1;
  // Residency self-check: all workgroups must report in, otherwise the
  // completion protocol would deadlock. (Occupancy-API sizing should make
  // this impossible; a failure means something else holds CUs.)
  const auto t0 = std::chrono::steady_clock::now();
  while (__atomic_load_n(&mb_->alive, __ATOMIC_ACQUIRE) <
         static_cast<u32>(grid_)) {
    if (std::chrono::steady_clock::now() - t0 > std::chrono::seconds(2)) {
      const u32 alive = __atomic_load_n(&mb_->alive, __ATOMIC_ACQUIRE);
      if (svc_debug()) {
        std::fprintf(stderr, "[svc] residency failure: %u of %d alive\n", alive,
                     grid_);
      }
      quit_locked();
      throw std::runtime_error("scan service grid failed to become resident");
    }
    std::this_thread::yield();
  }
  if (svc_debug()) {
    std::fprintf(stderr, "[svc] launch #%llu: %d workgroups resident in %.1f us\n",
                 static_cast<unsigned long long>(relaunches_), grid_,
                 std::chrono::duration<double, std::micro>(
                     std::chrono::steady_clock::now() - t0)
                     .count());
  }
}

void ScanService::quit_locked() {
  if (!running_) return;
  if (!quit_bounded()) {
    throw std::runtime_error("scan service kernel refused to exit");
  }
}

void ScanService::park() {
  std::lock_guard<std::mutex> lk(mu_);
  if (!running_) return;
  (void)hipSetDevice(device_);
  if (__atomic_load_n(&mb_->svc_state, __ATOMIC_ACQUIRE) == SVC_RETIRED) {
    SBG_HIP_CHECK(hipStreamSynchronize(stream_));
    running_ = false;
    return;
  }
  quit_locked();
}

ScanResult ScanService::scan4(const ScanRequest& rq, i64 begin, i64 end) {
  std::lock_guard<std::mutex> lk(mu_);
  SBG_HIP_CHECK(hipSetDevice(device_));

  // Matcher delta (content compare: pointers can be reused across engine
  // lifetimes, so identity alone is not a safe key).
  if (std::memcmp(&mb_->matcher_staging, rq.matcher, sizeof(Avail3Matcher)) !=
      0) {
    std::memcpy(&mb_->matcher_staging, rq.matcher, sizeof(Avail3Matcher));
    matcher_epoch_ += 1;
  }

  // Pool delta against the pinned shadow copy.
  int keep = 0;
  {
    const u64* a = reinterpret_cast<const u64*>(mb_->pool_staging);
    const u64* b = reinterpret_cast<const u64*>(rq.tables);
    const int limit = std::min(shadow_n_, rq.n) * 4;
    int i = 0;
    while (i < limit && a[i] == b[i]) i++;
    keep = i / 4;
    if (keep < rq.n) {
      std::memcpy(mb_->pool_staging + keep, rq.tables + keep,
                  sizeof(ttable) * static_cast<size_t>(rq.n - keep));
    }
    shadow_n_ = rq.n;
  }

  // Publish the request.
  const ttable T1 = rq.target & rq.mask;
  const ttable T0 = ~rq.target & rq.mask;
  mb_->hdr[HDR_N_KEEP] = static_cast<u64>(static_cast<u32>(rq.n)) |
                         (static_cast<u64>(static_cast<u32>(keep)) << 32);
  mb_->hdr[HDR_EPOCH_CALL] =
      static_cast<u64>(matcher_epoch_) |
      (static_cast<u64>(rq.count_all ? 1u : 0u) << 32);
  mb_->hdr[HDR_BEGIN] = static_cast<u64>(begin);
  mb_->hdr[HDR_END] = static_cast<u64>(end);
  mb_->hdr[HDR_SEED] = rq.seed;
  for (int w = 0; w < 4; w++) {
    mb_->hdr[HDR_T1 + w] = T1.w[w];
    mb_->hdr[HDR_T0 + w] = T0.w[w];
  }
  ensure_running_locked();
  const u64 s = ++seq_;
  __atomic_store_n(&mb_->req_seq, s, __ATOMIC_RELEASE);

  // Wait for the response. Every exit path is bounded: a retire race
  // relaunches, a stuck scan is aborted via the escape stream.
  const auto t0 = std::chrono::steady_clock::now();
  bool escalated = false;
  bool reported = false;
  int spins = 0;
  for (;;) {
    if (__atomic_load_n(&mb_->resp_seq, __ATOMIC_ACQUIRE) == s) break;
    if (__atomic_load_n(&mb_->svc_state, __ATOMIC_ACQUIRE) == SVC_RETIRED) {
      // The kernel retired just as we published: reap and relaunch; the
      // request is still in the mailbox and the fresh kernel serves it.
      SBG_HIP_CHECK(hipStreamSynchronize(stream_));
      running_ = false;
      ensure_running_locked();
    }
    const auto dt = std::chrono::steady_clock::now() - t0;
    if (svc_debug() && !reported && dt > std::chrono::seconds(1)) {
      reported = true;
      std::fprintf(stderr,
                   "[svc] slow request: seq=%llu resp=%llu state=%u alive=%u "
                   "n=%d range=%lld\n",
                   static_cast<unsigned long long>(s),
                   static_cast<unsigned long long>(mb_->resp_seq),
                   mb_->svc_state, mb_->alive, rq.n,
                   static_cast<long long>(end - begin));
    }
    if (!escalated && dt > std::chrono::seconds(10)) {
      escalated = true;
      static const u32 one = 1;
      (void)hipMemcpyAsync(&d_svc_->ctl.abort, &one, sizeof(u32),
                           hipMemcpyHostToDevice, esc_stream_);
    }
    if (dt > std::chrono::seconds(20)) {
      quit_locked();
      throw std::runtime_error("scan service request timed out");
    }
    if (((spins++) & 0x3FF) == 0x3FF) std::this_thread::yield();
  }
  if (escalated) {
    quit_locked();
    throw std::runtime_error("scan service request needed an abort");
  }

  ScanResult out;
  out.evaluated = mb_->r_evaluated;
  if (out.evaluated > static_cast<u64>(end - begin)) {
    // A response that claims more work than the range is corruption;
    // treat it like a service failure (caller falls back to one-shot).
    quit_locked();
    throw std::runtime_error("scan service returned corrupt counts");
  }
  if (mb_->r_found != 0) {
    out.found = true;
    for (int i = 0; i < 10; i++) {
      out.res[i] = static_cast<u16>(mb_->r_res_w[i >> 2] >> ((i & 3) * 16));
    }
  }
  return out;
}

// ---------------------------------------------------------------------------
// GpuEngine host wrapper.
// ---------------------------------------------------------------------------

struct GpuEngine::Impl {
  int device = 0;
  hipStream_t stream = nullptr;
  ttable* d_pool = nullptr;
  DevCtl* d_ctl = nullptr;
  Hit7* d_hits = nullptr;
  Avail3Matcher* d_matcher = nullptr;
  ttable* h_pool = nullptr;  // pinned staging: pageable-source async
                             // copies cost ~100s of us on small scans
  const Avail3Matcher* last_matcher = nullptr;  // uploaded-matcher cache
  u64 hit_cap = 0;
  DevCtl* h_ctl = nullptr;  // pinned staging
  std::string name;
  std::shared_ptr<ScanService> svc;  // lazily acquired for k=4 scans
  bool svc_tried = false;
  bool svc_broken = false;

  ~Impl() {
    if (d_pool != nullptr) (void)hipFree(d_pool);
    if (d_matcher != nullptr) (void)hipFree(d_matcher);
    if (d_ctl != nullptr) (void)hipFree(d_ctl);
    if (d_hits != nullptr) (void)hipFree(d_hits);
    if (h_ctl != nullptr) (void)hipHostFree(h_ctl);
    if (h_pool != nullptr) (void)hipHostFree(h_pool);
    if (stream != nullptr) (void)hipStreamDestroy(stream);
  }
};

bool gpu_available() { return gpu_count() > 0; }

int gpu_count() {
  int count = 0;
  if (hipGetDeviceCount(&count) != hipSuccess) return 0;
  return count;
}

std::unique_ptr<GpuEngine> GpuEngine::create(int device, std::string* err) {
  int count = 0;
  hipError_t e = hipGetDeviceCount(&count);
  if (e != hipSuccess || count == 0) {
    if (err != nullptr) {
      *err = e == hipSuccess ? "no HIP devices visible" : hipGetErrorString(e);
    }
    return nullptr;
  }
  try {
    auto impl = std::make_unique<Impl>();
    if (device >= 0) {
      SBG_HIP_CHECK(hipSetDevice(device));
      impl->device = device;
    } else {
      SBG_HIP_CHECK(hipGetDevice(&impl->device));
    }
    hipDeviceProp_t prop;
    SBG_HIP_CHECK(hipGetDeviceProperties(&prop, impl->device));
    impl->name = prop.name;
    SBG_HIP_CHECK(hipStreamCreate(&impl->stream));
    SBG_HIP_CHECK(hipMalloc(&impl->d_pool, sizeof(ttable) * MAX_GATES));
    SBG_HIP_CHECK(hipMalloc(&impl->d_ctl, sizeof(DevCtl)));
    SBG_HIP_CHECK(hipMalloc(&impl->d_matcher, sizeof(Avail3Matcher)));
    SBG_HIP_CHECK(hipHostMalloc(&impl->h_ctl, sizeof(DevCtl)));
    SBG_HIP_CHECK(hipHostMalloc(&impl->h_pool, sizeof(ttable) * MAX_GATES));
    // Hit buffer for the 7-LUT frontier: default 16M hits (768 MB) per
    // chunk; overridable for memory-constrained runs. Allocated lazily on
    // the first 7-LUT scan (gate-mode engines never need it).
    const char* cap_env = std::getenv("SBOXGATES_HIT_CAP");
    impl->hit_cap = cap_env != nullptr ? std::strtoull(cap_env, nullptr, 10)
                                       : (1ULL << 24);
    return std::unique_ptr<GpuEngine>(new GpuEngine(impl.release()));
  } catch (const std::exception& ex) {
    if (err != nullptr) *err = ex.what();
    return nullptr;
  }
}

GpuEngine::~GpuEngine() { delete impl_; }

int GpuEngine::device() const { return impl_->device; }
std::string GpuEngine::device_name() const { return impl_->name; }

bool GpuEngine::scan4_service_active() {
  Impl* im = impl_;
  if (im->svc != nullptr) return true;
  if (im->svc_broken || im->svc_tried) return false;
  im->svc_tried = true;
  const char* off = std::getenv("SBOXGATES_NO_SVC");
  if (off != nullptr && off[0] != '\0' && off[0] != '0') {
    im->svc_broken = true;
    return false;
  }
  std::string err;
  im->svc = ScanService::acquire(im->device, &err);
  if (im->svc == nullptr) {
    im->svc_broken = true;
    return false;
  }
  return true;
}

ScanResult GpuEngine::scan(int k, const ScanRequest& rq, i64 begin, i64 end) {
  ScanResult out;
  const i64 total = n_choose_k(rq.n, k == 4 ? 3 : k);
  if (begin >= total) return out;
  if (end > total) end = total;
  if (begin >= end) return out;

  Impl* im = impl_;
  SBG_HIP_CHECK(hipSetDevice(im->device));

  // Service only for small/medium ranges: its 64-WG grid is sized for the
  // request/response round trip, not for multi-million-candidate scans —
  // those go to the one-shot kernel with a work-sized grid.
  if (k == 4 && end - begin <= (1 << 20) && scan4_service_active()) {
    if (rq.matcher == nullptr) throw std::runtime_error("scan4 needs matcher");
    try {
      return im->svc->scan4(rq, begin, end);
    } catch (const std::exception& e) {
      // Service failure (residency/timeout): permanently fall back to the
      // one-shot launch path for this engine.
      std::fprintf(stderr, "sboxgates: scan service disabled: %s\n", e.what());
      im->svc.reset();
      im->svc_broken = true;
    }
  } else if (k != 4 && im->svc != nullptr) {
    // Do not let an idle-resident service delay other kernels.
    im->svc->park();
  }

  // Upload the pool (through pinned staging) and reset the control block.
  std::memcpy(im->h_pool, rq.tables, sizeof(ttable) * rq.n);
  SBG_HIP_CHECK(hipMemcpyAsync(im->d_pool, im->h_pool, sizeof(ttable) * rq.n,
                               hipMemcpyHostToDevice, im->stream));
  std::memset(im->h_ctl, 0, sizeof(DevCtl));

  ScanArgs args;
  args.pool = im->d_pool;
  args.ctl = im->d_ctl;
  args.hits = im->d_hits;
  args.hit_cap = im->hit_cap;
  args.matcher = im->d_matcher;
  args.T1 = rq.target & rq.mask;
  args.T0 = ~rq.target & rq.mask;
  args.n = rq.n;
  args.begin = begin;
  args.end = end;
  args.excl = rq.excl_low64;
  args.seed = rq.seed;
  args.count_all = rq.count_all ? 1 : 0;
  args.slices7 = 1;

  if (k == 3 || k == 4) {
    if (k == 4) {
      if (rq.matcher == nullptr) throw std::runtime_error("scan4 needs matcher");
      if (im->last_matcher != rq.matcher) {
        // The matcher is per-engine and immutable; upload once.
        SBG_HIP_CHECK(hipMemcpy(im->d_matcher, rq.matcher, sizeof(Avail3Matcher),
                                hipMemcpyHostToDevice));
        im->last_matcher = rq.matcher;
      }
    }
    SBG_HIP_CHECK(hipMemcpyAsync(im->d_ctl, im->h_ctl, sizeof(DevCtl),
                                 hipMemcpyHostToDevice, im->stream));
    i64 range = end - begin;
    int grid = static_cast<int>(std::min<i64>((range + SCAN_BLOCK - 1) / SCAN_BLOCK,
                                              4096));
    if (k == 3) {
      hipLaunchKernelGGL(k_scan3, dim3(grid), dim3(SCAN_BLOCK), 0, im->stream, args);
    } else {
      hipLaunchKernelGGL(k_scan4, dim3(grid), dim3(SCAN_BLOCK), 0, im->stream, args);
    }
    SBG_HIP_CHECK(hipGetLastError());
    SBG_HIP_CHECK(hipMemcpyAsync(im->h_ctl, im->d_ctl, sizeof(DevCtl),
                                 hipMemcpyDeviceToHost, im->stream));
    SBG_HIP_CHECK(hipStreamSynchronize(im->stream));
  } else if (k == 5) {
    // Seed the triple queue at the triple containing `begin`.
    gatenum first[5];
    nth_combination(begin, rq.n, 5, 0, first);
    i64 t_begin = combination_rank(first, 3, rq.n);
    gatenum last[5];
    nth_combination(end - 1, rq.n, 5, 0, last);
    i64 t_last = combination_rank(last, 3, rq.n);
    im->h_ctl->queue = static_cast<unsigned long long>(t_begin);
    SBG_HIP_CHECK(hipMemcpyAsync(im->d_ctl, im->h_ctl, sizeof(DevCtl),
                                 hipMemcpyHostToDevice, im->stream));
    // Size the grid to the triple count: small scans (deep recursion) pay
    // for pool staging per block, so an always-2048 grid costs ~10x the
    // useful work there.
    i64 tcount = t_last - t_begin + 1;
    int grid = static_cast<int>(std::min<i64>((tcount + 7) / 8 + 1, 2048));
    hipLaunchKernelGGL(k_scan5, dim3(grid), dim3(SCAN_BLOCK), 0, im->stream, args);
    SBG_HIP_CHECK(hipGetLastError());
    SBG_HIP_CHECK(hipMemcpyAsync(im->h_ctl, im->d_ctl, sizeof(DevCtl),
                                 hipMemcpyDeviceToHost, im->stream));
    SBG_HIP_CHECK(hipStreamSynchronize(im->stream));
  } else if (k == 7) {
    if (im->d_hits == nullptr) {
      SBG_HIP_CHECK(hipMalloc(&im->d_hits, sizeof(Hit7) * im->hit_cap));
      // args was built before the allocation; refresh the pointer.
    }
    args.hits = im->d_hits;
    // Filter + assign, with overflow-driven range splitting.
    i64 lo = begin;
    while (lo < end) {
      i64 hi = end;
      for (;;) {
        std::memset(im->h_ctl, 0, sizeof(DevCtl));
        gatenum first[7];
        nth_combination(lo, rq.n, 7, 0, first);
        i64 q_begin = combination_rank(first, 4, rq.n);
        gatenum last[7];
        nth_combination(hi - 1, rq.n, 7, 0, last);
        i64 q_last = combination_rank(last, 4, rq.n);
        i64 qcount = q_last - q_begin + 1;
        // Sub-quad slicing: a single quad prefix can hold C(n-4,3) triples
        // (tens of thousands); without slicing a small scan runs on a
        // handful of blocks and leaves the chip idle.
        i64 want_units = (hi - lo + 8 * SCAN_BLOCK - 1) / (8 * SCAN_BLOCK);
        int slices = static_cast<int>(
            std::clamp<i64>(want_units / std::max<i64>(qcount, 1), 1, 256));
        ScanArgs a2 = args;
        a2.begin = lo;
        a2.end = hi;
        a2.slices7 = slices;
        im->h_ctl->queue = static_cast<unsigned long long>(q_begin * slices);
        SBG_HIP_CHECK(hipMemcpyAsync(im->d_ctl, im->h_ctl, sizeof(DevCtl),
                                     hipMemcpyHostToDevice, im->stream));
        int grid7 = static_cast<int>(
            std::min<i64>((qcount * slices + 3) / 4 + 1, 2048));
        hipLaunchKernelGGL(k_scan7_filter, dim3(grid7), dim3(SCAN_BLOCK), 0,
                           im->stream, a2);
        SBG_HIP_CHECK(hipGetLastError());
        SBG_HIP_CHECK(hipMemcpyAsync(im->h_ctl, im->d_ctl, sizeof(DevCtl),
                                     hipMemcpyDeviceToHost, im->stream));
        SBG_HIP_CHECK(hipStreamSynchronize(im->stream));
        if (im->h_ctl->overflow == 0) break;
        // Too many feasible combinations for the buffer: halve the range.
        hi = lo + (hi - lo) / 2;
        if (hi <= lo + 1) {
          throw std::runtime_error("7-LUT hit buffer too small for one combo");
        }
      }
      // Accounting invariant: h_ctl is zeroed at the top of every attempt,
      // so only the final (non-overflowed) attempt's counter lands here —
      // combinations re-scanned after an overflow are never double-counted.
      out.evaluated += im->h_ctl->evaluated;
      unsigned long long nhits = im->h_ctl->hit_count;
      if (nhits > im->hit_cap) nhits = im->hit_cap;
      if (nhits > 0 && !rq.count_all) {
        u64 items = nhits * LUT7_NUM_ORDERINGS * 8;
        int grid = static_cast<int>(
            std::min<u64>((items + SCAN_BLOCK - 1) / SCAN_BLOCK, 4096));
        hipLaunchKernelGGL(k_scan7_assign, dim3(grid), dim3(SCAN_BLOCK), 0,
                           im->stream, im->d_hits, nhits, im->d_ctl, rq.seed);
        SBG_HIP_CHECK(hipGetLastError());
        SBG_HIP_CHECK(hipMemcpyAsync(im->h_ctl, im->d_ctl, sizeof(DevCtl),
                                     hipMemcpyDeviceToHost, im->stream));
        SBG_HIP_CHECK(hipStreamSynchronize(im->stream));
        if (im->h_ctl->found != 0) {
          out.found = true;
          std::memcpy(out.res, im->h_ctl->res, sizeof(out.res));
          return out;
        }
      }
      lo = hi;
    }
    return out;
  } else {
    throw std::runtime_error("GpuEngine::scan: bad k");
  }

  out.evaluated = im->h_ctl->evaluated;
  if (im->h_ctl->found != 0) {
    out.found = true;
    std::memcpy(out.res, im->h_ctl->res, sizeof(out.res));
  }
  return out;
}

}  // namespace sbg
