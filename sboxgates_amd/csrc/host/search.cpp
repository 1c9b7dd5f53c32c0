// search.cpp — Kwan's iterative gate-addition search, the LUT search and
// the multi-output beam driver. See search.hpp for parity notes.

#include "sbg/search.hpp"

#include <algorithm>
#include <cassert>
#include <thread>
#include <vector>
#include <chrono>
#include <climits>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <stdexcept>

#include "sbg/comb.hpp"
#include "sbg/gpu.hpp"
#include "sbg/lutcover.hpp"
#include "sbg/xmlio.hpp"

namespace sbg {

// ---------------------------------------------------------------------------
// Gate-append primitives (parity: sboxgates.c:95-229).
// ---------------------------------------------------------------------------

gatenum add_gate(state* st, int type, gatenum gid1, gatenum gid2, metric_t metric) {
  assert(!(type == NOT && gid2 != NO_GATE));
  assert(type != IN && type != LUT);
  if (gid1 == NO_GATE || (gid2 == NO_GATE && type != NOT)) return NO_GATE;
  // Hard capacity check first: the reference's `num_gates > max_gates` test
  // alone writes one past gates[MAX_GATES-1] when max_gates == MAX_GATES
  // (the init_state default) and a mux chain reaches exactly MAX_GATES
  // gates — an out-of-bounds write in the reference; rejected here.
  if (st->num_gates >= MAX_GATES) return NO_GATE;
  if (st->num_gates > st->max_gates) return NO_GATE;
  if (metric == METRIC_SAT && st->sat_metric > st->max_sat_metric) return NO_GATE;
  assert(gid1 < st->num_gates);
  assert(gid2 < st->num_gates || type == NOT);
  assert(gid1 != gid2);

  st->sat_metric += sat_metric_of(type);
  gate& g = st->gates[st->num_gates];
  if (type == NOT) {
    g.table = ~st->gates[gid1].table;
  } else {
    g.table = gen_ttable_2(type, st->gates[gid1].table, st->gates[gid2].table);
  }
  g.type = type;
  g.in1 = gid1;
  g.in2 = gid2;
  g.in3 = NO_GATE;
  g.function = 0;
  st->num_gates += 1;
  return static_cast<gatenum>(st->num_gates - 1);
}

gatenum add_not_gate(state* st, gatenum gid, metric_t metric) {
  if (gid == NO_GATE) return NO_GATE;
  return add_gate(st, NOT, gid, NO_GATE, metric);
}

gatenum add_lut(state* st, u8 func, const ttable& table, gatenum g1, gatenum g2,
                gatenum g3) {
  if (g1 == NO_GATE || g2 == NO_GATE || g3 == NO_GATE ||
      st->num_gates >= MAX_GATES || st->num_gates > st->max_gates) {
    return NO_GATE;
  }
  assert(g1 < st->num_gates && g2 < st->num_gates && g3 < st->num_gates);
  assert(g1 != g2 && g2 != g3 && g3 != g1);
  gate& g = st->gates[st->num_gates];
  g.table = table;
  g.type = LUT;
  g.in1 = g1;
  g.in2 = g2;
  g.in3 = g3;
  g.function = func;
  st->num_gates += 1;
  return static_cast<gatenum>(st->num_gates - 1);
}

static gatenum add_and_gate(state* st, gatenum g1, gatenum g2, metric_t metric) {
  if (g1 == NO_GATE || g2 == NO_GATE) return NO_GATE;
  if (g1 == g2) return g1;
  return add_gate(st, AND, g1, g2, metric);
}

static gatenum add_or_gate(state* st, gatenum g1, gatenum g2, metric_t metric) {
  if (g1 == NO_GATE || g2 == NO_GATE) return NO_GATE;
  if (g1 == g2) return g1;
  return add_gate(st, OR, g1, g2, metric);
}

static gatenum add_xor_gate(state* st, gatenum g1, gatenum g2, metric_t metric) {
  if (g1 == NO_GATE || g2 == NO_GATE) return NO_GATE;
  // XOR(x, x) is constant FALSE — a degenerate multiplexer (both
  // half-space solutions equal the selector bit). The reference ABORTS
  // here (add_gate's gid1 != gid2 assert, sboxgates.c:103, reachable for
  // degenerate targets); skipping the variant lets the search continue.
  if (g1 == g2) return NO_GATE;
  return add_gate(st, XOR, g1, g2, metric);
}

gatenum add_boolfunc_2(state* st, const boolfunc& fun, gatenum g1, gatenum g2,
                       metric_t metric) {
  assert(fun.num_inputs == 2);
  if (g1 == NO_GATE || g2 == NO_GATE || st->num_gates > st->max_gates) return NO_GATE;
  if (metric == METRIC_SAT && st->sat_metric > st->max_sat_metric) return NO_GATE;
  if (fun.not_a) g1 = add_not_gate(st, g1, metric);
  if (fun.not_b) g2 = add_not_gate(st, g2, metric);
  gatenum gid = add_gate(st, fun.fun1, g1, g2, metric);
  if (fun.not_out) gid = add_not_gate(st, gid, metric);
  return gid;
}

gatenum add_boolfunc_3(state* st, const boolfunc& fun, gatenum g1, gatenum g2,
                       gatenum g3, metric_t metric) {
  if (g1 == NO_GATE || g2 == NO_GATE || (g3 == NO_GATE && fun.num_inputs == 3) ||
      st->num_gates > st->max_gates) {
    return NO_GATE;
  }
  if (metric == METRIC_SAT && st->sat_metric > st->max_sat_metric) return NO_GATE;
  if (fun.not_a) g1 = add_not_gate(st, g1, metric);
  if (fun.not_b) g2 = add_not_gate(st, g2, metric);
  if (fun.not_c) g3 = add_not_gate(st, g3, metric);
  gatenum out1 = add_gate(st, fun.fun1, g1, g2, metric);
  if (fun.not_out) {
    return add_not_gate(st, add_gate(st, fun.fun2, out1, g3, metric), metric);
  }
  return add_gate(st, fun.fun2, out1, g3, metric);
}

u8 eval_circuit(const state& st, u8 input) {
  // Straight DAG evaluation on a single input pattern — the ground-truth
  // oracle for tests (independent of cached truth tables).
  bool val[MAX_GATES];
  int ninputs = get_num_inputs(&st);
  for (int i = 0; i < st.num_gates; i++) {
    const gate& g = st.gates[i];
    switch (g.type) {
      case IN:
        val[i] = ((input >> i) & 1) != 0;
        break;
      case NOT:
        val[i] = !val[g.in1];
        break;
      case LUT: {
        int p = (val[g.in1] << 2) | (val[g.in2] << 1) | (val[g.in3] ? 1 : 0);
        val[i] = ((g.function >> p) & 1) != 0;
        break;
      }
      default: {
        int pat = ((val[g.in1] ? 1 : 0) << 1) | (val[g.in2] ? 1 : 0);
        val[i] = fun2_val(static_cast<u8>(g.type), static_cast<u8>(pat)) != 0;
        break;
      }
    }
  }
  (void)ninputs;
  u8 out = 0;
  for (int b = 0; b < 8; b++) {
    if (st.outputs[b] != NO_GATE && val[st.outputs[b]]) out |= 1u << b;
  }
  return out;
}

// ---------------------------------------------------------------------------
// Return-assertion (parity: ASSERT_AND_RETURN, sboxgates.h:31-44). A failed
// assertion indicates an engine/kernel bug; throw instead of abort so the
// Python bindings surface it loudly.
// ---------------------------------------------------------------------------
static gatenum assert_ret(gatenum ret, const ttable& target, const state* st,
                          const ttable& mask, const char* where) {
  if (ret == NO_GATE || tt_eq_mask(target, st->gates[ret].table, mask)) return ret;
  throw std::runtime_error(std::string("sboxgates: return assertion failed in ") +
                           where);
}

// ---------------------------------------------------------------------------
// Cell-requirement helpers for the step-3/4 scans: the forced function bits
// a candidate pair/triple imposes, and permutations thereof.
// ---------------------------------------------------------------------------

// Per-gate masked halves, precomputed once per node for the step-3/4a
// pair loops: p = t&T1, q = ~t&T1, r = t&T0, s = ~t&T0. A pair cell's
// masked content is then one AND of two halves ((±tx)&(±ty)&T1 =
// (±tx&T1)&(±ty&T1)), which removes the per-pair cell construction and
// the ~t materializations — the pair loops are ~60% of gate-mode host
// time (profiles/gate_mode_service.md).
struct GateHalves {
  ttable p, q, r, s;
};

static void fill_halves(GateHalves* out, const state* st, int n, const ttable& T1,
                        const ttable& T0) {
  for (int i = 0; i < n; i++) {
    const ttable& t = st->gates[i].table;
    out[i].p = t & T1;
    out[i].q = ~t & T1;
    out[i].r = t & T0;
    out[i].s = ~t & T0;
  }
}

// 4-cell requirements for a pair (x, y): bit p of req*/care is pattern
// p = vx<<1 | vy. Returns false if some cell is contradictory (no 2-input
// function can match).
static bool pair_requirements(const GateHalves& x, const GateHalves& y, u8* req1,
                              u8* care) {
  u8 r1 = 0, r0 = 0;
  for (int p = 0; p < 4; p++) {
    bool has1 = tt_any((p & 2 ? x.p : x.q) & (p & 1 ? y.p : y.q));
    bool has0 = tt_any((p & 2 ? x.r : x.s) & (p & 1 ? y.r : y.s));
    if (has1 && has0) return false;
    if (has1) r1 |= 1u << p;
    if (has0) r0 |= 1u << p;
  }
  *req1 = r1;
  *care = static_cast<u8>(r1 | r0);
  return true;
}

// Pattern-space truth table of a 2-input function: bit p = value at
// pattern p (the 4-bit encoding is bit-reversed; see boolfunc.hpp).
static inline u8 fun2_pattern_table(u8 fun) {
  u8 out = 0;
  for (u8 p = 0; p < 4; p++) out |= static_cast<u8>(fun2_val(fun, p) << p);
  return out;
}

// Swap the two input roles of a 4-bit pattern mask (pattern vx<<1|vy ->
// vy<<1|vx): bits 1 and 2 exchange.
static inline u8 swap_pair_patterns(u8 m) {
  return static_cast<u8>((m & 0x9) | ((m & 2) << 1) | ((m & 4) >> 1));
}

// ---------------------------------------------------------------------------
// Engine
// ---------------------------------------------------------------------------

Engine::Engine(const options& opt, DistCtx* ctx)
    : opt_(opt), ctx_(ctx != nullptr ? ctx : &local_) {
  if (opt_.seeded) {
    // Distinct per-rank streams from one seed.
    rng_.seed_splitmix(opt_.seed + 0x100000001ULL * static_cast<u64>(ctx_->rank()));
  }
  if (opt_.gpu != GPU_OFF) {
    std::string err;
    gpu_ = GpuEngine::create(opt_.gpu_device, &err);
    if (gpu_ == nullptr && opt_.gpu == GPU_FORCE) {
      throw std::runtime_error("sboxgates: GPU required but unavailable: " + err);
    }
  }
}

Engine::~Engine() = default;

const Avail3Matcher* Engine::matcher3() {
  if (matcher_ == nullptr) {
    matcher_ = std::make_unique<Avail3Matcher>();
    u8 funs[256], costs[256];
    for (int i = 0; i < opt_.num_avail_3; i++) {
      const boolfunc& f = opt_.avail_3[i];
      funs[i] = f.fun;
      costs[i] = static_cast<u8>(2 + (f.not_a ? 1 : 0) + (f.not_b ? 1 : 0) +
                                 (f.not_c ? 1 : 0) + (f.not_out ? 1 : 0));
    }
    build_avail3_matcher(funs, costs, opt_.num_avail_3, matcher_.get());
  }
  return matcher_.get();
}

bool Engine::gpu_active() const { return gpu_ != nullptr; }

void Engine::set_sbox(const u8 sbox[256], int num_inputs) {
  std::memcpy(sbox_, sbox, 256);
  num_inputs_ = num_inputs;
  for (u8 i = 0; i < 8; i++) g_target_[i] = generate_target(i, sbox_);
  num_outputs_ = 0;
  for (int i = 7; i >= 0; i--) {
    if (tt_any(g_target_[i])) {
      num_outputs_ = i + 1;
      break;
    }
  }
}

ScanResult Engine::scan(int k, const ScanRequest& rq, i64 begin, i64 end) {
  // Per-k GPU cutover: per-combination cost grows steeply with k (a 7-LUT
  // feasibility check walks 128 cells vs 32 for 5-LUT and 8 for 3-LUT), so
  // the range size where the GPU (incl. ~20-40us launch+upload overhead)
  // beats the CPU differs by ~100x across kinds. Measured on the AES bit-0
  // search: a size-only 2^16 threshold left 2.1e9 7-LUT combos on the CPU
  // and tripled the wall time. k=4 scans go through the persistent scan
  // service (~5-8 us/call vs the ~30 us one-shot floor), which moves the
  // cutover well down.
  // k=4 via the scan service: adaptive cutover. The service round trip is
  // roughly constant (~SVC_US), while the CPU path's per-candidate cost
  // varies ~10x with mask sparsity; route to the GPU when the estimated
  // CPU time for the range exceeds the round trip. SBOXGATES_GPU_MIN4
  // pins a fixed threshold instead.
  static const i64 min4_fixed = [] {
    const char* s = std::getenv("SBOXGATES_GPU_MIN4");
    return s != nullptr ? std::strtoll(s, nullptr, 10) : -1;
  }();
  i64 min4 = min4_fixed;
  if (min4 < 0) {
    constexpr double SVC_US = 12e-6;
    const double rate = stats_.scan_seconds3_cpu > 1e-4
                            ? stats_.candidates3_cpu / stats_.scan_seconds3_cpu
                            : 2e8;
    min4 = std::clamp<i64>(static_cast<i64>(rate * SVC_US), 1024, 1 << 16);
  }
  const i64 gpu_min =
      k == 7 ? 256
             : (k == 5 ? 4096
                       : (k == 4 && gpu_ != nullptr &&
                                  gpu_->scan4_service_active()
                              ? min4
                              : 1 << 14));
  bool use_gpu = gpu_ != nullptr && (opt_.gpu == GPU_FORCE || end - begin >= gpu_min);
  const auto t0 = std::chrono::steady_clock::now();
  ScanResult r;
  if (use_gpu) {
    stats_.gpu_scans += 1;
    r = gpu_->scan(k, rq, begin, end);
  } else {
    stats_.cpu_scans += 1;
    switch (k) {
      case 3: r = cpu_scan3(rq, begin, end); break;
      case 4: r = cpu_scan4(rq, begin, end); break;
      case 5: r = cpu_scan5(rq, begin, end); break;
      case 7: r = cpu_scan7(rq, begin, end); break;
      default: throw std::runtime_error("bad scan k");
    }
  }
  const double dt =
      std::chrono::duration<double>(std::chrono::steady_clock::now() - t0).count();
  if (!use_gpu && k <= 4) {
    stats_.candidates3_cpu += r.evaluated;
    stats_.scan_seconds3_cpu += dt;
  }
  switch (k) {
    case 3:
    case 4:
      stats_.candidates3 += r.evaluated;
      stats_.scan_seconds3 += dt;
      break;
    case 5:
      stats_.candidates5 += r.evaluated;
      stats_.scan_seconds5 += dt;
      break;
    default:
      stats_.candidates7 += r.evaluated;
      stats_.scan_seconds7 += dt;
      break;
  }
  if (opt_.verbosity >= 3) {
    std::printf("[%4d] scan%d: %lld candidates in %.3fs (%.3g cand/s)%s\n",
                ctx_->rank(), k, static_cast<long long>(r.evaluated), dt,
                dt > 0 ? r.evaluated / dt : 0.0, r.found ? " HIT" : "");
  }
  return r;
}

// Chunked symmetric scan with cross-rank agreement each chunk. Every rank
// executes the same number of allreduce rounds (computed from the global
// combination count), so the protocol cannot deadlock — the redesign of the
// reference's Isend/Irecv/cancel protocol (lut.c:665-740) for RCCL.
bool Engine::dist_scan_chunked(int k, const ScanRequest& rq, u64 chunk, u16 res[10]) {
  const i64 total = n_choose_k(rq.n, k);
  const int W = ctx_->world();
  const int r = ctx_->rank();
  const i64 start = total * r / W;
  const i64 stop = total * (r + 1) / W;
  const i64 maxlen = (total + W - 1) / W;
  const i64 nchunks = chunk == 0 ? 1 : (maxlen + static_cast<i64>(chunk) - 1) /
                                           static_cast<i64>(chunk);

  bool found_local = false;
  i64 pos = start;
  for (i64 c = 0; c < nchunks; c++) {
    if (!found_local && pos < stop) {
      i64 e = chunk == 0 ? stop : std::min(stop, pos + static_cast<i64>(chunk));
      ScanResult rr = scan(k, rq, pos, e);
      pos = e;
      if (rr.found) {
        found_local = true;
        std::memcpy(res, rr.res, sizeof(u16) * 10);
      }
    }
    if (W > 1) {
      int winner = ctx_->allreduce_min(found_local ? r : INT_MAX);
      if (winner != INT_MAX) {
        ctx_->bcast(res, sizeof(u16) * 10, winner);
        return true;
      }
    } else if (found_local) {
      return true;
    }
  }
  return false;
}

bool Engine::distributed_lut_body(const WorkMsg& w, u16 res[10], bool* found5) {
  // Build the pool view. The tables live inside the (broadcast) state.
  static thread_local std::vector<ttable> pool;
  pool.resize(w.st.num_gates);
  for (int i = 0; i < w.st.num_gates; i++) pool[i] = w.st.gates[i].table;

  ScanRequest rq;
  rq.tables = pool.data();
  rq.n = w.st.num_gates;
  rq.target = w.target;
  rq.mask = w.mask;
  rq.seed = w.seed;
  rq.count_all = false;
  rq.excl_low64 = 0;
  for (int i = 0; i < 8 && w.inbits[i] != -1; i++) {
    if (w.inbits[i] < 64) rq.excl_low64 |= 1ULL << w.inbits[i];
  }

  *found5 = false;
  // 5/7-LUT chunk sizes: fixed constants (env-overridable for tests) so
  // every rank derives the same chunk count from the broadcast state.
  static const u64 CHUNK5 = [] {
    const char* s = std::getenv("SBOXGATES_CHUNK5");
    return s != nullptr ? std::strtoull(s, nullptr, 10) : (1ULL << 30);
  }();
  static const u64 CHUNK7 = [] {
    const char* s = std::getenv("SBOXGATES_CHUNK7");
    return s != nullptr ? std::strtoull(s, nullptr, 10) : (1ULL << 30);
  }();
  if (w.st.num_gates >= 5) {
    if (w.verbosity >= 2 && ctx_->rank() == 0) std::printf("[   0] Search 5.\n");
    if (dist_scan_chunked(5, rq, CHUNK5, res)) {
      *found5 = true;
      return true;
    }
  }
  // 7-LUT phase; the gate check is symmetric (same broadcast state).
  if (!check_num_gates_possible(&w.st, 3, 0, METRIC_GATES)) return false;
  if (w.st.num_gates >= 7) {
    if (w.verbosity >= 2 && ctx_->rank() == 0) std::printf("[   0] Search 7.\n");
    return dist_scan_chunked(7, rq, CHUNK7, res);
  }
  return false;
}

void Engine::worker_loop() {
  // Parity: sboxgates.c:618-642 — block on broadcast work until quit.
  for (;;) {
    WorkMsg w;
    ctx_->bcast(&w, sizeof(WorkMsg), 0);
    if (w.kind == WORK_QUIT) return;
    u16 res[10];
    bool found5;
    distributed_lut_body(w, res, &found5);
  }
}

void Engine::stop_workers() {
  if (ctx_->world() <= 1 || ctx_->rank() != 0) return;
  WorkMsg w;
  std::memset(&w, 0, sizeof(w));
  w.kind = WORK_QUIT;
  ctx_->bcast(&w, sizeof(WorkMsg), 0);
}

gatenum Engine::lut_search(state* st, const ttable& target, const ttable& mask,
                           const i8* inbits, const gatenum* gate_order) {
  // --- 3-LUT scan (local to rank 0; parity: lut.c:501-523). ---
  {
    static thread_local std::vector<ttable> pool;
    pool.resize(st->num_gates);
    for (int i = 0; i < st->num_gates; i++) pool[i] = st->gates[i].table;
    ScanRequest rq;
    rq.tables = pool.data();
    rq.n = st->num_gates;
    rq.target = target;
    rq.mask = mask;
    rq.seed = rng_.next();
    rq.count_all = false;
    rq.excl_low64 = 0;  // reference parity: 3-LUT scan ignores inbits
    ScanResult r = scan(3, rq, 0, n_choose_k(rq.n, 3));
    if (r.found) {
      const ttable& ta = st->gates[r.res[1]].table;
      const ttable& tb = st->gates[r.res[2]].table;
      const ttable& tc = st->gates[r.res[3]].table;
      ttable nt = gen_lut_ttable(static_cast<u8>(r.res[0]), ta, tb, tc);
      return assert_ret(
          add_lut(st, static_cast<u8>(r.res[0]), nt, r.res[1], r.res[2], r.res[3]),
          target, st, mask, "lut_search/3");
    }
    (void)gate_order;
  }

  if (!check_num_gates_possible(st, 2, 0, opt_.metric)) return NO_GATE;

  // --- Distributed 5/7-LUT search (parity: lut.c:525-631). ---
  WorkMsg w;
  std::memset(&w, 0, sizeof(w));
  w.kind = WORK_LUT_SEARCH;
  w.verbosity = opt_.verbosity;
  w.seed = rng_.next();
  w.target = target;
  w.mask = mask;
  std::memcpy(w.inbits, inbits, 8);
  w.st = *st;
  if (ctx_->world() > 1) ctx_->bcast(&w, sizeof(WorkMsg), 0);

  u16 res[10];
  bool found5 = false;
  bool found = distributed_lut_body(w, res, &found5);
  if (found && found5) {
    u8 fo = static_cast<u8>(res[0]);
    u8 fi = static_cast<u8>(res[1]);
    const ttable ta = st->gates[res[2]].table;
    const ttable tb = st->gates[res[3]].table;
    const ttable tc = st->gates[res[4]].table;
    const ttable td = st->gates[res[5]].table;
    const ttable te = st->gates[res[6]].table;
    if (opt_.verbosity >= 1) {
      std::printf("[%4d]   Selected 5LUT: %02x %02x    %3d %3d %3d %3d %3d\n",
                  ctx_->rank(), fo, fi, res[2], res[3], res[4], res[5], res[6]);
    }
    ttable t_outer = gen_lut_ttable(fo, ta, tb, tc);
    ttable t_inner = gen_lut_ttable(fi, t_outer, td, te);
    return assert_ret(
        add_lut(st, fi, t_inner, add_lut(st, fo, t_outer, res[2], res[3], res[4]),
                res[5], res[6]),
        target, st, mask, "lut_search/5");
  }
  if (found) {
    u8 fo = static_cast<u8>(res[0]);
    u8 fm = static_cast<u8>(res[1]);
    u8 fi = static_cast<u8>(res[2]);
    if (opt_.verbosity >= 1) {
      std::printf("[%4d]   Selected 7LUT: %02x %02x %02x %3d %3d %3d %3d %3d %3d %3d\n",
                  ctx_->rank(), fo, fm, fi, res[3], res[4], res[5], res[6], res[7],
                  res[8], res[9]);
    }
    ttable t_outer = gen_lut_ttable(fo, st->gates[res[3]].table,
                                    st->gates[res[4]].table, st->gates[res[5]].table);
    ttable t_middle = gen_lut_ttable(fm, st->gates[res[6]].table,
                                     st->gates[res[7]].table, st->gates[res[8]].table);
    ttable t_inner = gen_lut_ttable(fi, t_outer, t_middle, st->gates[res[9]].table);
    return assert_ret(
        add_lut(st, fi, t_inner, add_lut(st, fo, t_outer, res[3], res[4], res[5]),
                add_lut(st, fm, t_middle, res[6], res[7], res[8]), res[9]),
        target, st, mask, "lut_search/7");
  }
  if (opt_.verbosity >= 2) {
    std::printf("[%4d] No LUTs found. Num gates: %d\n", ctx_->rank(),
                st->num_gates - get_num_inputs(st));
  }
  return NO_GATE;
}

gatenum Engine::create_circuit(state* st, const ttable& target, const ttable& mask,
                               const i8* inbits) {
  // Randomized gate visit order (parity: sboxgates.c:285-299).
  gatenum gate_order[MAX_GATES];
  for (int i = 0; i < st->num_gates; i++) {
    gate_order[i] = static_cast<gatenum>(st->num_gates - 1 - i);
  }
  if (opt_.randomize) {
    for (u32 i = st->num_gates - 1; i > 0; i--) {
      u64 j = rng_.below(i + 1);
      gatenum t = gate_order[i];
      gate_order[i] = gate_order[j];
      gate_order[j] = t;
    }
  }

  const ttable T1 = target & mask;
  const ttable T0 = ~target & mask;
  stats_.nodes += 1;
  const auto t_s12 = std::chrono::steady_clock::now();

  // Step 1: an existing gate already realizes the map (sboxgates.c:301-308).
  for (int i = 0; i < st->num_gates; i++) {
    if (tt_eq_mask(target, st->gates[gate_order[i]].table, mask)) {
      return assert_ret(gate_order[i], target, st, mask, "step1");
    }
  }

  // Step 2: an inverse realizes the map -> append NOT (sboxgates.c:310-321).
  if (!check_num_gates_possible(st, 1, sat_metric_of(NOT), opt_.metric)) {
    return NO_GATE;
  }
  for (int i = 0; i < st->num_gates; i++) {
    if (tt_eq_mask(target, ~st->gates[gate_order[i]].table, mask)) {
      return assert_ret(add_not_gate(st, gate_order[i], opt_.metric), target, st,
                        mask, "step2");
    }
  }
  stats_.step12_seconds +=
      std::chrono::duration<double>(std::chrono::steady_clock::now() - t_s12)
          .count();

  // Step 3: a pair combined by one available gate (sboxgates.c:323-350).
  // Implemented via 4-cell forced-bit requirements instead of per-function
  // truth-table evaluation; masked equality (see header note).
  if (!check_num_gates_possible(st, 1, sat_metric_of(AND), opt_.metric)) {
    return NO_GATE;
  }
  struct PhaseTimer {
    double* acc;
    std::chrono::steady_clock::time_point t0;
    explicit PhaseTimer(double* a) : acc(a), t0(std::chrono::steady_clock::now()) {}
    ~PhaseTimer() {
      *acc += std::chrono::duration<double>(std::chrono::steady_clock::now() - t0)
                  .count();
    }
  };
  static thread_local std::vector<GateHalves> halves;
  halves.resize(st->num_gates);
  fill_halves(halves.data(), st, st->num_gates, T1, T0);
  {
  PhaseTimer pt3(&stats_.step3_seconds);
  for (int i = 0; i < st->num_gates; i++) {
    const gatenum gi = gate_order[i];
    for (int k = i + 1; k < st->num_gates; k++) {
      const gatenum gk = gate_order[k];
      u8 req1, care;
      if (!pair_requirements(halves[gi], halves[gk], &req1, &care)) continue;
      const u8 req1s = swap_pair_patterns(req1);
      const u8 cares = swap_pair_patterns(care);
      for (int m = 0; opt_.avail_gates[m].num_inputs != 0; m++) {
        const u8 fpat = fun2_pattern_table(opt_.avail_gates[m].fun);
        if ((fpat & care) == req1) {
          return assert_ret(add_boolfunc_2(st, opt_.avail_gates[m], gi, gk, opt_.metric),
                            target, st, mask, "step3");
        }
        if (!opt_.avail_gates[m].ab_commutative && (fpat & cares) == req1s) {
          return assert_ret(add_boolfunc_2(st, opt_.avail_gates[m], gk, gi, opt_.metric),
                            target, st, mask, "step3");
        }
      }
    }
  }
  }

  if (opt_.lut_graph) {
    gatenum ret = lut_search(st, target, mask, inbits, gate_order);
    if (ret != NO_GATE) return assert_ret(ret, target, st, mask, "lut_search");
  } else {
    // Step 4a: pairs with NOT-augmented functions (sboxgates.c:358-386).
    if (!check_num_gates_possible(st, 2, sat_metric_of(AND) + sat_metric_of(NOT),
                                  opt_.metric)) {
      return NO_GATE;
    }
    // Without -n the NOT-augmented function list is empty: the whole pair
    // sweep would compute requirements for zero candidate functions (the
    // reference pays this too, sboxgates.c:366-386 — measured at 22 s of
    // the AES bit-0 CPU run).
    if (opt_.avail_not[0].num_inputs != 0) {
    PhaseTimer pt4a(&stats_.step4a_seconds);
    for (int i = 0; i < st->num_gates; i++) {
      const gatenum gi = gate_order[i];
      for (int k = i + 1; k < st->num_gates; k++) {
        const gatenum gk = gate_order[k];
        u8 req1, care;
        if (!pair_requirements(halves[gi], halves[gk], &req1, &care)) continue;
        const u8 req1s = swap_pair_patterns(req1);
        const u8 cares = swap_pair_patterns(care);
        for (int m = 0; opt_.avail_not[m].num_inputs != 0; m++) {
          const u8 fpat = fun2_pattern_table(opt_.avail_not[m].fun);
          if ((fpat & care) == req1) {
            return assert_ret(
                add_boolfunc_2(st, opt_.avail_not[m], gi, gk, opt_.metric), target,
                st, mask, "step4a");
          }
          if (!opt_.avail_not[m].ab_commutative && (fpat & cares) == req1s) {
            return assert_ret(
                add_boolfunc_2(st, opt_.avail_not[m], gk, gi, opt_.metric), target,
                st, mask, "step4a");
          }
        }
      }
    }
    }

    // Step 4b: triples realized by an available composed 3-input function
    // (sboxgates.c:388-435) — the gate-mode hot loop, run as a k=4 scan
    // (GPU kernel k_scan4 on large pools): cell-requirement screen + one
    // matcher-bitmap probe per argument order; all 6 orders (improvement
    // over the reference's 4 orders with mis-indexed commutativity gates).
    if (!check_num_gates_possible(st, 3, 2 * sat_metric_of(AND) + sat_metric_of(NOT),
                                  opt_.metric)) {
      return NO_GATE;
    }
    if (opt_.num_avail_3 > 0 && st->num_gates >= 3) {
      static thread_local std::vector<ttable> pool4;
      pool4.resize(st->num_gates);
      for (int i = 0; i < st->num_gates; i++) pool4[i] = st->gates[i].table;
      ScanRequest rq;
      rq.tables = pool4.data();
      rq.n = st->num_gates;
      rq.target = target;
      rq.mask = mask;
      rq.excl_low64 = 0;
      rq.seed = rng_.next();
      rq.count_all = false;
      rq.matcher = matcher3();
      ScanResult r = scan(4, rq, 0, n_choose_k(rq.n, 3));
      if (r.found) {
        const boolfunc& f = opt_.avail_3[r.res[0]];
        const int* sel = TRIPLE_PERMS6[r.res[1]];
        const gatenum ids[3] = {r.res[2], r.res[3], r.res[4]};
        return assert_ret(add_boolfunc_3(st, f, ids[sel[0]], ids[sel[1]],
                                         ids[sel[2]], opt_.metric),
                          target, st, mask, "step4b");
      }
    }
  }

  // Step 5: multiplex on an input bit and recurse on the half-spaces
  // (sboxgates.c:438-607).
  i8 next_inbits[8];
  u8 bitp = 0;
  while (bitp < 6 && inbits[bitp] != -1) {
    next_inbits[bitp] = inbits[bitp];
    bitp += 1;
  }
  next_inbits[bitp] = -1;
  next_inbits[bitp + 1] = -1;

  state best;
  gatenum best_out = NO_GATE;
  best.num_gates = 0;
  best.sat_metric = 0;

  for (int bit = 0; bit < get_num_inputs(st); bit++) {
    bool skip = false;
    for (int i = 0; i < bitp; i++) {
      if (inbits[i] == bit) { skip = true; break; }
    }
    if (skip) continue;
    next_inbits[bitp] = static_cast<i8>(bit);

    const ttable fsel = st->gates[bit].table;  // selection bit
    state nst;
    gatenum nst_out = NO_GATE;

    // Cross-bit branch-and-bound (gate mode): once an earlier bit
    // produced `best`, later bits' candidates are only ever KEPT when
    // strictly smaller (ties keep the first), so bound their sub-searches
    // to strictly fewer gates. Measured: 3.4x fewer recursion nodes /
    // 3.4x wall on AES bit 0 (186 s -> 55 s CPU-only), at a ~0.4-gate
    // mean cost on single-shot runs (the tight bound reorders the greedy
    // exploration) that -i 2 at the higher speed more than recovers.
    // Not applied in LUT mode (its searches are scan-dominated; quality
    // of the headline LUT artifacts stays byte-for-byte reproducible)
    // nor under the SAT metric (its bound check is pre-append, so a
    // tight bound would not be strict). SBOXGATES_NO_PRUNE=1 disables.
    static const bool prune_off = [] {
      const char* e = std::getenv("SBOXGATES_NO_PRUNE");
      return e != nullptr && e[0] != '\0' && e[0] != '0';
    }();
    const bool have_best = best.num_gates != 0;
    const gatenum best_bound =
        !prune_off && have_best && opt_.metric == METRIC_GATES &&
                !opt_.lut_graph && best.num_gates >= 2
            ? static_cast<gatenum>(best.num_gates - 2)
            : MAX_GATES;

    if (opt_.lut_graph) {  // LUT-based multiplexer.
      copy_state(nst, *st);
      if (nst.max_gates > best_bound) nst.max_gates = best_bound;
      nst.max_gates -= 1;  // Room for the multiplexer.
      gatenum fb = create_circuit(&nst, target, mask & ~fsel, next_inbits);
      if (fb == NO_GATE) continue;
      gatenum fc = create_circuit(&nst, target, mask & fsel, next_inbits);
      if (fc == NO_GATE) continue;
      nst.max_gates = st->max_gates;  // restore the caller's bound exactly
                                      // (it was clamped by best_bound)

      if (fb == fc) {
        nst_out = fb;
      } else if (fb == bit) {
        nst_out = add_and_gate(&nst, fb, fc, opt_.metric);
        if (nst_out == NO_GATE) continue;
      } else if (fc == bit) {
        nst_out = add_or_gate(&nst, fb, fc, opt_.metric);
        if (nst_out == NO_GATE) continue;
      } else {
        ttable mux_table = gen_lut_ttable(0xac, nst.gates[bit].table,
                                          nst.gates[fb].table, nst.gates[fc].table);
        nst_out = add_lut(&nst, 0xac, mux_table, static_cast<gatenum>(bit), fb, fc);
        if (nst_out == NO_GATE) continue;
      }
      assert(tt_eq_mask(target, nst.gates[nst_out].table, mask));
    } else {  // Try both AND- and OR-based multiplexers; keep the smaller.
      state nst_and;
      copy_state(nst_and, *st);
      if (nst_and.max_gates > best_bound) nst_and.max_gates = best_bound;
      nst_and.max_gates -= 2;
      nst_and.max_sat_metric -= sat_metric_of(AND) + sat_metric_of(XOR);

      gatenum mux_out_and = NO_GATE;
      gatenum fb = create_circuit(&nst_and, target & ~fsel, mask & ~fsel, next_inbits);
      if (fb != NO_GATE) {
        gatenum fc = create_circuit(&nst_and, nst_and.gates[fb].table ^ target,
                                    mask & fsel, next_inbits);
        nst_and.max_gates = st->max_gates;  // caller's bound (was clamped)
        nst_and.max_sat_metric += sat_metric_of(AND) + sat_metric_of(XOR);
        gatenum andg = add_and_gate(&nst_and, fc, static_cast<gatenum>(bit), opt_.metric);
        mux_out_and = add_xor_gate(&nst_and, fb, andg, opt_.metric);
      }

      state nst_or;
      copy_state(nst_or, *st);
      if (nst_or.max_gates > best_bound) nst_or.max_gates = best_bound;
      if (mux_out_and != NO_GATE) {
        nst_or.max_gates = nst_and.num_gates;
        nst_or.max_sat_metric = nst_and.sat_metric;
      }
      nst_or.max_gates -= 2;
      nst_or.max_sat_metric -= sat_metric_of(OR) + sat_metric_of(XOR);

      gatenum mux_out_or = NO_GATE;
      gatenum fd = create_circuit(&nst_or, ~target & fsel, mask & fsel, next_inbits);
      if (fd != NO_GATE) {
        gatenum fe = create_circuit(&nst_or, nst_or.gates[fd].table ^ target,
                                    mask & ~fsel, next_inbits);
        nst_or.max_gates += 2;
        nst_or.max_sat_metric += sat_metric_of(OR) + sat_metric_of(XOR);
        gatenum org = add_or_gate(&nst_or, fe, static_cast<gatenum>(bit), opt_.metric);
        mux_out_or = add_xor_gate(&nst_or, fd, org, opt_.metric);
        nst_or.max_gates = st->max_gates;
        nst_or.max_sat_metric = st->max_sat_metric;
      }
      if (mux_out_and == NO_GATE && mux_out_or == NO_GATE) continue;

      bool pick_and;
      if (opt_.metric == METRIC_GATES) {
        pick_and = mux_out_or == NO_GATE ||
                   (mux_out_and != NO_GATE && nst_and.num_gates < nst_or.num_gates);
      } else {
        pick_and = mux_out_or == NO_GATE ||
                   (mux_out_and != NO_GATE && nst_and.sat_metric < nst_or.sat_metric);
      }
      if (pick_and) {
        copy_state(nst, nst_and);
        nst_out = mux_out_and;
      } else {
        copy_state(nst, nst_or);
        nst_out = mux_out_or;
      }
    }

    // Keep the best sub-state by the active metric (sboxgates.c:593-606).
    if (opt_.metric == METRIC_GATES) {
      if (best.num_gates == 0 || nst.num_gates < best.num_gates) {
        copy_state(best, nst);
        best_out = nst_out;
      }
    } else {
      if (best.sat_metric == 0 || nst.sat_metric < best.sat_metric) {
        copy_state(best, nst);
        best_out = nst_out;
      }
    }
  }

  if (best.num_gates == 0) return NO_GATE;
  copy_state(*st, best);
  return assert_ret(best_out, target, st, mask, "step5");
}

void Engine::save_checkpoint(const state& st) {
  if (!opt_.save_states) return;
  std::string path = save_state(st, opt_.output_dir);
  if (!path.empty()) saved_files_.push_back(path);
}

void Engine::generate_graph_one_output(const state& st_in) {
  if (opt_.jobs > 1 && ctx_->world() == 1) {
    generate_graph_one_output_jobs(st_in);
    return;
  }
  // Parity: sboxgates.c:661-688.
  assert(opt_.iterations > 0);
  state st = st_in;
  if (opt_.verbosity >= 0) {
    std::printf("Generating graphs for output %d...\n", opt_.oneoutput);
  }
  for (int iter = 0; iter < opt_.iterations; iter++) {
    state nst = st;
    i8 bits[8] = {-1, -1, -1, -1, -1, -1, -1, -1};
    const ttable mask = tt_mask_for_inputs(get_num_inputs(&st));
    nst.outputs[opt_.oneoutput] =
        create_circuit(&nst, g_target_[opt_.oneoutput], mask, bits);
    if (nst.outputs[opt_.oneoutput] == NO_GATE) {
      if (opt_.verbosity >= 0) {
        std::printf("(%d/%d): Not found.\n", iter + 1, opt_.iterations);
      }
      continue;
    }
    if (opt_.verbosity >= 0) {
      std::printf("(%d/%d): %d gates. SAT metric: %d\n", iter + 1, opt_.iterations,
                  nst.num_gates - get_num_inputs(&nst), nst.sat_metric);
    }
    save_checkpoint(nst);
    if (opt_.metric == METRIC_GATES) {
      if (nst.num_gates < st.max_gates) st.max_gates = nst.num_gates;
    } else {
      if (nst.sat_metric < st.max_sat_metric) st.max_sat_metric = nst.sat_metric;
    }
  }
}

// Parallel independent iterations: batches of `jobs` threads, each with
// its own engine (rotating over visible GPUs when present); search bounds
// tighten between batches. Semantics: the same iteration count as the
// serial driver, with bound tightening at batch granularity instead of
// per-iteration — every produced circuit is identical in kind and
// checkpointed the same way.
void Engine::generate_graph_one_output_jobs(const state& st_in) {
  assert(opt_.iterations > 0);
  state st = st_in;
  if (opt_.verbosity >= 0) {
    std::printf("Generating graphs for output %d (%d parallel jobs)...\n",
                opt_.oneoutput, opt_.jobs);
  }
  const int devices = gpu_ != nullptr ? std::max(1, gpu_count()) : 0;
  int done = 0;
  while (done < opt_.iterations) {
    const int batch = std::min(opt_.jobs, opt_.iterations - done);
    std::vector<state> results(batch);
    std::vector<char> found(batch, 0);
    std::vector<std::thread> threads;
    for (int j = 0; j < batch; j++) {
      threads.emplace_back([&, j] {
        options wopt = opt_;
        wopt.jobs = 1;
        wopt.verbosity = -1;
        if (wopt.seeded) wopt.seed = opt_.seed + 0x9E37 * (done + j + 1);
        if (devices > 0) wopt.gpu_device = j % devices;
        try {
          Engine we(wopt);
          we.set_sbox(sbox_, num_inputs_);
          state nst = st;
          i8 bits[8] = {-1, -1, -1, -1, -1, -1, -1, -1};
          const ttable mask = tt_mask_for_inputs(get_num_inputs(&st));
          gatenum out = we.create_circuit(&nst, g_target_[opt_.oneoutput], mask,
                                          bits);
          if (out != NO_GATE) {
            nst.outputs[opt_.oneoutput] = out;
            results[j] = nst;
            found[j] = 1;
          }
        } catch (const std::exception& e) {
          std::fprintf(stderr, "search job %d failed: %s\n", done + j, e.what());
        }
      });
    }
    for (auto& t : threads) t.join();
    for (int j = 0; j < batch; j++) {
      if (!found[j]) {
        if (opt_.verbosity >= 0) {
          std::printf("(%d/%d): Not found.\n", done + j + 1, opt_.iterations);
        }
        continue;
      }
      const state& nst = results[j];
      if (opt_.verbosity >= 0) {
        std::printf("(%d/%d): %d gates. SAT metric: %d\n", done + j + 1,
                    opt_.iterations, nst.num_gates - get_num_inputs(&nst),
                    nst.sat_metric);
      }
      save_checkpoint(nst);
      if (opt_.metric == METRIC_GATES) {
        if (nst.num_gates < st.max_gates) st.max_gates = nst.num_gates;
      } else {
        if (nst.sat_metric < st.max_sat_metric) st.max_sat_metric = nst.sat_metric;
      }
    }
    done += batch;
  }
}

static int count_state_outputs(const state& st) {
  int n = 0;
  for (int i = 0; i < 8; i++) {
    if (st.outputs[i] != NO_GATE) n += 1;
  }
  return n;
}

void Engine::generate_graph(const state& st_in) {
  // Multi-output beam search, keeping up to `beam` tied-minimum start
  // states per added output (parity: sboxgates.c:701-788). With
  // opt.jobs > 1 (single-process), the (start state x missing output)
  // searches of one iteration run as parallel independent jobs (engines
  // rotate over visible GPUs); bounds fold between iterations instead of
  // between tasks — slightly weaker mid-iteration pruning, same results
  // semantics.
  int num_start_states = 1;
  state start_states[20];
  start_states[0] = st_in;

  const bool parallel = opt_.jobs > 1 && ctx_->world() == 1;
  const int devices = gpu_ != nullptr ? std::max(1, gpu_count()) : 0;

  int num_outputs;
  while ((num_outputs = count_state_outputs(start_states[0])) < num_outputs_) {
    gatenum max_gates = MAX_GATES;
    int max_sat_metric = INT_MAX;
    state out_states[20];
    int num_out_states = 0;

    auto consider = [&](const state& st) {
      // Fold a successful search result into the beam.
      if (opt_.metric == METRIC_GATES) {
        if (max_gates > st.num_gates) {
          max_gates = st.num_gates;
          num_out_states = 0;
        }
        if (st.num_gates <= max_gates) {
          if (num_out_states < std::min(20, opt_.beam)) {
            out_states[num_out_states++] = st;
          } else if (opt_.verbosity >= 0) {
            std::printf("Output state buffer full! Throwing away valid state.\n");
          }
        }
      } else {
        if (max_sat_metric > st.sat_metric) {
          max_sat_metric = st.sat_metric;
          num_out_states = 0;
        }
        if (st.sat_metric <= max_sat_metric) {
          if (num_out_states < std::min(20, opt_.beam)) {
            out_states[num_out_states++] = st;
          } else if (opt_.verbosity >= 0) {
            std::printf("Output state buffer full! Throwing away valid state.\n");
          }
        }
      }
    };

    for (int iter = 0; iter < opt_.iterations; iter++) {
      if (opt_.verbosity >= 0) {
        std::printf("Generating circuits with %d output%s. (%d/%d)\n", num_outputs + 1,
                    num_outputs == 0 ? "" : "s", iter + 1, opt_.iterations);
      }

      // Collect this iteration's tasks.
      struct Task {
        int current;
        u8 output;
      };
      std::vector<Task> tasks;
      for (int current = 0; current < num_start_states; current++) {
        for (u8 output = 0; output < num_outputs_; output++) {
          if (start_states[current].outputs[output] != NO_GATE) continue;
          tasks.push_back({current, output});
        }
      }

      if (!parallel) {
        for (const Task& task : tasks) {
          if (opt_.verbosity >= 0) {
            std::printf("Generating circuit for output %d...\n", task.output);
          }
          i8 bits[8] = {-1, -1, -1, -1, -1, -1, -1, -1};
          state st = start_states[task.current];
          if (opt_.metric == METRIC_GATES) {
            st.max_gates = max_gates;
          } else {
            st.max_sat_metric = max_sat_metric;
          }
          const ttable mask = tt_mask_for_inputs(get_num_inputs(&st));
          st.outputs[task.output] = create_circuit(&st, g_target_[task.output], mask, bits);
          if (st.outputs[task.output] == NO_GATE) {
            if (opt_.verbosity >= 0) {
              std::printf("No solution for output %d.\n", task.output);
            }
            continue;
          }
          assert(tt_eq_mask(g_target_[task.output],
                            st.gates[st.outputs[task.output]].table, mask));
          save_checkpoint(st);
          consider(st);
        }
      } else {
        // Bounds for every task in this round are fixed at round start.
        const gatenum round_max_gates = max_gates;
        const int round_max_sat = max_sat_metric;
        std::vector<state> results(tasks.size());
        std::vector<char> ok(tasks.size(), 0);
        size_t next = 0;
        while (next < tasks.size()) {
          const size_t batch = std::min<size_t>(opt_.jobs, tasks.size() - next);
          std::vector<std::thread> threads;
          for (size_t j = 0; j < batch; j++) {
            const size_t ti = next + j;
            threads.emplace_back([&, ti, j] {
              const Task& task = tasks[ti];
              options wopt = opt_;
              wopt.jobs = 1;
              wopt.verbosity = -1;
              if (wopt.seeded) {
                wopt.seed = opt_.seed + 0x51ED * (iter * 1024 + static_cast<int>(ti) + 1);
              }
              if (devices > 0) wopt.gpu_device = static_cast<int>(j) % devices;
              try {
                Engine we(wopt);
                we.set_sbox(sbox_, num_inputs_);
                state st = start_states[task.current];
                if (opt_.metric == METRIC_GATES) {
                  st.max_gates = round_max_gates;
                } else {
                  st.max_sat_metric = round_max_sat;
                }
                i8 bits[8] = {-1, -1, -1, -1, -1, -1, -1, -1};
                const ttable mask = tt_mask_for_inputs(get_num_inputs(&st));
                st.outputs[task.output] =
                    we.create_circuit(&st, g_target_[task.output], mask, bits);
                if (st.outputs[task.output] != NO_GATE) {
                  results[ti] = st;
                  ok[ti] = 1;
                }
              } catch (const std::exception& e) {
                std::fprintf(stderr, "beam job (state %d, output %d) failed: %s\n",
                             task.current, task.output, e.what());
              }
            });
          }
          for (auto& t : threads) t.join();
          next += batch;
        }
        for (size_t ti = 0; ti < tasks.size(); ti++) {
          if (!ok[ti]) {
            if (opt_.verbosity >= 0) {
              std::printf("No solution for output %d.\n", tasks[ti].output);
            }
            continue;
          }
          save_checkpoint(results[ti]);
          consider(results[ti]);
        }
      }
    }
    if (num_out_states == 0) {
      // No output could be added within bounds; stop rather than loop
      // forever (the reference would loop with an empty beam).
      if (opt_.verbosity >= 0) std::printf("No solution found.\n");
      return;
    }
    if (opt_.verbosity >= 0) {
      if (opt_.metric == METRIC_GATES) {
        std::printf("Found %d state%s with %d gates.\n", num_out_states,
                    num_out_states == 1 ? "" : "s",
                    max_gates - get_num_inputs(&out_states[0]));
      } else {
        std::printf("Found %d state%s with SAT metric %d.\n", num_out_states,
                    num_out_states == 1 ? "" : "s", max_sat_metric);
      }
    }
    for (int i = 0; i < num_out_states; i++) start_states[i] = out_states[i];
    num_start_states = num_out_states;
  }
}

}  // namespace sbg
