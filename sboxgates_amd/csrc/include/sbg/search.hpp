// search.hpp — the search engine: Kwan's iterative gate-addition algorithm
// (steps 1-5), the LUT search (3-LUT scan + distributed 5/7-LUT scans),
// and the multi-output beam-search driver.
//
// Behavioral parity: sboxgates.c:280-788 and lut.c:489-631 in the
// reference. Known deliberate improvements over the reference (documented,
// not bugs): masked equality is used in the step-3/4 pair scans where the
// reference requires full equality against target&mask (sboxgates.c:338 —
// which can never match when the mask is partial); step-4 triples try all
// 6 argument orders (the reference tries 4, gated on mis-indexed
// commutativity flags, sboxgates.c:411-425); the 7-LUT search has no
// 100k-combination frontier cap (lut.c:291).
#pragma once

#include <memory>
#include <string>
#include <vector>

#include "sbg/common.hpp"
#include "sbg/dist.hpp"
#include "sbg/options.hpp"
#include "sbg/rng.hpp"
#include "sbg/scan.hpp"
#include "sbg/state.hpp"

namespace sbg {

class GpuEngine;

struct SearchStats {
  u64 candidates3 = 0;
  u64 candidates5 = 0;
  u64 candidates7 = 0;
  u64 gpu_scans = 0;
  u64 cpu_scans = 0;
  double scan_seconds3 = 0;  // wall time inside 3-input scans (incl. k=4)
  double scan_seconds5 = 0;
  double scan_seconds7 = 0;
  // CPU-path-only k=3/4 totals: feed the adaptive GPU cutover (the CPU
  // rate swings ~10x between full and sparse masks, so the size threshold
  // where the scan-service round trip wins moves with it).
  u64 candidates3_cpu = 0;
  double scan_seconds3_cpu = 0;
  // Host-side phase timers for the gate-mode recursion (where the wall
  // that is NOT scan time goes; see profiles/gate_mode_service.md).
  u64 nodes = 0;              // create_circuit invocations
  double step12_seconds = 0;  // step 1/2 existing-gate and inverse sweeps
  double step3_seconds = 0;   // step 3 pair loop (host)
  double step4a_seconds = 0;  // step 4a NOT-augmented pair loop (host)
};

class Engine {
 public:
  explicit Engine(const options& opt, DistCtx* ctx = nullptr);
  ~Engine();

  // Target setup. sbox is a full 256-entry table (tail zeroed for smaller
  // S-boxes); num_inputs in [1,8].
  void set_sbox(const u8 sbox[256], int num_inputs);
  int num_inputs() const { return num_inputs_; }
  int num_outputs() const { return num_outputs_; }
  const ttable& target(int bit) const { return g_target_[bit]; }

  // Fresh initial state over the S-box's inputs.
  void initial_state(state& st) const { init_state(st, num_inputs_); }

  // Search drivers (rank 0). Parity: sboxgates.c:661-688, 701-788.
  // With opt.jobs > 1 (and no distributed ctx), one-output iterations run
  // as parallel independent jobs, each with its own engine/GPU; bounds
  // tighten between batches of `jobs` iterations.
  void generate_graph_one_output(const state& st);
  void generate_graph(const state& st);

  // One recursive circuit construction for an arbitrary target/mask
  // (exposed for tests and fine-grained use).
  gatenum create_circuit(state* st, const ttable& target, const ttable& mask,
                         const i8* inbits);

  // Worker loop for ranks != 0 (parity: sboxgates.c:618-642): blocks on
  // broadcast work until a quit message arrives.
  void worker_loop();
  // Rank 0: release workers (parity: sboxgates.c:790-795).
  void stop_workers();

  // Scan dispatch (used by create_circuit internals, tests, and bench):
  // runs on GPU when available and the range is large enough, else CPU.
  ScanResult scan(int k, const ScanRequest& rq, i64 begin, i64 end);

  bool gpu_active() const;
  // Cached step-4 matcher built from this engine's avail_3 vocabulary.
  const Avail3Matcher* matcher3();
  const SearchStats& stats() const { return stats_; }
  const std::vector<std::string>& saved_files() const { return saved_files_; }
  options& opt() { return opt_; }
  Xorshift1024& rng() { return rng_; }

 private:
  gatenum lut_search(state* st, const ttable& target, const ttable& mask,
                     const i8* inbits, const gatenum* gate_order);
  // Symmetric distributed 5/7 search body executed by every rank.
  bool distributed_lut_body(const WorkMsg& work, u16 res[10], bool* found5);
  bool dist_scan_chunked(int k, const ScanRequest& rq, u64 chunk, u16 res[10]);
  void generate_graph_one_output_jobs(const state& st);
  void save_checkpoint(const state& st);

  options opt_;
  LocalCtx local_;
  DistCtx* ctx_;
  Xorshift1024 rng_;
  u8 sbox_[256] = {};
  int num_inputs_ = 0;
  int num_outputs_ = 0;
  ttable g_target_[8] = {};
  std::unique_ptr<GpuEngine> gpu_;
  std::unique_ptr<Avail3Matcher> matcher_;
  SearchStats stats_;
  std::vector<std::string> saved_files_;
};

// --- Gate-append primitives (parity: sboxgates.c:95-229). Exposed for
// tests and codegen round-trips. All return NO_GATE on bound violations.
gatenum add_gate(state* st, int type, gatenum gid1, gatenum gid2, metric_t metric);
gatenum add_not_gate(state* st, gatenum gid, metric_t metric);
gatenum add_lut(state* st, u8 func, const ttable& table, gatenum g1, gatenum g2,
                gatenum g3);
gatenum add_boolfunc_2(state* st, const boolfunc& fun, gatenum g1, gatenum g2,
                       metric_t metric);
gatenum add_boolfunc_3(state* st, const boolfunc& fun, gatenum g1, gatenum g2,
                       gatenum g3, metric_t metric);

// Evaluates the circuit on one input pattern (0..2^n-1); returns the output
// byte assembled from the state's output gates. Used by tests as the
// ground-truth correctness oracle (the reference only checks truth tables).
u8 eval_circuit(const state& st, u8 input);

}  // namespace sbg
