// options.hpp — user-facing search configuration (flag-for-flag parity with
// the reference options struct, sboxgates.h:49-66, plus MI355X additions:
// seed control, GPU selection, output directory).
#pragma once

#include <string>

#include "sbg/boolfunc.hpp"
#include "sbg/state.hpp"

namespace sbg {

enum gpu_mode_t : i32 {
  GPU_AUTO = 0,   // use a GPU when one is visible
  GPU_OFF = 1,
  GPU_FORCE = 2,  // fail loudly if no GPU / kernels unavailable
};

struct options {
  std::string fname;        // Input file (S-box table, or XML for -c/-d).
  std::string gfname;       // -g: initial graph file.
  int iterations = 1;       // -i
  int oneoutput = -1;       // -o
  int permute = 0;          // -p
  metric_t metric = METRIC_GATES;  // -s selects SAT
  bool output_c = false;    // -c
  bool output_dot = false;  // -d
  bool output_hip = false;  // --convert-hip (new: HIP device-function codegen)
  bool lut_graph = false;   // -l
  bool randomize = true;    // always on, as in the reference (sboxgates.c:1070)
  bool try_nots = false;    // -n
  int verbosity = 0;        // -v (counted)

  boolfunc avail_gates[17]; // terminated by num_inputs == 0
  boolfunc avail_not[49];
  boolfunc avail_3[257];
  int num_avail_3 = 0;

  // MI355X-native additions.
  bool seeded = false;      // --seed given: deterministic RNG
  u64 seed = 0;
  gpu_mode_t gpu = GPU_AUTO;
  int gpu_device = -1;      // HIP device index; -1 = current device
  int num_gpus = 1;         // CLI --gpus: in-process devices (threads)
  int beam = 20;            // --beam: tied-state beam width (<= 20; the
                            // reference hard-codes 20, sboxgates.c:704)
  int jobs = 1;             // --jobs: parallel independent search
                            // iterations (one engine per job; jobs rotate
                            // over visible GPUs)
  std::string output_dir;   // where XML checkpoints are written ("" = CWD)
  bool save_states = true;  // library callers may disable checkpoint writes

  // Fills avail_gates from a 16-bit gate-set bitfield (bit i = 2-input
  // function i available). Parity: sboxgates.c:870-880.
  void set_avail_gates(u32 bitfield) {
    int gatep = 0;
    for (int i = 0; i < 16; i++) {
      if (bitfield & (1u << i)) avail_gates[gatep++] = make_2_input_fun(static_cast<u8>(i));
    }
    avail_gates[gatep].num_inputs = 0;
  }

  // Derives avail_not / avail_3 from avail_gates. Call after set_avail_gates
  // and after try_nots is final (parity: sboxgates.c:974-981).
  void derive_function_lists() {
    int num = 0;
    if (try_nots) num = get_not_functions(avail_gates, avail_not);
    avail_not[num].num_inputs = 0;
    num_avail_3 = get_3_input_function_list(avail_gates, avail_3, try_nots);
    avail_3[num_avail_3].num_inputs = 0;
  }
};

// Default gate set: AND + OR + XOR (bitfield 194; sboxgates.c:1078).
constexpr u32 DEFAULT_GATE_BITFIELD = 2 + 64 + 128;

}  // namespace sbg
