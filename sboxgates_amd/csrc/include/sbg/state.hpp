// state.hpp — circuit state: fixed-capacity gate DAG with per-gate cached
// truth tables, 8 output slots, and branch-and-bound bounds for both
// metrics.
//
// Layout parity with the reference (state.h:72-88): member order, 32-byte
// ttable alignment and zero-padding are preserved so that the Speck-based
// state fingerprint (state.c:56-105) produces identical file names for
// identical circuits, and so the state can be broadcast between ranks as a
// flat byte blob.
#pragma once

#include <climits>
#include <cstddef>
#include <cstring>
#include <string>

#include "sbg/boolfunc.hpp"
#include "sbg/common.hpp"
#include "sbg/ttable.hpp"

namespace sbg {

enum metric_t : i32 { METRIC_GATES = 0, METRIC_SAT = 1 };

struct gate {
  ttable table;     // Cached truth table of this gate.
  i32 type;         // gate_type.
  gatenum in1;      // NO_GATE for inputs.
  gatenum in2;      // NO_GATE for NOT gates and inputs.
  gatenum in3;      // Input 3 if LUT, else NO_GATE.
  u8 function;      // For LUTs: the 8-bit lookup table.
};
static_assert(sizeof(gate) == 64, "gate layout must match the reference");

struct state {
  i32 max_sat_metric;    // Current maximum accepted SAT metric.
  i32 sat_metric;        // SAT metric of the current state.
  gatenum max_gates;     // Current maximum accepted number of gates.
  gatenum num_gates;     // Current number of gates.
  gatenum outputs[8];    // Gate number of each output, or NO_GATE.
  gate gates[MAX_GATES];
};
static_assert(sizeof(state) == 32 + 64 * MAX_GATES,
              "state layout must match the reference");

// Copies only the live prefix (header + gates[0..num_gates)) — the struct
// is 32 KB but live circuits are usually < 100 gates, and the step-5 mux
// recursion copies states several times per node. Dead slots in dst are
// left stale, which is safe: every consumer (fingerprint, XML writer, DAG
// eval, scans, rank broadcast) interprets only the live prefix.
inline void copy_state(state& dst, const state& src) {
  std::memcpy(&dst, &src,
              offsetof(state, gates) + sizeof(gate) * src.num_gates);
}

// SAT-metric cost model (parity: state.c:168-191). Calling with LUT aborts
// in the reference; here it returns a sentinel the callers treat as invalid.
int sat_metric_of(int type);

// Number of leading IN gates (parity: state.c:193-199).
int get_num_inputs(const state* st);

// Truth table of S-box output bit `bit` (sbox != nullptr), or of raw input
// bit `bit` (sbox == nullptr). Parity: state.c:232-250.
ttable generate_target(u8 bit, const u8* sbox);

// Unique-ish 32-bit fingerprint of a state, used in save file names
// (parity: state.c:56-105 — Speck-round-based hash over the gate array).
u32 state_fingerprint(const state& st);

// The save file name the reference would use:
// O-GGG-MMMM-NNNN...-FFFFFFFF.xml (parity: state.c:107-125, state.h:90-96).
std::string state_file_name(const state& st);

// Initializes st as an empty circuit over num_inputs input gates.
void init_state(state& st, int num_inputs);

// Whether a solution with `add` more gates / `add_sat` more SAT cost is
// still within bounds (parity: sboxgates.c:270-278).
bool check_num_gates_possible(const state* st, int add, int add_sat,
                              metric_t metric);

}  // namespace sbg
