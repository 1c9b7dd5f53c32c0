// codegen.hpp — circuit-graph conversion to source text.
//
// Behavioral parity: convert_graph.c in the reference (C function for pure
// 2-input graphs, CUDA with LOP3.LUT inline asm when LUT gates are present,
// Graphviz DOT). New in this implementation: a HIP/CDNA backend emitting a
// bitsliced __device__ function (gfx950 has no LOP3 analog; LUTs lower to
// minterm expressions the compiler folds to v_bfi/v_xor3 sequences), and
// the reference's output-slot iteration bug (convert_graph.c:121,164 bounds
// the output loop by the input count) is fixed: all 8 slots are scanned.
#pragma once

#include <string>

#include "sbg/state.hpp"

namespace sbg {

enum codegen_lang : i32 {
  LANG_AUTO = 0,  // C, or CUDA if the graph contains LUT gates (reference rule)
  LANG_C = 1,
  LANG_CUDA = 2,
  LANG_HIP = 3,
};

// 256-character '0'/'1' dump of a truth table, 16 per line (debug helper;
// parity: convert_graph.c:28-46).
std::string ttable_to_string(const ttable& t);

// Graphviz DOT digraph (parity: convert_graph.c:48-85).
std::string graph_to_dot(const state& st);

// C / CUDA / HIP source for the circuit. Returns empty string and sets
// *err if the state has no outputs.
std::string graph_to_source(const state& st, codegen_lang lang, std::string* err);

}  // namespace sbg
