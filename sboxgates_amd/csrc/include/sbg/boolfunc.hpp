// boolfunc.hpp — the search's gate vocabulary: 2-input Boolean functions,
// NOT-augmented derived functions, and the closure of 3-input functions
// reachable as fun2(fun1(A,B),C) with optional input/output NOTs.
//
// Behavioral parity: boolfunc.c/boolfunc.h in the reference. The 4-bit
// 2-input function encoding IS the gate_type enum value (AND=0b0001 means
// "output 1 only at A=1,B=1"), and the 8-bit 3-input function bit p is the
// output at input pattern p = A<<2 | B<<1 | C — the same conventions the
// gates.xsd "function" attribute and generate_lut_ttable use.
#pragma once

#include "sbg/common.hpp"
#include "sbg/ttable.hpp"

namespace sbg {

// Gate types. Numeric values 0..15 are the canonical 2-input truth tables;
// values must match the reference enum (state.h:37-57) for XML-name order
// and function-bit semantics.
enum gate_type : i32 {
  FALSE_GATE = 0,
  AND = 1,
  A_AND_NOT_B = 2,
  A = 3,
  NOT_A_AND_B = 4,
  B = 5,
  XOR = 6,
  OR = 7,
  NOR = 8,
  XNOR = 9,
  NOT_B = 10,
  A_OR_NOT_B = 11,
  NOT_A = 12,
  NOT_A_OR_B = 13,
  NAND = 14,
  TRUE_GATE = 15,
  NOT = 16,
  IN = 17,
  LUT = 18,
  GATE_END = 0xff
};

// Display/XML names, indexed by gate_type (parity: state.c:33-52, gates.xsd).
extern const char* const gate_name[19];

// Value of 2-input function `fun` at input pattern bit = A<<1 | B.
// fun bit 3 is the A=1,B=1 entry (parity: boolfunc.c:22-25).
SBG_HD inline u8 fun2_val(u8 fun, u8 bit) { return (fun >> (3 - bit)) & 1; }

// Truth table of a 2-input gate applied to two truth tables
// (parity: boolfunc.c:136-157). Implemented directly from the 4-bit
// function encoding instead of a 16-way switch: minterm expansion.
// NOTE the encoding is bit-REVERSED relative to the pattern index:
// fun bit k is the output at pattern A<<1|B = 3-k (so AND=0b0001,
// NOR=0b1000), unlike the 3-input byte where bit p <-> pattern p.
SBG_HD inline ttable gen_ttable_2(int fun, const ttable& a, const ttable& b) {
  ttable r = tt_zero_table();
  if (fun & 1) r |= a & b;     // pattern 3: A=1,B=1
  if (fun & 2) r |= a & ~b;    // pattern 2: A=1,B=0
  if (fun & 4) r |= ~a & b;    // pattern 1: A=0,B=1
  if (fun & 8) r |= ~a & ~b;   // pattern 0: A=0,B=0
  return r;
}

// Truth table of a 3-input LUT: function bit p = output at
// p = in1<<2 | in2<<1 | in3 (parity: state.c:201-230).
SBG_HD inline ttable gen_lut_ttable(u8 function, const ttable& a,
                                    const ttable& b, const ttable& c) {
  ttable r = tt_zero_table();
  for (int p = 0; p < 8; p++) {
    if (function & (1u << p)) {
      ttable cell = (p & 4 ? a : ~a) & (p & 2 ? b : ~b) & (p & 1 ? c : ~c);
      r |= cell;
    }
  }
  return r;
}

// A (possibly composed) Boolean function usable as a search gate.
// For 2-input entries fun == fun1 and fun2 is unused; for 3-input entries
// the realized circuit is fun2(fun1(±A,±B),±C) with optional output NOT.
struct boolfunc {
  int num_inputs = 0;  // 0 marks end-of-list, as in the reference.
  u8 fun = 0;          // Truth table: 4 bits (2-in) or 8 bits (3-in).
  i32 fun1 = GATE_END;
  i32 fun2 = GATE_END;
  bool not_a = false, not_b = false, not_c = false, not_out = false;
  bool ab_commutative = false;
  bool ac_commutative = false;
  bool bc_commutative = false;
};

// Builds the boolfunc for a raw 2-input function (parity: boolfunc.c:56-71).
boolfunc make_2_input_fun(u8 fun);

// Derives functions reachable by complementing the output of the available
// gates, excluding duplicates (parity: boolfunc.c:36-54). input list is
// terminated by num_inputs == 0; returns count written to out.
int get_not_functions(const boolfunc* input_funs, boolfunc* output_funs);

// Enumerates all distinct 3-input functions reachable as
// fun2(fun1(±A,±B),±C) [input/output NOTs only when try_nots], dedupe by
// 8-bit truth table, with pairwise-commutativity flags
// (parity: boolfunc.c:73-134). output_funs must hold >= 256 entries.
int get_3_input_function_list(const boolfunc* input_funs, boolfunc* output_funs,
                              bool try_nots);

// Truth table of a composed 3-input boolfunc (parity: boolfunc.c:159-186).
SBG_HD inline ttable gen_ttable_3(const boolfunc& f, const ttable& a,
                                  const ttable& b, const ttable& c) {
  return gen_lut_ttable(f.fun, a, b, c);
}

}  // namespace sbg
