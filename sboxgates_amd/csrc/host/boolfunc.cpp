// boolfunc.cpp — gate-vocabulary construction (host-side, runs once at
// startup). Behavioral parity: boolfunc.c in the reference.

#include "sbg/boolfunc.hpp"

#include <cstring>

namespace sbg {

const char* const gate_name[19] = {
    "FALSE", "AND",   "A_AND_NOT_B", "A",     "NOT_A_AND_B", "B",
    "XOR",   "OR",    "NOR",         "XNOR",  "NOT_B",       "A_OR_NOT_B",
    "NOT_A", "NOT_A_OR_B", "NAND",   "TRUE",  "NOT",         "IN",
    "LUT"};

boolfunc make_2_input_fun(u8 fun) {
  boolfunc ret;
  ret.num_inputs = 2;
  ret.fun = fun;
  ret.fun1 = fun;
  ret.fun2 = GATE_END;
  // Commutative iff the A=1,B=0 and A=0,B=1 entries agree (fun bits 1 and 2
  // in the reversed encoding).
  ret.ab_commutative = ((~(fun >> 1 ^ fun >> 2)) & 1) != 0;
  return ret;
}

static bool fun_in_list(u8 fun, const boolfunc* list) {
  for (int i = 0; list[i].num_inputs != 0; i++) {
    if (list[i].fun == fun) return true;
  }
  return false;
}

int get_not_functions(const boolfunc* input_funs, boolfunc* output_funs) {
  int outp = 0;
  output_funs[0].num_inputs = 0;
  for (int i = 0; input_funs[i].num_inputs != 0; i++) {
    u8 cfun = static_cast<u8>(~input_funs[i].fun & 0xF);
    if (!fun_in_list(cfun, input_funs) && !fun_in_list(cfun, output_funs)) {
      output_funs[outp] = input_funs[i];
      output_funs[outp].fun = cfun;
      output_funs[outp].not_out = !output_funs[outp].not_out;
      outp += 1;
      output_funs[outp].num_inputs = 0;
    }
  }
  return outp;
}

int get_3_input_function_list(const boolfunc* input_funs, boolfunc* output_funs,
                              bool try_nots) {
  boolfunc funs[256];
  for (auto& f : funs) { f.num_inputs = 0; f.fun1 = GATE_END; }
  bool have[256] = {};

  // Input-NOT patterns (bit2=not_a, bit1=not_b, bit0=not_c), ordered so that
  // decompositions with fewer NOT gates are discovered (and kept) first —
  // the same preference order the reference uses (boolfunc.c:80).
  static const u8 nots[8] = {0, 1, 2, 4, 3, 5, 6, 7};

  for (int notsp = 0; notsp < (try_nots ? 8 : 1); notsp++) {
    const u8 nv = nots[notsp];
    for (int i = 0; input_funs[i].num_inputs != 0; i++) {
      for (int k = 0; input_funs[k].num_inputs != 0; k++) {
        // Truth table of fun2_k(fun1_i(A^na, B^nb), C^nc).
        u8 fun = 0;
        for (u8 p = 0; p < 8; p++) {
          u8 a = ((p >> 2) & 1) ^ ((nv >> 2) & 1);
          u8 b = ((p >> 1) & 1) ^ ((nv >> 1) & 1);
          u8 c = (p & 1) ^ (nv & 1);
          u8 inner = fun2_val(input_funs[i].fun, static_cast<u8>(a << 1 | b));
          u8 v = fun2_val(input_funs[k].fun, static_cast<u8>(inner << 1 | c));
          fun |= static_cast<u8>(v << p);
        }
        if (!have[fun]) {
          have[fun] = true;
          boolfunc& f = funs[fun];
          f.num_inputs = 3;
          f.fun = fun;
          f.fun1 = input_funs[i].fun;
          f.fun2 = input_funs[k].fun;
          f.not_a = (nv & 4) != 0;
          f.not_b = (nv & 2) != 0;
          f.not_c = (nv & 1) != 0;
          f.not_out = false;
          f.ab_commutative = ((~(fun >> 2 ^ fun >> 4) & ~(fun >> 3 ^ fun >> 5)) & 1) != 0;
          f.ac_commutative = ((~(fun >> 1 ^ fun >> 4) & ~(fun >> 3 ^ fun >> 6)) & 1) != 0;
          f.bc_commutative = ((~(fun >> 1 ^ fun >> 2) & ~(fun >> 5 ^ fun >> 6)) & 1) != 0;
        }
      }
    }
  }

  // Functions reachable by appending a NOT to an already-found function.
  if (try_nots) {
    for (int i = 0; i < 256; i++) {
      int nfun = ~i & 0xff;
      if (have[i] && !have[nfun]) {
        have[nfun] = true;
        funs[nfun] = funs[i];
        funs[nfun].fun = static_cast<u8>(nfun);
        funs[nfun].not_out = true;
      }
    }
  }

  int outp = 0;
  for (int i = 0; i < 256; i++) {
    if (have[i]) output_funs[outp++] = funs[i];
  }
  if (outp < 256) output_funs[outp].num_inputs = 0;
  return outp;
}

}  // namespace sbg
