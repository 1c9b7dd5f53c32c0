// codegen.cpp — graph → C / CUDA / HIP / DOT text.

#include "sbg/codegen.hpp"

#include <cstdio>
#include <cstring>

namespace sbg {

std::string ttable_to_string(const ttable& t) {
  std::string out;
  out.reserve(256 + 16);
  for (int i = 0; i < 256; i++) {
    if (i != 0 && i % 16 == 0) out += '\n';
    out += tt_get_bit(t, i) ? '1' : '0';
  }
  out += '\n';
  return out;
}

std::string graph_to_dot(const state& st) {
  std::string out = "digraph sbox {\n";
  char buf[96];
  for (int gt = 0; gt < st.num_gates; gt++) {
    char gatename[24];
    if (st.gates[gt].type == IN) {
      std::snprintf(gatename, sizeof(gatename), "IN %d", gt);
    } else if (st.gates[gt].type == LUT) {
      std::snprintf(gatename, sizeof(gatename), "0x%02x", st.gates[gt].function);
    } else {
      std::snprintf(gatename, sizeof(gatename), "%s", gate_name[st.gates[gt].type]);
      for (char* p = gatename; *p != '\0'; p++) {
        if (*p == '_') *p = ' ';
      }
    }
    std::snprintf(buf, sizeof(buf), "  gt%d [label=\"%s\"];\n", gt, gatename);
    out += buf;
  }
  for (int gt = get_num_inputs(&st); gt < st.num_gates; gt++) {
    const gatenum ins[3] = {st.gates[gt].in1, st.gates[gt].in2, st.gates[gt].in3};
    for (gatenum in : ins) {
      if (in != NO_GATE) {
        std::snprintf(buf, sizeof(buf), "  gt%u -> gt%d;\n", in, gt);
        out += buf;
      }
    }
  }
  for (int i = 0; i < 8; i++) {
    if (st.outputs[i] != NO_GATE) {
      std::snprintf(buf, sizeof(buf), "  gt%u -> out%d;\n", st.outputs[i], i);
      out += buf;
    }
  }
  out += "}\n";
  return out;
}

namespace {

// Variable naming (parity: convert_graph.c:93-107): inputs are in.bN,
// output gates are (*)outN, everything else varN. Returns true when the
// variable needs a declaration.
bool variable_name(const state& st, gatenum g, char* buf, size_t bufsz, bool ptr_out) {
  if (g < get_num_inputs(&st)) {
    std::snprintf(buf, bufsz, "in.b%u", g);
    return false;
  }
  for (int i = 0; i < 8; i++) {
    if (st.outputs[i] == g) {
      std::snprintf(buf, bufsz, "%sout%d", ptr_out ? "*" : "", i);
      return false;
    }
  }
  std::snprintf(buf, bufsz, "var%u", g);
  return true;
}

// Bitsliced sum-of-products expression for a LUT byte over variables a,b,c
// (used by the HIP backend, which has no LOP3-style instruction to name).
std::string lut_expr(u8 func, const char* a, const char* b, const char* c) {
  if (func == 0) return "0";
  if (func == 0xff) return "~(bit_t)0";
  std::string out;
  char term[128];
  // Emit complemented-minterm form when it is shorter.
  bool invert = __builtin_popcount(func) > 4;
  u8 f = invert ? static_cast<u8>(~func) : func;
  bool first = true;
  for (int p = 0; p < 8; p++) {
    if (!((f >> p) & 1)) continue;
    std::snprintf(term, sizeof(term), "%s(%s%s & %s%s & %s%s)", first ? "" : " | ",
                  p & 4 ? "" : "~", a, p & 2 ? "" : "~", b, p & 1 ? "" : "~", c);
    out += term;
    first = false;
  }
  if (invert) return "~(" + out + ")";
  return out;
}

}  // namespace

std::string graph_to_source(const state& st, codegen_lang lang, std::string* err) {
  bool has_lut = false;
  for (int g = get_num_inputs(&st); g < st.num_gates; g++) {
    if (st.gates[g].type == LUT) { has_lut = true; break; }
  }
  if (lang == LANG_AUTO) lang = has_lut ? LANG_CUDA : LANG_C;
  if (lang == LANG_C && has_lut) lang = LANG_CUDA;  // reference rule

  int num_outputs = 0;
  int outp_num = 0;
  for (int outp = 0; outp < 8; outp++) {
    if (st.outputs[outp] != NO_GATE) {
      num_outputs += 1;
      outp_num = outp;
    }
  }
  if (num_outputs <= 0) {
    if (err != nullptr) *err = "no output gates in circuit";
    return "";
  }
  const bool ptr_ret = num_outputs > 1;

  std::string out;
  char buf[256];
  const char* TYPE = "bit_t";

  // Type definitions.
  if (lang == LANG_CUDA) {
    out +=
        "#define LUT(a,b,c,d,e) asm(\"lop3.b32 %0, %1, %2, %3, \"#e\";\" : "
        "\"=r\"(a): \"r\"(b), \"r\"(c), \"r\"(d));\n";
    out += "typedef int bit_t;\n";
  } else if (lang == LANG_HIP) {
    out += "typedef unsigned long long int bit_t;\n";
  } else {
    out += "typedef unsigned long long int bit_t;\n";
  }
  out += "typedef struct {\n";
  for (int i = 0; i < get_num_inputs(&st); i++) {
    std::snprintf(buf, sizeof(buf), "  %s b%d;\n", TYPE, i);
    out += buf;
  }
  out += "} bits;\n";

  // Function signature.
  const char* qual = lang == LANG_C ? "" : "__device__ __forceinline__ ";
  if (num_outputs > 1) {
    std::snprintf(buf, sizeof(buf), "%svoid s(bits in", qual);
    out += buf;
    for (int outp = 0; outp < 8; outp++) {
      if (st.outputs[outp] != NO_GATE) {
        std::snprintf(buf, sizeof(buf), ", %s *out%d", TYPE, outp);
        out += buf;
      }
    }
    out += ") {\n";
  } else {
    std::snprintf(buf, sizeof(buf), "%s%s s%d(bits in) {\n", qual, TYPE, outp_num);
    out += buf;
  }

  // Gate statements.
  char v1[16], v2[16], v3[16], vo[16], start[16];
  for (int g = get_num_inputs(&st); g < st.num_gates; g++) {
    const gate& gt = st.gates[g];
    if (gt.in1 != NO_GATE) variable_name(st, gt.in1, v1, sizeof(v1), ptr_ret);
    if (gt.in2 != NO_GATE) variable_name(st, gt.in2, v2, sizeof(v2), ptr_ret);
    if (gt.in3 != NO_GATE) variable_name(st, gt.in3, v3, sizeof(v3), ptr_ret);
    bool decl = variable_name(st, g, vo, sizeof(vo), ptr_ret);
    if (decl || vo[0] != '*') {
      std::snprintf(start, sizeof(start), "  %s ", TYPE);
    } else {
      std::snprintf(start, sizeof(start), "  ");
    }

    switch (gt.type) {
      case FALSE_GATE:  std::snprintf(buf, sizeof(buf), "%s%s = 0;\n", start, vo); break;
      case AND:         std::snprintf(buf, sizeof(buf), "%s%s = %s & %s;\n", start, vo, v1, v2); break;
      case A_AND_NOT_B: std::snprintf(buf, sizeof(buf), "%s%s = %s & ~%s;\n", start, vo, v1, v2); break;
      case A:           std::snprintf(buf, sizeof(buf), "%s%s = %s;\n", start, vo, v1); break;
      case NOT_A_AND_B: std::snprintf(buf, sizeof(buf), "%s%s = ~%s & %s;\n", start, vo, v1, v2); break;
      case B:           std::snprintf(buf, sizeof(buf), "%s%s = %s;\n", start, vo, v2); break;
      case XOR:         std::snprintf(buf, sizeof(buf), "%s%s = %s ^ %s;\n", start, vo, v1, v2); break;
      case OR:          std::snprintf(buf, sizeof(buf), "%s%s = %s | %s;\n", start, vo, v1, v2); break;
      case NOR:         std::snprintf(buf, sizeof(buf), "%s%s = ~(%s | %s);\n", start, vo, v1, v2); break;
      case XNOR:        std::snprintf(buf, sizeof(buf), "%s%s = (%s & %s) | (~%s & ~%s);\n", start, vo, v1, v2, v1, v2); break;
      case NOT_B:       std::snprintf(buf, sizeof(buf), "%s%s = ~%s;\n", start, vo, v2); break;
      case A_OR_NOT_B:  std::snprintf(buf, sizeof(buf), "%s%s = %s | ~%s;\n", start, vo, v1, v2); break;
      case NOT_A:       std::snprintf(buf, sizeof(buf), "%s%s = ~%s;\n", start, vo, v1); break;
      case NOT_A_OR_B:  std::snprintf(buf, sizeof(buf), "%s%s = ~%s | %s;\n", start, vo, v1, v2); break;
      case NAND:        std::snprintf(buf, sizeof(buf), "%s%s = ~(%s & %s);\n", start, vo, v1, v2); break;
      case TRUE_GATE:   std::snprintf(buf, sizeof(buf), "%s%s = ~0;\n", start, vo); break;
      case NOT:         std::snprintf(buf, sizeof(buf), "%s%s = ~%s;\n", start, vo, v1); break;
      case LUT:
        if (lang == LANG_HIP) {
          std::string expr = lut_expr(gt.function, v1, v2, v3);
          if (!decl && vo[0] == '*') {
            std::snprintf(buf, sizeof(buf), "  %s = %s;  /* LUT 0x%02x */\n", vo,
                          expr.c_str(), gt.function);
          } else {
            std::snprintf(buf, sizeof(buf), "  %s %s = %s;  /* LUT 0x%02x */\n",
                          TYPE, vo, expr.c_str(), gt.function);
          }
        } else if (!decl && vo[0] == '*') {
          // LUT gate that is itself an output: route through a temporary
          // (the reference would emit an invalid redeclaration here).
          std::snprintf(buf, sizeof(buf),
                        "  %s lt%d; LUT(lt%d, %s, %s, %s, 0x%02x); %s = lt%d;\n",
                        TYPE, g, g, v1, v2, v3, gt.function, vo, g);
        } else {
          std::snprintf(buf, sizeof(buf), "  %s %s; LUT(%s, %s, %s, %s, 0x%02x);\n",
                        TYPE, vo, vo, v1, v2, v3, gt.function);
        }
        break;
      default:
        if (err != nullptr) *err = "unknown gate type in graph";
        return "";
    }
    out += buf;

    if (!decl && num_outputs == 1) {
      variable_name(st, g, vo, sizeof(vo), ptr_ret);
      std::snprintf(buf, sizeof(buf), "  return %s;\n", vo);
      out += buf;
    }
  }
  out += "}\n";
  return out;
}

}  // namespace sbg
