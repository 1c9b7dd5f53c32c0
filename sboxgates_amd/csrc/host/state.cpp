// state.cpp — circuit-state helpers: metrics, targets, fingerprint,
// file naming. XML persistence lives in xmlio.cpp.

#include "sbg/state.hpp"

#include <cassert>
#include <cstdio>
#include <cstring>

namespace sbg {

int sat_metric_of(int type) {
  // Cost model parity: state.c:168-191.
  switch (type) {
    case FALSE_GATE:  return 1;
    case AND:         return 7;
    case A_AND_NOT_B: return 4;
    case A:           return 4;
    case NOT_A_AND_B: return 7;
    case B:           return 4;
    case XOR:         return 12;
    case OR:          return 7;
    case NOR:         return 7;
    case XNOR:        return 12;
    case NOT_B:       return 4;
    case A_OR_NOT_B:  return 7;
    case NOT_A:       return 4;
    case NOT_A_OR_B:  return 7;
    case NAND:        return 7;
    case TRUE_GATE:   return 1;
    case NOT:         return 4;
    case IN:          return 0;
    default:          return INT_MAX / 4;  // LUT: no SAT metric defined.
  }
}

int get_num_inputs(const state* st) {
  int inputs = 0;
  for (int i = 0; i < st->num_gates && st->gates[i].type == IN; i++) inputs++;
  return inputs;
}

ttable generate_target(u8 bit, const u8* sbox) {
  // Bit i of the table = bit `bit` of sbox[i] (or of i itself when sbox is
  // null, which yields the truth table of raw input bit `bit`).
  ttable t = tt_zero_table();
  for (int i = 0; i < 256; i++) {
    u8 v = sbox != nullptr ? sbox[i] : static_cast<u8>(i);
    tt_set_bit(t, i, (v >> bit) & 1);
  }
  return t;
}

// The Speck-like round function used by the fingerprint
// (parity: state.c:55-63). Not cryptographic; only for unique-ish names.
static inline u32 speck_round(u16 pt1, u16 pt2, u16 k1) {
  pt1 = static_cast<u16>((pt1 >> 7) | (pt1 << 9));
  pt1 = static_cast<u16>(pt1 + pt2);
  pt2 = static_cast<u16>((pt2 >> 14) | (pt2 << 2));
  pt1 ^= k1;
  pt2 ^= pt1;
  return (static_cast<u32>(pt1) << 16) | pt2;
}

u32 state_fingerprint(const state& st) {
  // Hash a normalized copy: bounds zeroed except max_gates, all gates beyond
  // num_gates zeroed, hashed only up to the live prefix — so the fingerprint
  // depends on the circuit, not the search bounds. Parity: state.c:65-105.
  state fps;
  std::memset(&fps, 0, sizeof(state));
  fps.max_gates = st.max_gates;
  fps.num_gates = st.num_gates;
  for (int i = 0; i < 8; i++) fps.outputs[i] = st.outputs[i];
  for (int i = 0; i < st.num_gates; i++) {
    fps.gates[i].table = st.gates[i].table;
    fps.gates[i].type = st.gates[i].type;
    fps.gates[i].in1 = st.gates[i].in1;
    fps.gates[i].in2 = st.gates[i].in2;
    fps.gates[i].in3 = st.gates[i].in3;
    fps.gates[i].function = st.gates[i].function;
  }
  u16 fp1 = 0, fp2 = 0;
  const u16* ptr = reinterpret_cast<const u16*>(&fps);
  size_t len = sizeof(state) - sizeof(gate) * (MAX_GATES - fps.num_gates);
  for (size_t p = 0; p < len / 2; p++) {
    u32 ct = speck_round(fp1, fp2, ptr[p]);
    fp1 = static_cast<u16>(ct >> 16);
    fp2 = static_cast<u16>(ct & 0xffff);
  }
  if (len & 1) {
    u32 ct = speck_round(fp1, fp2, reinterpret_cast<const u8*>(&fps)[len - 1]);
    fp1 = static_cast<u16>(ct >> 16);
    fp2 = static_cast<u16>(ct & 0xffff);
  }
  for (int r = 0; r < 22; r++) {
    u32 ct = speck_round(fp1, fp2, 0);
    fp1 = static_cast<u16>(ct >> 16);
    fp2 = static_cast<u16>(ct & 0xffff);
  }
  return (static_cast<u32>(fp1) << 16) | fp2;
}

std::string state_file_name(const state& st) {
  // O-GGG-MMMM-NNNN-FFFFFFFF.xml; NNNN = output bit numbers in order of
  // inclusion (gate-id order). Parity: state.c:107-125.
  char out[9];
  int num_outputs = 0;
  std::memset(out, 0, sizeof(out));
  for (int i = 0; i < st.num_gates; i++) {
    for (u8 k = 0; k < 8; k++) {
      if (st.outputs[k] == i) {
        out[num_outputs++] = static_cast<char>('0' + k);
        break;
      }
    }
  }
  char name[48];
  std::snprintf(name, sizeof(name), "%d-%03d-%04d-%s-%08x.xml", num_outputs,
                st.num_gates - get_num_inputs(&st), st.sat_metric, out,
                state_fingerprint(st));
  return std::string(name);
}

void init_state(state& st, int num_inputs) {
  std::memset(&st, 0, sizeof(state));
  st.max_sat_metric = INT_MAX;
  st.sat_metric = 0;
  st.max_gates = MAX_GATES;
  st.num_gates = static_cast<gatenum>(num_inputs);
  for (int i = 0; i < num_inputs; i++) {
    st.gates[i].type = IN;
    st.gates[i].table = generate_target(static_cast<u8>(i), nullptr);
    st.gates[i].in1 = NO_GATE;
    st.gates[i].in2 = NO_GATE;
    st.gates[i].in3 = NO_GATE;
    st.gates[i].function = 0;
  }
  for (int i = 0; i < 8; i++) st.outputs[i] = NO_GATE;
}

bool check_num_gates_possible(const state* st, int add, int add_sat,
                              metric_t metric) {
  if (metric == METRIC_SAT && st->sat_metric + add_sat > st->max_sat_metric) {
    return false;
  }
  return st->num_gates + add <= st->max_gates;
}

}  // namespace sbg
