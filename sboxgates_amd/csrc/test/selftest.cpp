// selftest.cpp — native self-test binary, built with ASan+UBSan (and
// optionally TSan) by `make asan` / `make tsan`. Exercises the host engine
// end-to-end: vocabulary building, searches in both modes, XML round-trip,
// codegen, scans with planted solutions, and the in-process thread SPMD
// protocol. The reference has no sanitizer coverage at all.

#include <cassert>
#include <cstdio>
#include <cstring>
#include <string>
#include <thread>
#include <vector>

#include "sbg/codegen.hpp"
#include "sbg/comb.hpp"
#include "sbg/lutcover.hpp"
#include "sbg/options.hpp"
#include "sbg/rng.hpp"
#include "sbg/sboxio.hpp"
#include "sbg/scan.hpp"
#include "sbg/search.hpp"
#include "sbg/state.hpp"
#include "sbg/threads_ctx.hpp"
#include "sbg/xmlio.hpp"

using namespace sbg;

#define CHECK(cond)                                                        \
  do {                                                                     \
    if (!(cond)) {                                                         \
      std::fprintf(stderr, "selftest FAILED: %s at %s:%d\n", #cond,        \
                   __FILE__, __LINE__);                                    \
      return 1;                                                            \
    }                                                                      \
  } while (0)

static const u8 kDesS1[64] = {
    0xe, 0x4, 0xd, 0x1, 0x2, 0xf, 0xb, 0x8, 0x3, 0xa, 0x6, 0xc, 0x5,
    0x9, 0x0, 0x7, 0x0, 0xf, 0x7, 0x4, 0xe, 0x2, 0xd, 0x1, 0xa, 0x6,
    0xc, 0xb, 0x9, 0x5, 0x3, 0x8, 0x4, 0x1, 0xe, 0x8, 0xd, 0x6, 0x2,
    0xb, 0xf, 0xc, 0x9, 0x7, 0x3, 0xa, 0x5, 0x0, 0xf, 0xc, 0x8, 0x2,
    0x4, 0x9, 0x1, 0x7, 0x5, 0xb, 0x3, 0xe, 0xa, 0x0, 0x6, 0xd};

static bool circuit_bit_ok(const state& st, const u8* sbox, int n, int bit) {
  for (int x = 0; x < (1 << n); x++) {
    if (((eval_circuit(st, static_cast<u8>(x)) >> bit) & 1) !=
        ((sbox[x] >> bit) & 1)) {
      return false;
    }
  }
  return true;
}

int main() {
  u8 sbox[256];
  u32 n = 0;
  std::string err;
  CHECK(load_sbox_table(kDesS1, 64, 0, sbox, &n, &err));
  CHECK(n == 6);

  // Gate-mode search.
  {
    options opt;
    opt.set_avail_gates(DEFAULT_GATE_BITFIELD);
    opt.seeded = true;
    opt.seed = 7;
    opt.gpu = GPU_OFF;
    opt.save_states = false;
    opt.verbosity = -1;
    opt.derive_function_lists();
    Engine eng(opt);
    eng.set_sbox(sbox, 6);
    state st;
    eng.initial_state(st);
    i8 bits[8] = {-1, -1, -1, -1, -1, -1, -1, -1};
    gatenum out = eng.create_circuit(&st, eng.target(0), tt_mask_for_inputs(6), bits);
    CHECK(out != NO_GATE);
    st.outputs[0] = out;
    CHECK(circuit_bit_ok(st, sbox, 6, 0));

    // XML round-trip + codegen on the result.
    std::string xml = state_to_xml(st);
    state st2;
    CHECK(state_from_xml(xml, &st2, &err));
    CHECK(state_to_xml(st2) == xml);
    std::string src = graph_to_source(st2, LANG_AUTO, &err);
    CHECK(!src.empty());
    CHECK(!graph_to_dot(st2).empty());
  }

  // LUT-mode search with 3-thread in-process SPMD.
  {
    ThreadGroup group(3);
    options opt;
    opt.set_avail_gates(DEFAULT_GATE_BITFIELD);
    opt.seeded = true;
    opt.seed = 21;
    opt.gpu = GPU_OFF;
    opt.lut_graph = true;
    opt.save_states = false;
    opt.verbosity = -1;
    opt.derive_function_lists();
    std::vector<std::thread> workers;
    for (int r = 1; r < 3; r++) {
      workers.emplace_back([&, r] {
        Engine we(opt, group.ctx(r));
        we.set_sbox(sbox, 6);
        we.worker_loop();
      });
    }
    Engine eng(opt, group.ctx(0));
    eng.set_sbox(sbox, 6);
    state st;
    eng.initial_state(st);
    i8 bits[8] = {-1, -1, -1, -1, -1, -1, -1, -1};
    gatenum out = eng.create_circuit(&st, eng.target(0), tt_mask_for_inputs(6), bits);
    eng.stop_workers();
    for (auto& t : workers) t.join();
    CHECK(out != NO_GATE);
    st.outputs[0] = out;
    CHECK(circuit_bit_ok(st, sbox, 6, 0));
  }

  // Scans with planted solutions over a random pool.
  {
    state st;
    init_state(st, 8);
    Xorshift1024 rng(123);
    while (st.num_gates < 30) {
      gatenum a = static_cast<gatenum>(rng.below(st.num_gates));
      gatenum b = static_cast<gatenum>(rng.below(st.num_gates));
      if (a == b) continue;
      add_gate(&st, XOR, a, b, METRIC_GATES);
    }
    std::vector<ttable> pool(st.num_gates);
    for (int i = 0; i < st.num_gates; i++) pool[i] = st.gates[i].table;
    ttable t_outer = gen_lut_ttable(0xE8, pool[3], pool[7], pool[12]);
    ttable target = gen_lut_ttable(0x4A, t_outer, pool[17], pool[22]);
    ScanRequest rq;
    rq.tables = pool.data();
    rq.n = st.num_gates;
    rq.target = target;
    rq.mask = tt_ones_table();
    rq.excl_low64 = 0;
    rq.seed = 5;
    rq.count_all = false;
    ScanResult r = cpu_scan5(rq, 0, n_choose_k(st.num_gates, 5));
    CHECK(r.found);
    ttable got_outer = gen_lut_ttable(static_cast<u8>(r.res[0]), pool[r.res[2]],
                                      pool[r.res[3]], pool[r.res[4]]);
    ttable got = gen_lut_ttable(static_cast<u8>(r.res[1]), got_outer,
                                pool[r.res[5]], pool[r.res[6]]);
    CHECK(tt_eq(got, target));
  }

  std::printf("selftest ok\n");
  return 0;
}
