// bindings.cpp — pybind11 bindings for the sboxgates-mi355x engine.
//
// The Python layer (sboxgates_amd.*) is orchestration only: bench driving,
// torch.distributed (RCCL) coordination callbacks, tests. All search
// compute stays in the native engine (host C++ + gfx950 HIP kernels).

#include <pybind11/functional.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstring>
#include <stdexcept>

#include "sbg/boolfunc.hpp"
#include "sbg/codegen.hpp"
#include "sbg/comb.hpp"
#include "sbg/dist.hpp"
#include "sbg/gpu.hpp"
#include "sbg/lutcover.hpp"
#include "sbg/options.hpp"
#include "sbg/rng.hpp"
#include "sbg/sboxio.hpp"
#include "sbg/scan.hpp"
#include "sbg/search.hpp"
#include "sbg/state.hpp"
#include "sbg/xmlio.hpp"

namespace py = pybind11;
using namespace sbg;

namespace {

ttable tt_from_bytes(const py::bytes& b) {
  std::string s = b;
  if (s.size() != 32) throw std::invalid_argument("ttable must be 32 bytes");
  ttable t;
  std::memcpy(&t, s.data(), 32);
  return t;
}

py::bytes tt_to_bytes(const ttable& t) {
  return py::bytes(reinterpret_cast<const char*>(&t), 32);
}

// Python-callback-backed DistCtx: the Python side provides rank/world and
// two callables (bcast(bytes, root) -> bytes; allreduce_min(int) -> int),
// typically implemented with torch.distributed over RCCL (GPU) or gloo
// (CPU tests).
class PyDistCtx : public DistCtx {
 public:
  PyDistCtx(int rank, int world, py::function bcast_fn, py::function min_fn)
      : rank_(rank), world_(world), bcast_fn_(std::move(bcast_fn)),
        min_fn_(std::move(min_fn)) {}

  int rank() const override { return rank_; }
  int world() const override { return world_; }

  void bcast(void* data, size_t n, int root) override {
    py::bytes inp(reinterpret_cast<const char*>(data), n);
    py::bytes out = bcast_fn_(inp, root);
    std::string s = out;
    if (s.size() != n) throw std::runtime_error("bcast size mismatch");
    std::memcpy(data, s.data(), n);
  }

  int allreduce_min(int v) override { return min_fn_(v).cast<int>(); }

 private:
  int rank_, world_;
  py::function bcast_fn_, min_fn_;
};

py::dict fun_to_dict(const boolfunc& f) {
  py::dict d;
  d["num_inputs"] = f.num_inputs;
  d["fun"] = static_cast<int>(f.fun);
  d["fun1"] = f.fun1;
  d["fun2"] = f.fun2;
  d["not_a"] = f.not_a;
  d["not_b"] = f.not_b;
  d["not_c"] = f.not_c;
  d["not_out"] = f.not_out;
  d["ab_commutative"] = f.ab_commutative;
  d["ac_commutative"] = f.ac_commutative;
  d["bc_commutative"] = f.bc_commutative;
  return d;
}

// Deterministically grows a state's gate pool with random 2-input gates —
// used by bench.py to build the synthetic scan workload (random-init
// "weights" analog: the pool content only shapes the data the kernels
// chew through, not the kernel work itself).
void grow_pool_random(state& st, int target_gates, u64 seed) {
  Xorshift1024 rng(seed);
  static const int kinds[4] = {XOR, AND, OR, A_AND_NOT_B};
  while (st.num_gates < target_gates && st.num_gates < MAX_GATES) {
    gatenum a = static_cast<gatenum>(rng.below(st.num_gates));
    gatenum b = static_cast<gatenum>(rng.below(st.num_gates));
    if (a == b) continue;
    int kind = kinds[rng.below(4)];
    gatenum g = add_gate(&st, kind, a, b, METRIC_GATES);
    if (g == NO_GATE) break;
    // Avoid duplicate truth tables (degenerate pools slow nothing but add
    // trivial hits): keep anyway — the reference pool also contains
    // near-duplicates; uniqueness is not required.
  }
}

}  // namespace

PYBIND11_MODULE(_core, m) {
  m.doc() = "sboxgates-mi355x native engine (host C++ + gfx950 HIP kernels)";

  // --- basics ---
  m.attr("MAX_GATES") = MAX_GATES;
  m.attr("NO_GATE") = static_cast<int>(NO_GATE);
  m.def("gpu_available", &gpu_available);
  m.def("gpu_count", &gpu_count);

  // --- ttable ops (bytes <-> 256-bit tables) ---
  m.def("generate_target", [](int bit, py::object sbox) {
    if (sbox.is_none()) return tt_to_bytes(generate_target(bit, nullptr));
    std::string s = py::bytes(sbox);
    if (s.size() != 256) throw std::invalid_argument("sbox must be 256 bytes");
    return tt_to_bytes(generate_target(bit, reinterpret_cast<const u8*>(s.data())));
  }, py::arg("bit"), py::arg("sbox") = py::none());
  m.def("mask_for_inputs", [](int n) { return tt_to_bytes(tt_mask_for_inputs(n)); });
  m.def("gen_ttable_2", [](int fun, py::bytes a, py::bytes b) {
    return tt_to_bytes(gen_ttable_2(fun, tt_from_bytes(a), tt_from_bytes(b)));
  });
  m.def("gen_lut_ttable", [](int fun, py::bytes a, py::bytes b, py::bytes c) {
    return tt_to_bytes(gen_lut_ttable(static_cast<u8>(fun), tt_from_bytes(a),
                                      tt_from_bytes(b), tt_from_bytes(c)));
  });
  m.def("tt_eq_mask", [](py::bytes a, py::bytes b, py::bytes mask) {
    return tt_eq_mask(tt_from_bytes(a), tt_from_bytes(b), tt_from_bytes(mask));
  });

  // --- boolfunc vocabulary ---
  m.def("make_2_input_fun", [](int fun) {
    return fun_to_dict(make_2_input_fun(static_cast<u8>(fun)));
  });
  m.def("function_lists", [](u32 bitfield, bool try_nots) {
    options o;
    o.set_avail_gates(bitfield);
    o.try_nots = try_nots;
    o.derive_function_lists();
    py::list gates, nots, threes;
    for (int i = 0; o.avail_gates[i].num_inputs != 0; i++) {
      gates.append(fun_to_dict(o.avail_gates[i]));
    }
    for (int i = 0; o.avail_not[i].num_inputs != 0; i++) {
      nots.append(fun_to_dict(o.avail_not[i]));
    }
    for (int i = 0; i < o.num_avail_3; i++) threes.append(fun_to_dict(o.avail_3[i]));
    return py::make_tuple(gates, nots, threes);
  });

  // --- combinatorics ---
  m.def("n_choose_k", &n_choose_k);
  m.def("nth_combination", [](i64 n, int num, int k) {
    std::vector<gatenum> ret(k);
    nth_combination(n, num, k, 0, ret.data());
    return ret;
  });
  m.def("combination_rank", [](std::vector<gatenum> c, int num) {
    return combination_rank(c.data(), static_cast<int>(c.size()), num);
  });
  m.def("decode_pair", [](i64 q, int m) {
    int d, e;
    decode_pair(q, m, &d, &e);
    return py::make_tuple(d, e);
  });

  // --- lutcover primitives (for oracle tests) ---
  m.def("naive_check_n_lut_possible",
        [](int num, py::bytes target, py::bytes mask, std::vector<py::bytes> tables) {
          std::vector<ttable> tt;
          for (auto& b : tables) tt.push_back(tt_from_bytes(b));
          return naive_check_n_lut_possible(num, tt_from_bytes(target),
                                            tt_from_bytes(mask), tt.data());
        });
  m.def("naive_get_lut_function",
        [](py::bytes a, py::bytes b, py::bytes c, py::bytes target, py::bytes mask) {
          u8 func;
          bool ok = naive_get_lut_function(tt_from_bytes(a), tt_from_bytes(b),
                                           tt_from_bytes(c), tt_from_bytes(target),
                                           tt_from_bytes(mask), &func);
          return py::make_tuple(ok, static_cast<int>(func));
        });
  m.def("lut5_solve", [](std::vector<py::bytes> tables, py::bytes target,
                         py::bytes mask, u64 rnd) {
    std::vector<ttable> tt;
    for (auto& b : tables) tt.push_back(tt_from_bytes(b));
    ttable T = tt_from_bytes(target), M = tt_from_bytes(mask);
    u32 p1, p0;
    if (!lut5_p_masks(tt.data(), T & M, ~T & M, &p1, &p0)) {
      return py::make_tuple(false, 0, 0, 0);
    }
    u8 fo, fi;
    int split;
    if (!lut5_solve_from_p(p1, p0, rnd, &fo, &fi, &split)) {
      return py::make_tuple(false, 0, 0, 0);
    }
    return py::make_tuple(true, static_cast<int>(fo), static_cast<int>(fi),
                          static_cast<int>(split));
  });
  m.def("lut7_solve", [](std::vector<py::bytes> tables, py::bytes target,
                         py::bytes mask, u64 rnd) -> py::tuple {
    std::vector<ttable> tt;
    for (auto& b : tables) tt.push_back(tt_from_bytes(b));
    ttable T = tt_from_bytes(target), M = tt_from_bytes(mask);
    u64 p1[2], p0[2];
    if (!lut7_p_masks(tt.data(), T & M, ~T & M, p1, p0)) {
      return py::make_tuple(false, py::none());
    }
    for (int o = 0; o < LUT7_NUM_ORDERINGS; o++) {
      u8 ord[7];
      lut7_ordering(o, ord);
      u8 fo, fm, fi;
      if (lut7_solve_ordering(p1, p0, ord, rnd, &fo, &fm, &fi)) {
        py::list ret;
        ret.append(static_cast<int>(fo));
        ret.append(static_cast<int>(fm));
        ret.append(static_cast<int>(fi));
        for (int j = 0; j < 7; j++) ret.append(static_cast<int>(ord[j]));
        return py::make_tuple(true, py::object(ret));
      }
    }
    return py::make_tuple(false, py::object(py::none()));
  });
  m.def("splits5", []() {
    py::list out;
    for (auto& sp : SPLITS5) {
      py::list row;
      for (u8 v : sp) row.append(static_cast<int>(v));
      out.append(row);
    }
    return out;
  });
  m.def("lut7_ordering", [](int idx) {
    u8 ord[7];
    lut7_ordering(idx, ord);
    std::vector<int> out(ord, ord + 7);
    return out;
  });

  // --- state ---
  py::class_<state>(m, "State")
      .def(py::init([](int num_inputs) {
        state st;
        init_state(st, num_inputs);
        return st;
      }), py::arg("num_inputs"))
      .def_property_readonly("num_gates", [](const state& s) { return s.num_gates; })
      .def_property_readonly("num_inputs", [](const state& s) { return get_num_inputs(&s); })
      .def_property_readonly("sat_metric", [](const state& s) { return s.sat_metric; })
      .def_property("max_gates", [](const state& s) { return s.max_gates; },
                    [](state& s, int v) { s.max_gates = static_cast<gatenum>(v); })
      .def_property_readonly("outputs", [](const state& s) {
        std::vector<int> out;
        for (int i = 0; i < 8; i++) {
          out.push_back(s.outputs[i] == NO_GATE ? -1 : s.outputs[i]);
        }
        return out;
      })
      .def("set_output", [](state& s, int bit, int g) {
        s.outputs[bit] = g < 0 ? NO_GATE : static_cast<gatenum>(g);
      })
      .def("gate", [](const state& s, int i) {
        if (i < 0 || i >= s.num_gates) throw std::out_of_range("gate index");
        const gate& g = s.gates[i];
        py::dict d;
        d["type"] = g.type;
        d["type_name"] = gate_name[g.type];
        d["in1"] = g.in1 == NO_GATE ? -1 : g.in1;
        d["in2"] = g.in2 == NO_GATE ? -1 : g.in2;
        d["in3"] = g.in3 == NO_GATE ? -1 : g.in3;
        d["function"] = static_cast<int>(g.function);
        d["table"] = tt_to_bytes(g.table);
        return d;
      })
      .def("add_gate", [](state& s, int type, int g1, int g2) {
        gatenum r = add_gate(&s, type, static_cast<gatenum>(g1),
                             g2 < 0 ? NO_GATE : static_cast<gatenum>(g2), METRIC_GATES);
        return r == NO_GATE ? -1 : static_cast<int>(r);
      }, py::arg("type"), py::arg("g1"), py::arg("g2") = -1)
      .def("add_lut", [](state& s, int func, int g1, int g2, int g3) {
        ttable t = gen_lut_ttable(static_cast<u8>(func), s.gates[g1].table,
                                  s.gates[g2].table, s.gates[g3].table);
        gatenum r = add_lut(&s, static_cast<u8>(func), t, static_cast<gatenum>(g1),
                            static_cast<gatenum>(g2), static_cast<gatenum>(g3));
        return r == NO_GATE ? -1 : static_cast<int>(r);
      })
      .def("eval", [](const state& s, int input) {
        return static_cast<int>(eval_circuit(s, static_cast<u8>(input)));
      })
      .def("fingerprint", [](const state& s) { return state_fingerprint(s); })
      .def("file_name", [](const state& s) { return state_file_name(s); })
      .def("to_xml", [](const state& s) { return state_to_xml(s); })
      .def("save", [](const state& s, const std::string& dir) {
        return save_state(s, dir);
      }, py::arg("dir") = std::string())
      .def_static("from_xml", [](const std::string& xml) {
        state st;
        std::string err;
        if (!state_from_xml(xml, &st, &err)) throw std::runtime_error(err);
        return st;
      })
      .def_static("load", [](const std::string& path) {
        state st;
        std::string err;
        if (!load_state(path, &st, &err)) throw std::runtime_error(err);
        return st;
      })
      .def("copy", [](const state& s) { return state(s); })
      .def("grow_pool_random", [](state& s, int target, u64 seed) {
        grow_pool_random(s, target, seed);
      });

  // --- codegen ---
  m.def("graph_to_dot", &graph_to_dot);
  m.def("graph_to_source", [](const state& st, const std::string& lang) {
    codegen_lang l = LANG_AUTO;
    if (lang == "c") l = LANG_C;
    else if (lang == "cuda") l = LANG_CUDA;
    else if (lang == "hip") l = LANG_HIP;
    std::string err;
    std::string src = graph_to_source(st, l, &err);
    if (src.empty()) throw std::runtime_error(err);
    return src;
  }, py::arg("state"), py::arg("lang") = "auto");
  m.def("ttable_to_string", [](py::bytes t) {
    return ttable_to_string(tt_from_bytes(t));
  });

  // --- sbox io ---
  m.def("load_sbox_file", [](const std::string& path, int permute) {
    u8 sbox[256];
    u32 num_inputs;
    std::string err;
    if (!load_sbox_file(path, permute, sbox, &num_inputs, &err)) {
      throw std::runtime_error(err);
    }
    return py::make_tuple(py::bytes(reinterpret_cast<char*>(sbox), 256),
                          static_cast<int>(num_inputs));
  }, py::arg("path"), py::arg("permute") = 0);
  m.def("load_sbox_table", [](std::vector<int> table, int permute) {
    std::vector<u8> t8(table.begin(), table.end());
    u8 sbox[256];
    u32 num_inputs;
    std::string err;
    if (!load_sbox_table(t8.data(), static_cast<int>(t8.size()), permute, sbox,
                         &num_inputs, &err)) {
      throw std::runtime_error(err);
    }
    return py::make_tuple(py::bytes(reinterpret_cast<char*>(sbox), 256),
                          static_cast<int>(num_inputs));
  }, py::arg("table"), py::arg("permute") = 0);

  // --- options ---
  py::class_<options>(m, "Options")
      .def(py::init([]() {
        options o;
        o.set_avail_gates(DEFAULT_GATE_BITFIELD);
        o.verbosity = -1;  // library default: quiet
        return o;
      }))
      .def_readwrite("iterations", &options::iterations)
      .def_readwrite("jobs", &options::jobs)
      .def_readwrite("oneoutput", &options::oneoutput)
      .def_readwrite("permute", &options::permute)
      .def_readwrite("lut_graph", &options::lut_graph)
      .def_readwrite("try_nots", &options::try_nots)
      .def_readwrite("verbosity", &options::verbosity)
      .def_readwrite("seeded", &options::seeded)
      .def_readwrite("seed", &options::seed)
      .def_readwrite("output_dir", &options::output_dir)
      .def_readwrite("save_states", &options::save_states)
      .def_property("metric",
                    [](const options& o) { return o.metric == METRIC_SAT ? "sat" : "gates"; },
                    [](options& o, const std::string& v) {
                      o.metric = v == "sat" ? METRIC_SAT : METRIC_GATES;
                    })
      .def_property("gpu",
                    [](const options& o) {
                      return o.gpu == GPU_OFF ? "off" : (o.gpu == GPU_FORCE ? "force" : "auto");
                    },
                    [](options& o, const std::string& v) {
                      o.gpu = v == "off" ? GPU_OFF : (v == "force" ? GPU_FORCE : GPU_AUTO);
                    })
      .def("set_avail_gates", [](options& o, u32 bf) { o.set_avail_gates(bf); })
      .def("derive_function_lists", &options::derive_function_lists);

  // --- dist ctx ---
  py::class_<DistCtx>(m, "DistCtx");
  py::class_<PyDistCtx, DistCtx>(m, "PyDistCtx")
      .def(py::init<int, int, py::function, py::function>(), py::arg("rank"),
           py::arg("world"), py::arg("bcast"), py::arg("allreduce_min"));

  // --- engine ---
  py::class_<Engine>(m, "Engine")
      .def(py::init([](const options& opt, DistCtx* ctx) {
        return new Engine(opt, ctx);
      }), py::arg("options"), py::arg("ctx") = nullptr,
         py::keep_alive<1, 3>())
      .def("set_sbox", [](Engine& e, py::bytes sbox, int num_inputs) {
        std::string s = sbox;
        if (s.size() != 256) throw std::invalid_argument("sbox must be 256 bytes");
        e.set_sbox(reinterpret_cast<const u8*>(s.data()), num_inputs);
      })
      .def_property_readonly("num_inputs", &Engine::num_inputs)
      .def_property_readonly("num_outputs", &Engine::num_outputs)
      .def_property_readonly("gpu_active", &Engine::gpu_active)
      .def("target", [](Engine& e, int bit) { return tt_to_bytes(e.target(bit)); })
      .def("initial_state", [](Engine& e) {
        state st;
        e.initial_state(st);
        return st;
      })
      .def("generate_graph", [](Engine& e, const state& st) { e.generate_graph(st); })
      .def("generate_graph_one_output",
           [](Engine& e, const state& st) { e.generate_graph_one_output(st); })
      .def("create_circuit", [](Engine& e, state& st, py::bytes target,
                                py::bytes mask) {
        i8 bits[8] = {-1, -1, -1, -1, -1, -1, -1, -1};
        gatenum r = e.create_circuit(&st, tt_from_bytes(target), tt_from_bytes(mask),
                                     bits);
        return r == NO_GATE ? -1 : static_cast<int>(r);
      })
      .def("worker_loop", &Engine::worker_loop)
      .def("stop_workers", &Engine::stop_workers)
      .def("saved_files", &Engine::saved_files)
      .def("stats", [](const Engine& e) {
        py::dict d;
        d["candidates3"] = e.stats().candidates3;
        d["candidates5"] = e.stats().candidates5;
        d["candidates7"] = e.stats().candidates7;
        d["gpu_scans"] = e.stats().gpu_scans;
        d["cpu_scans"] = e.stats().cpu_scans;
        d["scan_seconds3"] = e.stats().scan_seconds3;
        d["scan_seconds5"] = e.stats().scan_seconds5;
        d["scan_seconds7"] = e.stats().scan_seconds7;
        d["nodes"] = e.stats().nodes;
        d["step12_seconds"] = e.stats().step12_seconds;
        d["step3_seconds"] = e.stats().step3_seconds;
        d["step4a_seconds"] = e.stats().step4a_seconds;
        return d;
      })
      .def("scan_pool", [](Engine& e, int k, const state& st, py::bytes target,
                           py::bytes mask, i64 begin, i64 end, u64 seed,
                           bool count_all) {
        std::vector<ttable> pool(st.num_gates);
        for (int i = 0; i < st.num_gates; i++) pool[i] = st.gates[i].table;
        ScanRequest rq;
        rq.tables = pool.data();
        rq.n = st.num_gates;
        rq.target = tt_from_bytes(target);
        rq.mask = tt_from_bytes(mask);
        rq.excl_low64 = 0;
        rq.seed = seed;
        rq.count_all = count_all;
        if (k == 4) rq.matcher = e.matcher3();
        ScanResult r = e.scan(k, rq, begin, end);
        std::vector<int> res(r.res, r.res + 10);
        return py::make_tuple(r.found, res, r.evaluated);
      }, py::arg("k"), py::arg("state"), py::arg("target"), py::arg("mask"),
         py::arg("begin"), py::arg("end"), py::arg("seed") = 0,
         py::arg("count_all") = false);
}
