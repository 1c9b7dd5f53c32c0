// main.cpp — sboxgates command-line driver.
//
// Flag-for-flag compatible with the reference CLI (sboxgates.c:43-73,
// 895-1174): -a -c -d -g -i -l -n -o -p -s -v plus INPUT_FILE, same
// defaults (gates = AND|OR|XOR, iterations = 1), same conflict rules
// (-c/-d mutually exclusive, -l/-s mutually exclusive). MI355X additions:
// --convert-hip, --seed N, --cpu / --gpu, --output-dir DIR.
//
// Unlike the reference there is no MPI launcher: --gpus N drives N devices
// from ONE process (one host thread + one engine per GPU, shared-memory
// coordination over the same chunked protocol). The Python layer
// (sboxgates_amd.parallel, torchrun one process per GPU over RCCL) covers
// multi-process deployment.

#include <getopt.h>
#include <dirent.h>
#include <sys/stat.h>
#include <sys/types.h>

#include <atomic>
#include <climits>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <thread>
#include <vector>

#include "sbg/codegen.hpp"
#include "sbg/gpu.hpp"
#include "sbg/options.hpp"
#include "sbg/sboxio.hpp"
#include "sbg/search.hpp"
#include "sbg/threads_ctx.hpp"
#include "sbg/xmlio.hpp"

namespace {

const char* kVersion = "sboxgates-mi355x 1.0";

void print_help(const char* prog) {
  std::printf(
      "Usage: %s [OPTION...] INPUT_FILE\n"
      "Generates graphs of Boolean gates or 3-input LUTs that realize a specified\n"
      "S-box. Generated graphs can be converted to C/CUDA/HIP source code or to\n"
      "Graphviz DOT format.\n\n"
      " Graph generation\n"
      "  -a, --available-gates=gates   Specify the set of available gates\n"
      "                                (bitfield 0-65535).\n"
      "  -g, --graph=graph             Load graph from file as initial state.\n"
      "                                (For use with -o.)\n"
      "  -i, --iterations=iterations   Set number of iterations per step.\n"
      "  -l, --lut                     Generate LUT graph. Results in smaller\n"
      "                                graphs but takes significantly more time.\n"
      "  -n, --append-not              Try to generate more boolean functions by\n"
      "                                appending NOT gates.\n"
      "  -o, --single-output=output    Generate single-output graph for specified\n"
      "                                output.\n"
      "  -p, --permute=value           Permute the input S-box by XORing it with\n"
      "                                value.\n"
      "  -s, --sat-metric              Use SAT metric.\n"
      "  -v, --verbose                 Increase verbosity.\n\n"
      " Graph conversion\n"
      "  -c, --convert-c               Convert input file to a C or CUDA function.\n"
      "  -d, --convert-dot             Convert input file to a DOT digraph.\n"
      "      --convert-hip             Convert input file to a HIP device function.\n\n"
      " MI355X engine\n"
      "      --seed=N                  Deterministic RNG seed.\n"
      "      --cpu                     Disable the GPU path.\n"
      "      --gpu                     Require the GPU path (fail if no device).\n"
      "      --output-dir=DIR          Directory for XML checkpoint files.\n"
      "      --resume-dir=DIR          Write checkpoints to DIR and resume the\n"
      "                                search from the best state file already\n"
      "                                in it (most outputs, then fewest gates).\n"
      "      --gpus=N                  Drive N GPUs from this process (one\n"
      "                                worker thread per device).\n"
      "      --beam=N                  Tied-state beam width for multi-output\n"
      "                                search (default 20).\n"
      "      --max-gates=N             Bound the search to N total gates\n"
      "                                (including inputs).\n"
      "      --jobs=N                  Run -i iterations as N parallel jobs\n"
      "                                (single-output mode; jobs rotate over\n"
      "                                visible GPUs).\n\n"
      "  -?, --help                    Give this help list.\n"
      "  -V, --version                 Print program version.\n",
      prog);
}

int fail(const char* msg, const char* arg) {
  if (arg != nullptr) {
    std::fprintf(stderr, "%s: %s\n", msg, arg);
  } else {
    std::fprintf(stderr, "%s\n", msg);
  }
  return 1;
}

}  // namespace

int main(int argc, char** argv) {
  std::setvbuf(stdout, nullptr, _IOLBF, 0);  // keep logs on SIGKILL'd runs
  sbg::options opt;
  opt.set_avail_gates(sbg::DEFAULT_GATE_BITFIELD);

  enum { OPT_HIP = 1000, OPT_SEED, OPT_CPU, OPT_GPU, OPT_OUTDIR, OPT_HELP,
         OPT_GPUS, OPT_BEAM, OPT_JOBS, OPT_MAXG, OPT_RESUME };
  static const struct option long_opts[] = {
      {"available-gates", required_argument, nullptr, 'a'},
      {"convert-c", no_argument, nullptr, 'c'},
      {"convert-dot", no_argument, nullptr, 'd'},
      {"graph", required_argument, nullptr, 'g'},
      {"iterations", required_argument, nullptr, 'i'},
      {"lut", no_argument, nullptr, 'l'},
      {"append-not", no_argument, nullptr, 'n'},
      {"single-output", required_argument, nullptr, 'o'},
      {"permute", required_argument, nullptr, 'p'},
      {"sat-metric", no_argument, nullptr, 's'},
      {"verbose", no_argument, nullptr, 'v'},
      {"convert-hip", no_argument, nullptr, OPT_HIP},
      {"seed", required_argument, nullptr, OPT_SEED},
      {"cpu", no_argument, nullptr, OPT_CPU},
      {"gpu", no_argument, nullptr, OPT_GPU},
      {"output-dir", required_argument, nullptr, OPT_OUTDIR},
      {"resume-dir", required_argument, nullptr, OPT_RESUME},
      {"gpus", required_argument, nullptr, OPT_GPUS},
      {"beam", required_argument, nullptr, OPT_BEAM},
      {"jobs", required_argument, nullptr, OPT_JOBS},
      {"max-gates", required_argument, nullptr, OPT_MAXG},
      {"help", no_argument, nullptr, OPT_HELP},
      {"version", no_argument, nullptr, 'V'},
      {nullptr, 0, nullptr, 0}};

  int ch;
  char* endptr = nullptr;
  long v;
  int max_gates_bound = -1;  // --max-gates: bound on TOTAL gates (incl. inputs)
  std::string resume_dir;    // --resume-dir: pick up the best state in DIR
  while ((ch = getopt_long(argc, argv, "a:cdg:i:lno:p:svV", long_opts, nullptr)) != -1) {
    switch (ch) {
      case 'a':
        v = std::strtol(optarg, &endptr, 10);
        if (*endptr != '\0' || v <= 0 || v > 65535) {
          return fail("Bad available gates value", optarg);
        }
        opt.set_avail_gates(static_cast<sbg::u32>(v));
        break;
      case 'c': opt.output_c = true; break;
      case 'd': opt.output_dot = true; break;
      case 'g': opt.gfname = optarg; break;
      case 'i':
        v = std::strtol(optarg, &endptr, 10);
        if (*endptr != '\0' || v < 1) return fail("Bad iterations value", optarg);
        opt.iterations = static_cast<int>(v);
        break;
      case 'l': opt.lut_graph = true; break;
      case 'n': opt.try_nots = true; break;
      case 'o':
        v = std::strtol(optarg, &endptr, 10);
        if (*endptr != '\0' || v < 0 || v > 7) return fail("Bad output value", optarg);
        opt.oneoutput = static_cast<int>(v);
        break;
      case 'p':
        v = std::strtol(optarg, &endptr, 10);
        if (*endptr != '\0' || v < 0 || v > 255) {
          return fail("Bad permutation value", optarg);
        }
        opt.permute = static_cast<int>(v);
        break;
      case 's': opt.metric = sbg::METRIC_SAT; break;
      case 'v': opt.verbosity += 1; break;
      case OPT_HIP: opt.output_hip = true; break;
      case OPT_SEED:
        opt.seeded = true;
        opt.seed = std::strtoull(optarg, &endptr, 0);
        if (*endptr != '\0') return fail("Bad seed value", optarg);
        break;
      case OPT_CPU: opt.gpu = sbg::GPU_OFF; break;
      case OPT_GPU: opt.gpu = sbg::GPU_FORCE; break;
      case OPT_OUTDIR:
        opt.output_dir = optarg;
        (void)mkdir(optarg, 0755);  // best-effort; open errors surface later
        break;
      case OPT_RESUME:
        opt.output_dir = optarg;
        resume_dir = optarg;
        (void)mkdir(optarg, 0755);
        break;
      case OPT_GPUS:
        v = std::strtol(optarg, &endptr, 10);
        if (*endptr != '\0' || v < 1 || v > 64) return fail("Bad --gpus value", optarg);
        opt.num_gpus = static_cast<int>(v);
        break;
      case OPT_BEAM:
        v = std::strtol(optarg, &endptr, 10);
        if (*endptr != '\0' || v < 1 || v > 20) return fail("Bad --beam value", optarg);
        opt.beam = static_cast<int>(v);
        break;
      case OPT_JOBS:
        v = std::strtol(optarg, &endptr, 10);
        if (*endptr != '\0' || v < 1 || v > 64) return fail("Bad --jobs value", optarg);
        opt.jobs = static_cast<int>(v);
        break;
      case OPT_MAXG:
        v = std::strtol(optarg, &endptr, 10);
        if (*endptr != '\0' || v < 1 || v > 500) {
          return fail("Bad --max-gates value", optarg);
        }
        max_gates_bound = static_cast<int>(v);
        break;
      case 'V': std::printf("%s\n", kVersion); return 0;
      case OPT_HELP: print_help(argv[0]); return 0;
      default: return 1;
    }
  }

  if (optind < argc) opt.fname = argv[optind];

  // Conflict rules (parity: sboxgates.c:958-968).
  int conv = (opt.output_c ? 1 : 0) + (opt.output_dot ? 1 : 0) + (opt.output_hip ? 1 : 0);
  if (conv > 1) return fail("Cannot combine conversion options", nullptr);
  if (opt.lut_graph && opt.metric == sbg::METRIC_SAT) {
    return fail("SAT metric can not be combined with LUT graph generation", nullptr);
  }
  if (opt.fname.empty()) return fail("Input file name argument missing", nullptr);

  opt.derive_function_lists();

  try {
    // Conversion mode: input file is a graph XML (parity: sboxgates.c:1096-1113).
    if (conv == 1) {
      sbg::state st;
      std::string err;
      if (!sbg::load_state(opt.fname, &st, &err)) {
        return fail("Error when reading state file", err.c_str());
      }
      if (opt.output_dot) {
        std::fputs(sbg::graph_to_dot(st).c_str(), stdout);
        return 0;
      }
      sbg::codegen_lang lang = opt.output_hip ? sbg::LANG_HIP : sbg::LANG_AUTO;
      std::string src = sbg::graph_to_source(st, lang, &err);
      if (src.empty()) return fail("Conversion failed", err.c_str());
      std::fputs(src.c_str(), stdout);
      return 0;
    }

    // Search mode.
    sbg::u8 sbox[256];
    sbg::u32 num_inputs = 0;
    std::string err;
    if (!sbg::load_sbox_file(opt.fname, opt.permute, sbox, &num_inputs, &err)) {
      return fail("Error loading S-box", err.c_str());
    }
    if (opt.verbosity >= 2) {
      std::printf("Loaded %u input S-box:\n", num_inputs);
      for (int i = 0; i < (1 << num_inputs); i++) {
        std::printf("%02x%s", sbox[i], (i + 1) % 16 ? " " : "\n");
      }
    }

    sbg::ThreadGroup group(opt.num_gpus);
    std::vector<std::thread> workers;
    std::atomic<bool> worker_failed{false};
    const int dev_count = sbg::gpu_count();
    for (int r = 1; r < opt.num_gpus; r++) {
      workers.emplace_back([&, r] {
        try {
          sbg::options wopt = opt;
          wopt.gpu_device = dev_count > 0 ? r % dev_count : -1;
          sbg::Engine we(wopt, group.ctx(r));
          we.set_sbox(sbox, static_cast<int>(num_inputs));
          we.worker_loop();
        } catch (const std::exception& e) {
          std::fprintf(stderr, "worker %d: %s\n", r, e.what());
          worker_failed = true;
          std::exit(1);  // protocol is broken; do not deadlock rank 0
        }
      });
    }
    sbg::options opt0 = opt;
    opt0.gpu_device = dev_count > 0 ? 0 : -1;
    sbg::Engine engine(opt0, group.ctx(0));
    engine.set_sbox(sbox, static_cast<int>(num_inputs));

    if (opt.verbosity >= 1) {
      std::printf("Available gates: NOT ");
      for (int i = 0; opt.avail_gates[i].num_inputs != 0; i++) {
        std::printf("%s ", sbg::gate_name[opt.avail_gates[i].fun]);
      }
      std::printf("\nGenerated gates: ");
      for (int i = 0; opt.avail_not[i].num_inputs != 0; i++) {
        std::printf("%s ", sbg::gate_name[opt.avail_not[i].fun]);
      }
      std::printf("\nGenerated 3-input gates: ");
      for (int i = 0; i < opt.num_avail_3; i++) {
        std::printf("%02x ", opt.avail_3[i].fun);
      }
      std::printf("\n");
      std::printf("GPU: %s\n", engine.gpu_active() ? "active" : "off (CPU path)");
    }

    if (opt.oneoutput >= engine.num_outputs()) {
      std::fprintf(stderr,
                   "Error: Can't generate output bit %d. Target S-box only has %d "
                   "outputs.\n",
                   opt.oneoutput, engine.num_outputs());
      return 1;
    }

    sbg::state st;
    if (!resume_dir.empty() && opt.gfname.empty()) {
      // Resume: scan DIR for state files and pick the most advanced one
      // (most wired outputs, then fewest gates). The beam driver then
      // continues adding outputs from that state — checkpoint/restart
      // across budget windows for long multi-output runs.
      engine.initial_state(st);
      int best_outputs = -1;
      std::string best_file;
      if (DIR* d = opendir(resume_dir.c_str())) {
        while (struct dirent* e = readdir(d)) {
          std::string name = e->d_name;
          if (name.size() < 5 || name.substr(name.size() - 4) != ".xml") continue;
          sbg::state cand;
          std::string lerr;
          if (!sbg::load_state(resume_dir + "/" + name, &cand, &lerr)) continue;
          if (sbg::get_num_inputs(&cand) != static_cast<int>(num_inputs)) continue;
          int outs = 0;
          for (int i = 0; i < 8; i++) {
            if (cand.outputs[i] != sbg::NO_GATE) outs += 1;
          }
          if (outs > best_outputs ||
              (outs == best_outputs && cand.num_gates < st.num_gates)) {
            best_outputs = outs;
            st = cand;
            best_file = name;
          }
        }
        closedir(d);
      }
      if (best_outputs > 0) {
        std::printf("Resuming from %s/%s (%d output%s, %d gates).\n",
                    resume_dir.c_str(), best_file.c_str(), best_outputs,
                    best_outputs == 1 ? "" : "s",
                    st.num_gates - sbg::get_num_inputs(&st));
      }
    } else if (opt.gfname.empty()) {
      engine.initial_state(st);
    } else if (!sbg::load_state(opt.gfname, &st, &err)) {
      return fail("Error when reading state file", err.c_str());
    } else {
      std::printf("Loaded %s.\n", opt.gfname.c_str());
    }
    if (max_gates_bound > 0) {
      st.max_gates = static_cast<sbg::gatenum>(max_gates_bound);
    }

    if (opt.oneoutput != -1) {
      engine.generate_graph_one_output(st);
    } else {
      engine.generate_graph(st);
    }
    engine.stop_workers();
    for (auto& t : workers) t.join();
    if (worker_failed.load()) return 1;
    if (opt.verbosity >= 1) {
      const auto& st_ = engine.stats();
      auto rate = [](sbg::u64 c, double s) { return s > 0 ? c / s : 0.0; };
      std::printf("Scan totals: 3-in %llu (%.3g/s), 5LUT %llu (%.3g/s), "
                  "7LUT %llu (%.3g/s); %llu GPU / %llu CPU scans\n",
                  (unsigned long long)st_.candidates3,
                  rate(st_.candidates3, st_.scan_seconds3),
                  (unsigned long long)st_.candidates5,
                  rate(st_.candidates5, st_.scan_seconds5),
                  (unsigned long long)st_.candidates7,
                  rate(st_.candidates7, st_.scan_seconds7),
                  (unsigned long long)st_.gpu_scans,
                  (unsigned long long)st_.cpu_scans);
      std::printf("Host phases: %llu nodes; step1/2 %.2fs, step3 %.2fs, "
                  "step4a %.2fs\n",
                  (unsigned long long)st_.nodes, st_.step12_seconds,
                  st_.step3_seconds, st_.step4a_seconds);
    }
  } catch (const std::exception& e) {
    std::fprintf(stderr, "%s\n", e.what());
    return 1;
  }
  return 0;
}
