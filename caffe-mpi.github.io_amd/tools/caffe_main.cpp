// caffe_main.cpp — the reference CLI surface (tools/caffe.cpp:28-46,:120):
//   caffe train -solver=... [-gpu=0|all] [-snapshot=...] [-weights=...]
//   caffe test  -model=... -weights=... [-gpu=0] [-iterations=50]
//   caffe time  -model=... [-gpu=0] [-iterations=50]
//   caffe device_query [-gpu=0]
// Single-process/single-GPU here ("all" selects device 0); multi-GPU
// training runs one process per GPU through bench.py / the C ABI
//  — the reference's in-process thread-per-GPU P2PManager is
// replaced by that launcher model.
#include <csignal>
#include <cstdio>
#include <cstring>
#include <map>
#include <string>

#include "../csrc/solver.hpp"

using namespace camd;

// ---- signal → SolverAction plumbing (reference util/signal_handler.cpp:
// handlers only set flags; Solver polls via the action-request callback
// once per iteration).  Defaults: -sigint_effect=stop -sighup_effect=snapshot
// (tools/caffe.cpp:31-36).
static volatile sig_atomic_t g_got_sigint = 0;
static volatile sig_atomic_t g_got_sighup = 0;
static SolverAction g_sigint_action = SolverAction::STOP;
static SolverAction g_sighup_action = SolverAction::SNAPSHOT;

static void handle_signal(int sig) {
  if (sig == SIGINT) g_got_sigint = 1;
  else if (sig == SIGHUP) g_got_sighup = 1;
}

static SolverAction parse_effect(const std::string& s, SolverAction dflt) {
  if (s == "stop") return SolverAction::STOP;
  if (s == "snapshot") return SolverAction::SNAPSHOT;
  if (s == "none") return SolverAction::NONE;
  if (!s.empty())
    fprintf(stderr, "unknown signal effect '%s' (stop|snapshot|none)\n",
            s.c_str());
  return dflt;
}

static SolverAction action_request() {
  if (g_got_sigint) {
    g_got_sigint = 0;
    return g_sigint_action;
  }
  if (g_got_sighup) {
    g_got_sighup = 0;
    return g_sighup_action;
  }
  return SolverAction::NONE;
}

static void install_signal_handlers(
    const std::map<std::string, std::string>& flags) {
  auto it = flags.find("sigint_effect");
  if (it != flags.end())
    g_sigint_action = parse_effect(it->second, SolverAction::STOP);
  it = flags.find("sighup_effect");
  if (it != flags.end())
    g_sighup_action = parse_effect(it->second, SolverAction::SNAPSHOT);
  struct sigaction sa;
  memset(&sa, 0, sizeof(sa));
  sa.sa_handler = handle_signal;
  sigaction(SIGINT, &sa, nullptr);
  sigaction(SIGHUP, &sa, nullptr);
}

static std::map<std::string, std::string> parse_flags(int argc, char** argv,
                                                      int start) {
  std::map<std::string, std::string> flags;
  for (int i = start; i < argc; ++i) {
    std::string a = argv[i];
    if (a.rfind("-", 0) != 0) continue;
    a = a.substr(a[1] == '-' ? 2 : 1);
    const auto eq = a.find('=');
    if (eq != std::string::npos) {
      flags[a.substr(0, eq)] = a.substr(eq + 1);
    } else if (i + 1 < argc && argv[i + 1][0] != '-') {
      flags[a] = argv[++i];
    } else {
      flags[a] = "";
    }
  }
  return flags;
}

static void setup_device(const std::map<std::string, std::string>& flags) {
  auto it = flags.find("gpu");
  if (it == flags.end()) {
    Engine::get().mode = Mode::CPU;
    return;
  }
  int dev = 0;
  if (it->second != "all" && !it->second.empty()) dev = atoi(it->second.c_str());
  Engine::get().set_mode_gpu(dev);
}

namespace camd {
extern int g_syn_shape[3];
}

int main(int argc, char** argv) {
  // synthetic data shape for dataset-less runs: CAFFE_SYN_SHAPE=CxHxW[xK]
  if (const char* ss = getenv("CAFFE_SYN_SHAPE")) {
    int c = 0, h = 0, w = 0, k = 0;
    if (sscanf(ss, "%dx%dx%dx%d", &c, &h, &w, &k) >= 3) {
      camd::g_syn_shape[0] = c;
      camd::g_syn_shape[1] = h;
      camd::g_syn_shape[2] = w;
      if (k > 0) Engine::get().syn_classes = k;
    }
  }
  if (argc < 2) {
    fprintf(stderr,
            "usage: caffe <train|test|time|device_query> [flags]\n");
    return 1;
  }
  const std::string cmd = argv[1];
  auto flags = parse_flags(argc, argv, 2);
  try {
    if (cmd == "device_query") {
      int n = 0;
      HIP_CHECK(hipGetDeviceCount(&n));
      for (int i = 0; i < n; ++i) {
        hipDeviceProp_t p;
        HIP_CHECK(hipGetDeviceProperties(&p, i));
        printf("Device %d: %s, %d CUs, %.1f GB, gcnArch %s\n", i, p.name,
               p.multiProcessorCount, p.totalGlobalMem / 1.073741824e9,
               p.gcnArchName);
      }
      return 0;
    }
    if (cmd == "train") {
      CHECK_(flags.count("solver")) << "train needs -solver";
      setup_device(flags);
      install_signal_handlers(flags);
      Solver solver(parse_prototxt_file(flags["solver"]));
      solver.set_action_request(&action_request);
      if (flags.count("snapshot") && !flags["snapshot"].empty())
        solver.Restore(flags["snapshot"]);
      else if (flags.count("weights") && !flags["weights"].empty())
        solver.LoadWeights(flags["weights"]);
      const long max_iter = solver.param()->inum("max_iter", 0);
      long todo = max_iter - solver.iter();
      if (flags.count("iterations"))
        todo = atol(flags["iterations"].c_str());
      CHECK_GT_(todo, 0);
      solver.Step((int)todo);
      solver.print_perf_report();
      if (solver.param()->boolean("snapshot_after_train", true))
        solver.Snapshot();
      if (solver.early_exit())
        fprintf(stderr, "Optimization stopped early.\n");
      fprintf(stderr, "Optimization Done.\n");
      return 0;
    }
    if (cmd == "test") {
      CHECK_(flags.count("model")) << "test needs -model";
      setup_device(flags);
      Net net(parse_prototxt_file(flags["model"]), Phase::TEST);
      if (flags.count("weights")) net.LoadWeights(flags["weights"]);
      const int iters = flags.count("iterations")
                            ? atoi(flags["iterations"].c_str())
                            : 50;
      std::map<std::string, double> scores;
      for (int i = 0; i < iters; ++i) {
        net.Forward();
        for (auto& kv : net.scores()) scores[kv.first] += kv.second;
      }
      for (auto& kv : scores)
        printf("%s = %g\n", kv.first.c_str(), kv.second / iters);
      return 0;
    }
    if (cmd == "time") {
      CHECK_(flags.count("model")) << "time needs -model";
      setup_device(flags);
      Engine::get().seed = 1371;  // tools/caffe.cpp:365 pins 1371
      Net net(parse_prototxt_file(flags["model"]), Phase::TRAIN);
      const int iters = flags.count("iterations")
                            ? atoi(flags["iterations"].c_str())
                            : 10;
      net.time_layers(iters);
      return 0;
    }
    fprintf(stderr, "unknown command %s\n", cmd.c_str());
    return 1;
  } catch (const std::exception& e) {
    fprintf(stderr, "FATAL: %s\n", e.what());
    return 1;
  }
}
