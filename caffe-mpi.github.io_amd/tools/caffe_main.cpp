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
#include <sys/stat.h>
#include <sys/wait.h>
#include <unistd.h>

#include <map>
#include <string>
#include <vector>

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

// ---- multi-GPU train (`-gpu=all` / `-gpu=0,1,...`): the reference's
// in-process thread-per-GPU P2PManager (parallel.cpp) is replaced by the
// MI355X-native one-process-per-GPU model — the parent forks one worker per
// device BEFORE any HIP call (HIP init + fork is unsafe), the RCCL unique id
// rendezvous is a file in a private tmp dir (the reference used MPI_Bcast;
// single-node needs neither MPI nor a network), and the parent prints the
// reference's overall line (parallel.cpp:85) from per-rank perf files.

// device count probed in a throwaway child so the parent never inits HIP
static int device_count_scout() {
  int fds[2];
  if (pipe(fds) != 0) return 0;
  const pid_t pid = fork();
  if (pid == 0) {
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) n = 0;
    (void)!write(fds[1], &n, sizeof(n));
    _exit(0);
  }
  close(fds[1]);
  int n = 0;
  if (read(fds[0], &n, sizeof(n)) != sizeof(n)) n = 0;
  close(fds[0]);
  waitpid(pid, nullptr, 0);
  return n;
}

// total train-phase Data batch from the net message — the reference splits
// the prototxt batch across solvers (parallel.cpp:284-348), remainder to
// the first ranks
static long train_batch_size(const PMsgPtr& sp) {
  PMsgPtr net_msg;
  if (sp->has("net"))
    net_msg = parse_prototxt_file(sp->str("net"));
  else
    net_msg = sp->sub("net_param");
  if (!net_msg) return 0;
  for (auto& lm : net_msg->subs("layer")) {
    bool test_only = false;
    for (auto& inc : lm->subs("include"))
      if (inc->str("phase") == "TEST") test_only = true;
    if (test_only || lm->str("type") != "Data") continue;
    if (auto dp = lm->sub("data_param")) return dp->inum("batch_size", 0);
    return 0;
  }
  return 0;
}

// multi-node configuration from the environment (the reference's
// Clusters shim, clusters.cpp: node_rank/node_count + MPI_Bcast of the
// ncclUniqueId; here the id travels over TCP — csrc/bootstrap.cpp):
// CAFFE_NNODES, CAFFE_NODE_RANK, MASTER_ADDR, MASTER_PORT.
struct NodeEnv {
  int nnodes = 1, node_rank = 0, port = 29500;
  std::string master = "127.0.0.1";
};
static NodeEnv node_env() {
  NodeEnv e;
  if (const char* v = getenv("CAFFE_NNODES")) e.nnodes = atoi(v);
  if (const char* v = getenv("CAFFE_NODE_RANK")) e.node_rank = atoi(v);
  if (const char* v = getenv("MASTER_ADDR")) e.master = v;
  if (const char* v = getenv("MASTER_PORT")) e.port = atoi(v);
  CHECK_GT_(e.nnodes, 0);
  CHECK_LT_(e.node_rank, e.nnodes);
  return e;
}

static int run_train_rank(std::map<std::string, std::string> flags, int dev,
                          int rank, int world,
                          const std::string& rdv_dir) {
  Engine& E = Engine::get();
  if (dev >= 0)
    E.set_mode_gpu(dev);
  else
    E.mode = Mode::CPU;
  const NodeEnv ne = node_env();
  // ONE communicator spans every GPU of every node: global rank =
  // node_rank * local_world + local_rank (parallel.cpp:166-169)
  const int local_rank = rank, local_world = world;
  const int grank = ne.node_rank * local_world + local_rank;
  const int gworld = ne.nnodes * local_world;
  E.rank = grank;  // fillers + synthetic/LMDB sharding use seed+rank
  E.world = gworld;
  install_signal_handlers(flags);
  auto sp = parse_prototxt_file(flags["solver"]);
  int batch_override = 0;
  if (local_world > 1) {
    // the reference divides the prototxt batch among THIS node's solvers
    // (parallel.cpp:284-348); multi-node multiplies the effective batch
    const long total = train_batch_size(sp);
    if (total > 0)
      batch_override = (int)(total / local_world +
                             (local_rank < total % local_world ? 1 : 0));
  }
  Solver solver(sp, batch_override);
  solver.set_action_request(&action_request);
  if (grank != 0) solver.set_snapshot_enabled(false);
  if (gworld > 1) {
    uint8_t uid[128];
    if (ne.nnodes > 1) {
      // TCP bootstrap: global rank 0 creates + serves the id; every
      // other rank (any node) fetches from MASTER_ADDR:MASTER_PORT
      if (grank == 0) {
        rccl_unique_id(uid);
        CHECK_EQ_(uid_serve(uid, sizeof(uid), ne.port, gworld - 1), 0)
            << "uid bootstrap serve failed on port " << ne.port;
      } else {
        CHECK_EQ_(uid_fetch(uid, sizeof(uid), ne.master.c_str(), ne.port,
                            120),
                  0)
            << "uid bootstrap fetch from " << ne.master << ":" << ne.port
            << " failed";
      }
    } else {
      const std::string uid_path = rdv_dir + "/rccl_uid";
      if (rank == 0) {
        rccl_unique_id(uid);
        const std::string tmp = uid_path + ".tmp";
        FILE* f = fopen(tmp.c_str(), "wb");
        CHECK_(f) << "cannot write " << tmp;
        fwrite(uid, 1, sizeof(uid), f);
        fclose(f);
        CHECK_EQ_(rename(tmp.c_str(), uid_path.c_str()), 0);
      } else {
        bool got = false;
        for (int i = 0; i < 6000 && !got; ++i) {  // up to 60 s
          if (FILE* f = fopen(uid_path.c_str(), "rb")) {
            got = fread(uid, 1, sizeof(uid), f) == sizeof(uid);
            fclose(f);
          }
          if (!got) usleep(10000);
        }
        CHECK_(got) << "rank " << rank << ": no RCCL uid rendezvous";
      }
    }
    solver.set_comm(make_rccl_comm(grank, gworld, uid));
    solver.bcast_weights();  // initial weight bcast, parallel.cpp:208-227
  }
  if (flags.count("snapshot") && !flags["snapshot"].empty())
    solver.Restore(flags["snapshot"]);
  else if (flags.count("weights") && !flags["weights"].empty())
    solver.LoadWeights(flags["weights"]);
  const long max_iter = solver.param()->inum("max_iter", 0);
  long todo = max_iter - solver.iter();
  if (flags.count("iterations")) todo = atol(flags["iterations"].c_str());
  CHECK_GT_(todo, 0);
  // perf accounting is Step-call-granular (first call untimed): warm up
  // with 2 iters then time the rest — the reference's skip-iters-0..1
  // rule (solver.cpp:299)
  const long warm = std::min<long>(2, todo);
  solver.Step((int)warm);
  if (todo > warm) solver.Step((int)(todo - warm));
  solver.print_perf_report();
  if (!rdv_dir.empty()) {
    const std::string pf = rdv_dir + "/perf_" + std::to_string(rank);
    if (FILE* f = fopen(pf.c_str(), "w")) {
      fprintf(f, "%.17g\n", solver.perf_img_per_sec());
      fclose(f);
    }
  }
  if (E.rank == 0 &&
      solver.param()->boolean("snapshot_after_train", true))
    solver.Snapshot();
  if (solver.early_exit())
    fprintf(stderr, "Optimization stopped early.\n");
  if (E.rank == 0) fprintf(stderr, "Optimization Done.\n");
  return 0;
}

static int run_multi_gpu_train(
    const std::map<std::string, std::string>& flags,
    const std::vector<int>& devs) {
  char tmpl[] = "/tmp/caffe_amd_rdv_XXXXXX";
  CHECK_(mkdtemp(tmpl)) << "mkdtemp failed";
  const std::string rdv_dir = tmpl;
  // children own SIGINT/SIGHUP (delivered group-wide by the terminal);
  // the parent only waits
  signal(SIGINT, SIG_IGN);
  signal(SIGHUP, SIG_IGN);
  const int world = (int)devs.size();
  std::vector<pid_t> pids;
  for (int rank = 0; rank < world; ++rank) {
    const pid_t pid = fork();
    CHECK_GE_(pid, 0) << "fork failed";
    if (pid == 0) {
      int rc = 1;
      try {
        rc = run_train_rank(flags, devs[rank], rank, world, rdv_dir);
      } catch (const std::exception& e) {
        fprintf(stderr, "FATAL (rank %d): %s\n", rank, e.what());
      }
      _exit(rc);
    }
    pids.push_back(pid);
  }
  int rc = 0;
  for (pid_t pid : pids) {
    int status = 0;
    waitpid(pid, &status, 0);
    if (!WIFEXITED(status) || WEXITSTATUS(status) != 0) rc = 1;
  }
  if (rc == 0) {
    double total = 0;
    bool any = false;
    for (int rank = 0; rank < world; ++rank) {
      const std::string pf = rdv_dir + "/perf_" + std::to_string(rank);
      if (FILE* f = fopen(pf.c_str(), "r")) {
        double v = 0;
        if (fscanf(f, "%lg", &v) == 1 && v > 0) {
          total += v;
          any = true;
        }
        fclose(f);
      }
    }
    if (any)
      fprintf(stderr, "Overall multi-GPU performance: %.1f img/sec\n",
              total);
  }
  for (int rank = 0; rank < world; ++rank)
    unlink((rdv_dir + "/perf_" + std::to_string(rank)).c_str());
  unlink((rdv_dir + "/rccl_uid").c_str());
  rmdir(rdv_dir.c_str());
  return rc;
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
      const std::string gpuflag =
          flags.count("gpu") ? flags["gpu"] : std::string();
      if (gpuflag == "all" || gpuflag.find(',') != std::string::npos) {
        std::vector<int> devs;
        if (gpuflag == "all") {
          const int n = device_count_scout();
          CHECK_GT_(n, 0) << "-gpu=all but no HIP device visible";
          for (int i = 0; i < n; ++i) devs.push_back(i);
        } else {
          size_t pos = 0;
          while (pos < gpuflag.size()) {
            size_t next = gpuflag.find(',', pos);
            if (next == std::string::npos) next = gpuflag.size();
            devs.push_back(atoi(gpuflag.substr(pos, next - pos).c_str()));
            pos = next + 1;
          }
        }
        return run_multi_gpu_train(flags, devs);
      }
      // `-gpu` with no value selects device 0 (old setup_device rule).
      // Without the flag the solver file decides: solver_mode GPU (the
      // reference default, caffe.proto SolverParameter) + device_id —
      // tools/caffe.cpp:154 honors solver_param.solver_mode the same way
      int dev = -1;
      if (flags.count("gpu")) {
        dev = atoi(gpuflag.c_str());
      } else {
        auto sp = parse_prototxt_file(flags["solver"]);
        if (sp->str("solver_mode", "CPU") == "GPU")
          dev = (int)sp->inum("device_id", 0);
      }
      return run_train_rank(flags, dev, 0, 1, std::string());
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
        // advance the data-stream cursor per iteration (successive test
        // batches; with LMDB this walks the set from record 0)
        Engine::get().data_iter = (uint64_t)i;
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
