// ensemble_main.cpp — zkensembled: standalone synthetic ZooKeeper ensemble.
//
// Runs the in-process ensemble (ensemble.hpp) as its own server process so
// daemon-level integration tests (and operators) can point registrard — or
// any real ZooKeeper client — at it over TCP. Prints the bound ports on
// stdout as a JSON line, then serves until SIGINT/SIGTERM.
#include <signal.h>
#include <unistd.h>

#include <atomic>
#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <thread>

#include "ensemble.hpp"

using namespace registrar;

namespace {
std::atomic<int> g_signal{0};
void on_signal(int sig) { g_signal.store(sig); }
}  // namespace

int main(int argc, char** argv) {
  zk::EnsembleConfig cfg;
  cfg.ports.clear();
  int nservers = 1;
  for (int i = 1; i < argc; i++) {
    if (!strcmp(argv[i], "-n") && i + 1 < argc) {
      nservers = atoi(argv[++i]);
    } else if (!strcmp(argv[i], "-p") && i + 1 < argc) {
      cfg.ports.push_back(atoi(argv[++i]));
    } else if (!strcmp(argv[i], "--tick-ms") && i + 1 < argc) {
      cfg.tick_ms = atoi(argv[++i]);
    } else if (!strcmp(argv[i], "--latency-ms") && i + 1 < argc) {
      cfg.latency_ms = atoi(argv[++i]);
    } else if (!strcmp(argv[i], "--election-ms") && i + 1 < argc) {
      cfg.election_ms = atoi(argv[++i]);
    } else if (!strcmp(argv[i], "-b") && i + 1 < argc) {
      cfg.bind_host = argv[++i];
    } else if (!strcmp(argv[i], "-v")) {
      cfg.log_level = LogLevel::Info;
    } else {
      fprintf(stderr,
              "usage: %s [-n NSERVERS] [-p PORT]... [-b BINDHOST] [--tick-ms MS] "
              "[--latency-ms MS] [--election-ms MS] [-v]\n",
              argv[0]);
      return 1;
    }
  }
  if (cfg.ports.empty()) cfg.ports.assign(static_cast<size_t>(nservers), 0);

  signal(SIGINT, on_signal);
  signal(SIGTERM, on_signal);
  signal(SIGPIPE, SIG_IGN);

  zk::Ensemble ens(cfg);
  ens.start();
  std::string out = "{\"ports\":[";
  bool first = true;
  for (int p : ens.ports()) {
    if (!first) out += ',';
    first = false;
    out += std::to_string(p);
  }
  out += "],\"connect\":\"" + ens.connect_string() + "\"}\n";
  fputs(out.c_str(), stdout);
  fflush(stdout);

  while (g_signal.load() == 0) std::this_thread::sleep_for(std::chrono::milliseconds(100));
  ens.stop();
  return 0;
}
