// stress_main.cpp — native concurrency stress driver for sanitizer builds.
//
// The reference is single-threaded JS and needed no sanitizers; this build's
// core is heavily multi-threaded (IO-loop pool, sharded state, cross-thread
// watch delivery), so TSan/ASan passes run this driver in CI
// (make tsan / make asan):
//   - an in-process 3-server ensemble,
//   - N concurrent registrar clients doing register → heartbeat → unregister
//     cycles with watches armed,
//   - a chaos thread expiring random sessions and kill/restarting servers.
// Exit code 0 = every cycle behaved; sanitizers report races/leaks.
#include <dirent.h>
#include <execinfo.h>
#include <signal.h>
#include <sys/syscall.h>
#include <unistd.h>

#include <atomic>
#include <chrono>
#include <cstdio>
#include <cstring>
#include <thread>
#include <vector>

#include "ensemble.hpp"
#include "registrar.hpp"
#include "zkclient.hpp"

using namespace registrar;

namespace {
// Hang diagnosis: STRESS_WATCHDOG=<sec> arms a watchdog that, if the run
// overshoots, SIGUSR1s every thread (whose handler prints a backtrace) and
// aborts. Diagnostic aid; off by default.
void dump_backtrace(int) {
  void* frames[48];
  int n = backtrace(frames, 48);
  char hdr[64];
  int len = snprintf(hdr, sizeof(hdr), "--- thread %ld backtrace ---\n", syscall(SYS_gettid));
  ssize_t r = write(2, hdr, len);
  (void)r;
  backtrace_symbols_fd(frames, n, 2);
}

void arm_watchdog(int seconds) {
  struct sigaction sa;
  memset(&sa, 0, sizeof(sa));
  sa.sa_handler = dump_backtrace;
  sigaction(SIGUSR1, &sa, nullptr);
  std::thread([seconds] {
    std::this_thread::sleep_for(std::chrono::seconds(seconds));
    fprintf(stderr, "WATCHDOG: run overshot %ds; dumping all threads\n", seconds);
    DIR* d = opendir("/proc/self/task");
    if (d) {
      while (struct dirent* e = readdir(d)) {
        if (e->d_name[0] == '.') continue;
        long tid = atol(e->d_name);
        syscall(SYS_tgkill, getpid(), tid, SIGUSR1);
        std::this_thread::sleep_for(std::chrono::milliseconds(150));
      }
      closedir(d);
    }
    std::this_thread::sleep_for(std::chrono::seconds(1));
    _exit(42);
  }).detach();
}
}  // namespace

int main(int argc, char** argv) {
  signal(SIGPIPE, SIG_IGN);  // belt: library writes already use MSG_NOSIGNAL
  int nclients = 4;
  int seconds = 8;
  if (const char* wd = getenv("STRESS_WATCHDOG")) arm_watchdog(atoi(wd));
  for (int i = 1; i < argc; i++) {
    if (!strcmp(argv[i], "-c") && i + 1 < argc) nclients = atoi(argv[++i]);
    if (!strcmp(argv[i], "-t") && i + 1 < argc) seconds = atoi(argv[++i]);
  }

  Logger log("stress");
  log.set_level(LogLevel::Error);

  zk::EnsembleConfig ecfg;
  ecfg.ports = {0, 0, 0};
  ecfg.tick_ms = 50;
  ecfg.min_session_timeout_ms = 300;
  ecfg.log_level = LogLevel::Error;
  zk::Ensemble ens(ecfg);
  ens.start();

  std::vector<std::pair<std::string, int>> servers;
  for (int p : ens.ports()) servers.push_back({"127.0.0.1", p});

  std::atomic<bool> stop{false};
  std::atomic<uint64_t> cycles{0};
  std::atomic<uint64_t> failures{0};

  auto client_fn = [&](int idx) {
    while (!stop.load()) {
      zk::ZkClientConfig ccfg;
      for (auto& s : servers) ccfg.servers.push_back({s.first, s.second});
      ccfg.session_timeout_ms = 2000;
      ccfg.connect_timeout_ms = 1000;
      ccfg.connect_initial_delay_ms = 20;
      ccfg.connect_max_delay_ms = 100;
      ccfg.log_level = LogLevel::Fatal;
      zk::ZkClient client(std::move(ccfg), log);
      client.start();
      if (!client.wait_connected(5000)) continue;

      RegistrationConfig reg;
      reg.domain = "c" + std::to_string(idx) + ".stress.test";
      reg.type = "host";
      reg.admin_ip = "127.0.0.1";
      reg.hostname = "h" + std::to_string(idx);
      reg.settle_ms = 0;
      for (int a = 0; a < 20; a++) reg.aliases.push_back("a" + std::to_string(a) + ".c" + std::to_string(idx) + ".stress.test");

      while (!stop.load() && client.state() == zk::SessionState::Connected) {
        RegisterResult res = register_node(client, reg, log);
        if (res.rc != zk::kZOk) {
          failures.fetch_add(1);
          break;
        }
        // arm some watches like a Binder reader would
        client.exists(res.znodes[0], nullptr, true);
        std::vector<std::string> ch;
        client.get_children(domain_to_path(reg.domain), &ch, true);
        // exercise multi under chaos too: atomic delete+create of one node
        {
          std::vector<zk::ZkClient::MixedOp> mops;
          zk::ZkClient::MixedOp d;
          d.op = zk::kOpDelete;
          d.path = res.znodes[0];
          mops.push_back(d);
          zk::ZkClient::MixedOp cr;
          cr.op = zk::kOpCreate;
          cr.path = res.znodes[0];
          cr.data = "swap";
          cr.flags = zk::kEphemeral;
          mops.push_back(cr);
          client.multi(mops, nullptr);  // rc may be conn-loss under chaos
        }
        zk::RetryPolicy rp;
        rp.max_attempts = 1;
        rp.initial_delay_ms = 10;
        int rc = client.heartbeat(res.znodes, rp, nullptr);
        if (rc != zk::kZOk && rc != zk::kZConnectionLoss && rc != zk::kZSessionExpired) failures.fetch_add(1);
        unregister_node(client, res.znodes, log);
        cycles.fetch_add(1);
      }
      client.close();
    }
  };

  auto chaos_fn = [&] {
    uint64_t rng = 0x12345678;
    auto next = [&rng] {
      rng ^= rng << 13;
      rng ^= rng >> 7;
      rng ^= rng << 17;
      return rng;
    };
    while (!stop.load()) {
      std::this_thread::sleep_for(std::chrono::milliseconds(150 + next() % 250));
      switch (next() % 4) {
        case 0: {
          auto sids = ens.session_ids();
          if (!sids.empty()) ens.expire_session(sids[next() % sids.size()]);
          break;
        }
        case 1: {
          size_t idx = next() % 3;
          if (ens.server_up(idx)) {
            // keep at least one server up
            int up = 0;
            for (size_t i = 0; i < 3; i++) up += ens.server_up(i) ? 1 : 0;
            if (up > 1) ens.kill_server(idx);
          }
          break;
        }
        case 2: {
          for (size_t i = 0; i < 3; i++)
            if (!ens.server_up(i)) ens.restart_server(i);
          break;
        }
        default:
          ens.set_latency_ms(static_cast<int>(next() % 3));
          break;
      }
    }
    ens.set_latency_ms(0);
    for (size_t i = 0; i < 3; i++)
      if (!ens.server_up(i)) ens.restart_server(i);
  };

  std::vector<std::thread> threads;
  for (int i = 0; i < nclients; i++) threads.emplace_back(client_fn, i);
  std::thread chaos(chaos_fn);

  std::this_thread::sleep_for(std::chrono::seconds(seconds));
  stop.store(true);
  for (auto& t : threads) t.join();
  chaos.join();
  ens.stop();

  printf("stress: %llu cycles, %llu hard failures\n", static_cast<unsigned long long>(cycles.load()),
         static_cast<unsigned long long>(failures.load()));
  // hard failures are register errors while connected — tolerate a few from
  // chaos timing, fail on systematic breakage
  return (cycles.load() > 0 && failures.load() < cycles.load() / 4 + 16) ? 0 : 1;
}
