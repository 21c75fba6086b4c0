// loop.hpp — epoll event loop with monotonic timers and cross-thread post().
//
// The reference runs on Node's single-threaded event loop; the native
// equivalent here is one epoll loop per component
// (ZK client session, ensemble server, orchestrator), each owning all of its
// state on its loop thread. Cross-thread interaction goes through post(),
// which is the only thread-safe entry point.
#pragma once

#include <stdio.h>
#include <sys/epoll.h>
#include <sys/eventfd.h>
#include <time.h>
#include <unistd.h>

#include <atomic>
#include <cstdint>
#include <cstring>
#include <deque>
#include <functional>
#include <map>
#include <mutex>
#include <queue>
#include <stdexcept>
#include <thread>
#include <unordered_map>
#include <unordered_set>
#include <vector>

namespace registrar {

inline int64_t now_ms() {
  struct timespec ts;
  clock_gettime(CLOCK_MONOTONIC, &ts);
  return static_cast<int64_t>(ts.tv_sec) * 1000 + ts.tv_nsec / 1000000;
}

inline int64_t now_us() {
  struct timespec ts;
  clock_gettime(CLOCK_MONOTONIC, &ts);
  return static_cast<int64_t>(ts.tv_sec) * 1000000 + ts.tv_nsec / 1000;
}

class EventLoop {
 public:
  using FdCallback = std::function<void(uint32_t events)>;
  using TimerId = uint64_t;

  EventLoop() {
    epfd_ = epoll_create1(EPOLL_CLOEXEC);
    if (epfd_ < 0) throw std::runtime_error("epoll_create1 failed");
    wakeup_fd_ = eventfd(0, EFD_CLOEXEC | EFD_NONBLOCK);
    if (wakeup_fd_ < 0) throw std::runtime_error("eventfd failed");
    struct epoll_event ev;
    memset(&ev, 0, sizeof(ev));
    ev.events = EPOLLIN;
    ev.data.fd = wakeup_fd_;
    epoll_ctl(epfd_, EPOLL_CTL_ADD, wakeup_fd_, &ev);
  }

  ~EventLoop() {
    close(wakeup_fd_);
    close(epfd_);
  }

  EventLoop(const EventLoop&) = delete;
  EventLoop& operator=(const EventLoop&) = delete;

  // ---- fd registration (thread-safe: a connection may be registered on
  // this loop by an accepting thread that is not the loop thread) ----

  void add_fd(int fd, uint32_t events, FdCallback cb) {
    {
      std::lock_guard<std::mutex> g(fd_mu_);
      fd_cbs_[fd] = std::move(cb);  // callback installed before the fd can fire
    }
    struct epoll_event ev;
    memset(&ev, 0, sizeof(ev));
    ev.events = events;
    ev.data.fd = fd;
    if (epoll_ctl(epfd_, EPOLL_CTL_ADD, fd, &ev) < 0) {
      std::lock_guard<std::mutex> g(fd_mu_);
      fd_cbs_.erase(fd);
      throw std::runtime_error("epoll_ctl ADD failed");
    }
  }

  // Error-tolerant: the fd may already be closed/removed by another thread's
  // shutdown path; epoll_ctl itself is thread-safe.
  void mod_fd(int fd, uint32_t events) {
    struct epoll_event ev;
    memset(&ev, 0, sizeof(ev));
    ev.events = events;
    ev.data.fd = fd;
    epoll_ctl(epfd_, EPOLL_CTL_MOD, fd, &ev);
  }

  void del_fd(int fd) {
    epoll_ctl(epfd_, EPOLL_CTL_DEL, fd, nullptr);
    std::lock_guard<std::mutex> g(fd_mu_);
    fd_cbs_.erase(fd);
  }

  TimerId schedule(int64_t delay_ms, std::function<void()> fn) {
    TimerId id = next_timer_id_++;
    int64_t deadline = now_ms() + (delay_ms < 0 ? 0 : delay_ms);
    timers_[id] = std::move(fn);
    heap_.push({deadline, id});
    return id;
  }

  void cancel(TimerId id) { timers_.erase(id); }

  bool running() const { return running_.load(std::memory_order_acquire); }

  // ---- thread-safe API ----

  void post(std::function<void()> fn) {
    {
      std::lock_guard<std::mutex> g(post_mu_);
      posted_.push_back(std::move(fn));
    }
    wakeup();
  }

  void stop() {
    post([this] { running_.store(false, std::memory_order_release); });
  }

  // Timer arming from any thread (the heap itself is loop-thread-only).
  void schedule_from_any(int64_t delay_ms, std::function<void()> fn) {
    if (on_loop_thread()) {
      schedule(delay_ms, std::move(fn));
    } else {
      post([this, delay_ms, fn = std::move(fn)]() mutable { schedule(delay_ms, std::move(fn)); });
    }
  }

  // Run the loop until stop(). Call from the owning thread.
  void run() {
    running_.store(true, std::memory_order_release);
    loop_tid_.store(std::this_thread::get_id(), std::memory_order_release);
    std::vector<struct epoll_event> events(64);
    while (running_.load(std::memory_order_relaxed)) {
      int timeout = next_timeout_ms();
      int n = epoll_wait(epfd_, events.data(), static_cast<int>(events.size()), timeout);
      if (n < 0) {
        if (errno == EINTR) continue;
        throw std::runtime_error("epoll_wait failed");
      }
      for (int i = 0; i < n && running_.load(std::memory_order_relaxed); i++) {
        int fd = events[i].data.fd;
        if (fd == wakeup_fd_) {
          uint64_t v;
          while (read(wakeup_fd_, &v, sizeof(v)) > 0) {
          }
          drain_posted();
          continue;
        }
        FdCallback cb;
        {
          std::lock_guard<std::mutex> g(fd_mu_);
          auto it = fd_cbs_.find(fd);
          if (it != fd_cbs_.end()) cb = it->second;  // copy: cb may del_fd(fd)
        }
        if (cb) guarded([&] { cb(events[i].events); });
      }
      fire_timers();
      if (n == static_cast<int>(events.size())) events.resize(events.size() * 2);
    }
    drain_posted();
  }

  bool on_loop_thread() const {
    return std::this_thread::get_id() == loop_tid_.load(std::memory_order_acquire);
  }

 private:
  void wakeup() {
    uint64_t one = 1;
    ssize_t r = write(wakeup_fd_, &one, sizeof(one));
    (void)r;
  }

  // A throwing callback must not take down the loop thread (and with it the
  // daemon): log to stderr and keep serving. Logic errors still surface in
  // tests via the message.
  template <typename Fn>
  void guarded(Fn&& fn) {
    try {
      fn();
    } catch (const std::exception& e) {
      fprintf(stderr, "[eventloop] callback threw: %s\n", e.what());
    } catch (...) {
      fprintf(stderr, "[eventloop] callback threw (non-std exception)\n");
    }
  }

  void drain_posted() {
    std::deque<std::function<void()>> q;
    {
      std::lock_guard<std::mutex> g(post_mu_);
      q.swap(posted_);
    }
    for (auto& fn : q) guarded(fn);
  }

  int next_timeout_ms() {
    prune_heap();
    if (heap_.empty()) return 1000;
    int64_t dt = heap_.top().deadline - now_ms();
    if (dt < 0) dt = 0;
    if (dt > 60000) dt = 60000;
    return static_cast<int>(dt);
  }

  void prune_heap() {
    while (!heap_.empty() && timers_.find(heap_.top().id) == timers_.end()) heap_.pop();
  }

  void fire_timers() {
    int64_t now = now_ms();
    while (running_.load(std::memory_order_relaxed)) {
      prune_heap();
      if (heap_.empty() || heap_.top().deadline > now) break;
      TimerId id = heap_.top().id;
      heap_.pop();
      auto it = timers_.find(id);
      if (it == timers_.end()) continue;
      auto fn = std::move(it->second);
      timers_.erase(it);
      guarded(fn);
    }
  }

  struct HeapEntry {
    int64_t deadline;
    TimerId id;
    // FIFO among equal deadlines (ids are monotonic) so delayed responses
    // keep their submission order (latency-injection correctness)
    bool operator>(const HeapEntry& o) const {
      return deadline != o.deadline ? deadline > o.deadline : id > o.id;
    }
  };

  int epfd_ = -1;
  int wakeup_fd_ = -1;
  std::atomic<bool> running_{false};
  std::atomic<std::thread::id> loop_tid_{};
  std::mutex fd_mu_;
  std::unordered_map<int, FdCallback> fd_cbs_;
  std::unordered_map<TimerId, std::function<void()>> timers_;
  std::priority_queue<HeapEntry, std::vector<HeapEntry>, std::greater<HeapEntry>> heap_;
  TimerId next_timer_id_ = 1;
  std::mutex post_mu_;
  std::deque<std::function<void()>> posted_;
};

// Exponential backoff policy (reference parameters: lib/zk.js:38-42 — heartbeat
// retry 1 s → 30 s cap, ≤5 attempts; lib/zk.js:97-101 — connect retry 1 s → 90 s,
// infinite attempts).
struct Backoff {
  int64_t initial_ms = 1000;
  int64_t max_ms = 30000;
  int64_t max_attempts = 5;  // <0 = infinite
  int64_t attempt = 0;

  // delay before attempt N (first retry waits initial_ms)
  int64_t next_delay() {
    int64_t d = initial_ms;
    for (int64_t i = 0; i < attempt && d < max_ms; i++) d *= 2;
    if (d > max_ms) d = max_ms;
    attempt++;
    return d;
  }

  bool exhausted() const { return max_attempts >= 0 && attempt >= max_attempts; }
  void reset() { attempt = 0; }
};

}  // namespace registrar
