// epoll event loop — one loop per worker thread, SO_REUSEPORT listeners,
// non-blocking sockets, min-heap timers, eventfd cross-thread wakeup.
#pragma once

#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/epoll.h>
#include <sys/eventfd.h>
#include <sys/socket.h>
#include <unistd.h>

#include <atomic>
#include <functional>
#include <queue>
#include <vector>

#include "common.h"

namespace gw {

class EventLoop;

struct FdHandler {
  virtual ~FdHandler() = default;
  virtual void on_event(uint32_t events) = 0;
};

class EventLoop {
 public:
  EventLoop();
  ~EventLoop();

  void add(int fd, uint32_t events, FdHandler* h);
  void mod(int fd, uint32_t events, FdHandler* h);
  void del(int fd);

  using TimerFn = std::function<void()>;
  // returns timer id; fires once after delay_ms
  uint64_t add_timer(int64_t delay_ms, TimerFn fn);
  void cancel_timer(uint64_t id);

  // thread-safe: queue fn to run on this loop
  void post(std::function<void()> fn);

  void run();
  void stop() { running_.store(false, std::memory_order_relaxed); wakeup(); }
  void wakeup();

  // deferred deletion (safe while iterating epoll events)
  void defer_delete(std::function<void()> deleter) {
    garbage_.push_back(std::move(deleter));
  }

 private:
  struct Timer {
    int64_t when_us;
    uint64_t id;
    bool operator>(const Timer& o) const { return when_us > o.when_us; }
  };
  int epfd_ = -1;
  int wake_fd_ = -1;
  std::atomic<bool> running_{false};
  std::priority_queue<Timer, std::vector<Timer>, std::greater<Timer>> timers_;
  std::unordered_map<uint64_t, TimerFn> timer_fns_;
  uint64_t next_timer_id_ = 1;
  std::mutex post_mu_;
  std::vector<std::function<void()>> posted_;
  std::vector<std::function<void()>> garbage_;
  struct WakeHandler : FdHandler {
    EventLoop* loop;
    void on_event(uint32_t) override;
  } wake_handler_;
};

// ---- socket helpers ----
int make_listener(const std::string& host, int port, bool reuseport,
                  int backlog = 4096);
void set_nonblocking(int fd);
void set_nodelay(int fd);
// non-blocking connect; returns fd (in progress or connected), -1 on error
int start_connect(const std::string& host, int port);

}  // namespace gw
