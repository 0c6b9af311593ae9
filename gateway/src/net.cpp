#include "net.h"

#include <arpa/inet.h>
#include <fcntl.h>
#include <netdb.h>

#include <cstring>
#include <unordered_map>

namespace gw {

EventLoop::EventLoop() {
  epfd_ = epoll_create1(EPOLL_CLOEXEC);
  wake_fd_ = eventfd(0, EFD_NONBLOCK | EFD_CLOEXEC);
  wake_handler_.loop = this;
  add(wake_fd_, EPOLLIN, &wake_handler_);
}

EventLoop::~EventLoop() {
  if (wake_fd_ >= 0) close(wake_fd_);
  if (epfd_ >= 0) close(epfd_);
}

void EventLoop::WakeHandler::on_event(uint32_t) {
  uint64_t v;
  while (read(loop->wake_fd_, &v, 8) == 8) {}
  std::vector<std::function<void()>> fns;
  {
    std::lock_guard<std::mutex> lk(loop->post_mu_);
    fns.swap(loop->posted_);
  }
  for (auto& fn : fns) fn();
}

void EventLoop::add(int fd, uint32_t events, FdHandler* h) {
  epoll_event ev{};
  ev.events = events;
  ev.data.ptr = h;
  epoll_ctl(epfd_, EPOLL_CTL_ADD, fd, &ev);
}
void EventLoop::mod(int fd, uint32_t events, FdHandler* h) {
  epoll_event ev{};
  ev.events = events;
  ev.data.ptr = h;
  epoll_ctl(epfd_, EPOLL_CTL_MOD, fd, &ev);
}
void EventLoop::del(int fd) { epoll_ctl(epfd_, EPOLL_CTL_DEL, fd, nullptr); }

uint64_t EventLoop::add_timer(int64_t delay_ms, TimerFn fn) {
  uint64_t id = next_timer_id_++;
  timers_.push({now_us() + delay_ms * 1000, id});
  timer_fns_[id] = std::move(fn);
  return id;
}
void EventLoop::cancel_timer(uint64_t id) { timer_fns_.erase(id); }

void EventLoop::post(std::function<void()> fn) {
  {
    std::lock_guard<std::mutex> lk(post_mu_);
    posted_.push_back(std::move(fn));
  }
  wakeup();
}

void EventLoop::wakeup() {
  uint64_t one = 1;
  ssize_t n = write(wake_fd_, &one, 8);
  (void)n;
}

void EventLoop::run() {
  running_.store(true, std::memory_order_relaxed);
  epoll_event events[256];
  while (running_.load(std::memory_order_relaxed)) {
    int timeout = -1;
    int64_t now = now_us();
    while (!timers_.empty()) {
      const Timer& t = timers_.top();
      auto it = timer_fns_.find(t.id);
      if (it == timer_fns_.end()) {  // cancelled
        timers_.pop();
        continue;
      }
      if (t.when_us <= now) {
        TimerFn fn = std::move(it->second);
        timer_fns_.erase(it);
        timers_.pop();
        fn();
        now = now_us();
        continue;
      }
      timeout = (int)((t.when_us - now) / 1000) + 1;
      break;
    }
    int n = epoll_wait(epfd_, events, 256, timeout);
    for (int i = 0; i < n; ++i) {
      auto* h = (FdHandler*)events[i].data.ptr;
      if (h) h->on_event(events[i].events);
    }
    if (!garbage_.empty()) {
      for (auto& g : garbage_) g();
      garbage_.clear();
    }
  }
}

void set_nonblocking(int fd) {
  int fl = fcntl(fd, F_GETFL, 0);
  fcntl(fd, F_SETFL, fl | O_NONBLOCK);
}

void set_nodelay(int fd) {
  int one = 1;
  setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof one);
}

int make_listener(const std::string& host, int port, bool reuseport,
                  int backlog) {
  int fd = socket(AF_INET, SOCK_STREAM | SOCK_NONBLOCK | SOCK_CLOEXEC, 0);
  if (fd < 0) return -1;
  int one = 1;
  setsockopt(fd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof one);
  if (reuseport) setsockopt(fd, SOL_SOCKET, SO_REUSEPORT, &one, sizeof one);
  sockaddr_in addr{};
  addr.sin_family = AF_INET;
  addr.sin_port = htons((uint16_t)port);
  if (host.empty() || host == "0.0.0.0") {
    addr.sin_addr.s_addr = INADDR_ANY;
  } else if (inet_pton(AF_INET, host.c_str(), &addr.sin_addr) != 1) {
    close(fd);
    return -1;
  }
  if (bind(fd, (sockaddr*)&addr, sizeof addr) != 0) {
    close(fd);
    return -1;
  }
  if (listen(fd, backlog) != 0) {
    close(fd);
    return -1;
  }
  return fd;
}

// tiny resolver cache (health checker + upstream connects)
static std::mutex g_dns_mu;
static std::unordered_map<std::string, uint32_t> g_dns_cache;

static bool resolve_ipv4(const std::string& host, uint32_t* out) {
  {
    std::lock_guard<std::mutex> lk(g_dns_mu);
    auto it = g_dns_cache.find(host);
    if (it != g_dns_cache.end()) {
      *out = it->second;
      return true;
    }
  }
  in_addr a{};
  if (inet_pton(AF_INET, host.c_str(), &a) == 1) {
    *out = a.s_addr;
  } else if (host == "localhost") {
    *out = htonl(INADDR_LOOPBACK);
  } else {
    addrinfo hints{};
    hints.ai_family = AF_INET;
    hints.ai_socktype = SOCK_STREAM;
    addrinfo* res = nullptr;
    if (getaddrinfo(host.c_str(), nullptr, &hints, &res) != 0 || !res)
      return false;
    *out = ((sockaddr_in*)res->ai_addr)->sin_addr.s_addr;
    freeaddrinfo(res);
  }
  std::lock_guard<std::mutex> lk(g_dns_mu);
  g_dns_cache[host] = *out;
  return true;
}

int start_connect(const std::string& host, int port) {
  uint32_t ip;
  if (!resolve_ipv4(host, &ip)) return -1;
  int fd = socket(AF_INET, SOCK_STREAM | SOCK_NONBLOCK | SOCK_CLOEXEC, 0);
  if (fd < 0) return -1;
  set_nodelay(fd);
  sockaddr_in addr{};
  addr.sin_family = AF_INET;
  addr.sin_port = htons((uint16_t)port);
  addr.sin_addr.s_addr = ip;
  int rc = connect(fd, (sockaddr*)&addr, sizeof addr);
  if (rc != 0 && errno != EINPROGRESS) {
    close(fd);
    return -1;
  }
  return fd;
}

}  // namespace gw
