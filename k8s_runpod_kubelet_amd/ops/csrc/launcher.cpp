// Native pod-process launcher + event loop.
//
// MI355X-native replacement for the reference's deploy/terminate/status REST
// surface (reference pkg/virtual_kubelet/runpod_client.go:522-634 DeployPodREST,
// :711-739 TerminatePod, :772-818 GetDetailedPodStatus): instead of POSTing a
// pod spec to a cloud API and polling GET pods/{id}, we fork/exec the workload
// locally with its GPU binding environment (ROCR_VISIBLE_DEVICES), place it in
// a cgroup, and get *pushed* lifecycle events via pidfd+epoll — readiness from
// an AMDVK_READY_FD pipe the workload writes once its GPU context is up, exit
// via pidfd. Status latency is therefore bounded by the workload itself, not
// by a 10 s poll tick (reference kubelet.go:719).

#include "launcher.h"

#include <fcntl.h>
#include <signal.h>
#include <string.h>
#include <sys/epoll.h>
#include <sys/eventfd.h>
#include <sys/stat.h>
#include <sys/syscall.h>
#include <sys/types.h>
#include <sys/wait.h>
#include <unistd.h>

#include <cerrno>
#include <cstdio>
#include <fstream>
#include <stdexcept>

namespace amdvk {

namespace {

int PidfdOpen(pid_t pid) { return static_cast<int>(syscall(SYS_pidfd_open, pid, 0)); }

void WriteAll(int fd, const char* buf, size_t n) {
  while (n > 0) {
    ssize_t w = write(fd, buf, n);
    if (w <= 0) {
      if (errno == EINTR) continue;
      return;
    }
    buf += w;
    n -= static_cast<size_t>(w);
  }
}

bool WriteFileString(const std::string& path, const std::string& value) {
  int fd = open(path.c_str(), O_WRONLY | O_CLOEXEC);
  if (fd < 0) return false;
  ssize_t w = write(fd, value.c_str(), value.size());
  close(fd);
  return w == static_cast<ssize_t>(value.size());
}

bool MkdirP(const std::string& path) {
  std::string cur;
  for (size_t i = 0; i < path.size();) {
    size_t j = path.find('/', i + 1);
    if (j == std::string::npos) j = path.size();
    cur = path.substr(0, j);
    if (!cur.empty() && mkdir(cur.c_str(), 0755) != 0 && errno != EEXIST) return false;
    i = j;
  }
  return true;
}

}  // namespace

LaunchResult LaunchProcess(const LaunchSpec& spec) {
  LaunchResult res;
  if (spec.argv.empty()) {
    res.error = "empty argv";
    return res;
  }

  int ready_pipe[2] = {-1, -1};
  if (spec.ready_pipe) {
    if (pipe2(ready_pipe, O_CLOEXEC) != 0) {
      res.error = std::string("pipe2: ") + strerror(errno);
      return res;
    }
  }
  // Error-reporting pipe: child writes errno + message if exec fails.
  int err_pipe[2];
  if (pipe2(err_pipe, O_CLOEXEC) != 0) {
    res.error = std::string("pipe2: ") + strerror(errno);
    if (ready_pipe[0] >= 0) { close(ready_pipe[0]); close(ready_pipe[1]); }
    return res;
  }

  std::vector<char*> argv;
  argv.reserve(spec.argv.size() + 1);
  for (const auto& a : spec.argv) argv.push_back(const_cast<char*>(a.c_str()));
  argv.push_back(nullptr);

  std::vector<std::string> env_store = spec.env;
  if (spec.ready_pipe)
    env_store.push_back("AMDVK_READY_FD=" + std::to_string(ready_pipe[1]));
  std::vector<char*> envp;
  envp.reserve(env_store.size() + 1);
  for (auto& e : env_store) envp.push_back(const_cast<char*>(e.c_str()));
  envp.push_back(nullptr);

  pid_t pid = fork();
  if (pid < 0) {
    res.error = std::string("fork: ") + strerror(errno);
    if (ready_pipe[0] >= 0) { close(ready_pipe[0]); close(ready_pipe[1]); }
    close(err_pipe[0]); close(err_pipe[1]);
    return res;
  }

  if (pid == 0) {
    // ---- child ----
    if (spec.new_session) setsid();
    if (!spec.cgroup_dir.empty()) {
      // Move self into the pod cgroup before exec so all workload threads and
      // descendants inherit it.
      WriteFileString(spec.cgroup_dir + "/cgroup.procs", std::to_string(getpid()));
    }
    if (!spec.cwd.empty() && chdir(spec.cwd.c_str()) != 0) {
      const char msg[] = "chdir failed";
      WriteAll(err_pipe[1], msg, sizeof(msg) - 1);
      _exit(127);
    }
    auto redirect = [&](const std::string& path, int target) {
      if (path.empty()) return true;
      int fd = open(path.c_str(), O_WRONLY | O_CREAT | O_APPEND, 0644);
      if (fd < 0) return false;
      dup2(fd, target);
      close(fd);
      return true;
    };
    if (!redirect(spec.stdout_path, STDOUT_FILENO) ||
        !redirect(spec.stderr_path.empty() ? spec.stdout_path : spec.stderr_path,
                  STDERR_FILENO)) {
      const char msg[] = "log redirect failed";
      WriteAll(err_pipe[1], msg, sizeof(msg) - 1);
      _exit(127);
    }
    if (spec.ready_pipe) {
      // Keep the write end across exec (CLOEXEC was set by pipe2).
      int flags = fcntl(ready_pipe[1], F_GETFD);
      fcntl(ready_pipe[1], F_SETFD, flags & ~FD_CLOEXEC);
      close(ready_pipe[0]);
    }
    execvpe(argv[0], argv.data(), envp.data());
    std::string msg = std::string("execvpe ") + spec.argv[0] + ": " + strerror(errno);
    WriteAll(err_pipe[1], msg.c_str(), msg.size());
    _exit(127);
  }

  // ---- parent ----
  close(err_pipe[1]);
  if (ready_pipe[1] >= 0) close(ready_pipe[1]);

  char errbuf[256];
  ssize_t n = read(err_pipe[0], errbuf, sizeof(errbuf) - 1);  // EOF on exec success
  close(err_pipe[0]);
  if (n > 0) {
    errbuf[n] = '\0';
    waitpid(pid, nullptr, 0);
    if (ready_pipe[0] >= 0) close(ready_pipe[0]);
    res.error = errbuf;
    return res;
  }

  res.pid = pid;
  res.pidfd = PidfdOpen(pid);
  if (ready_pipe[0] >= 0) {
    int fl = fcntl(ready_pipe[0], F_GETFL);
    fcntl(ready_pipe[0], F_SETFL, fl | O_NONBLOCK);
  }
  res.ready_fd = ready_pipe[0];
  return res;
}

int OpenPidfd(int64_t pid) { return PidfdOpen(static_cast<pid_t>(pid)); }

int SignalProcess(int64_t pid, int sig, bool whole_group) {
  pid_t target = whole_group ? -static_cast<pid_t>(pid) : static_cast<pid_t>(pid);
  return kill(target, sig) == 0 ? 0 : errno;
}

bool CgroupCreate(const std::string& path, const std::string& cpu_max,
                  const std::string& memory_max) {
  if (!MkdirP(path)) return false;
  bool ok = true;
  if (!cpu_max.empty()) ok = WriteFileString(path + "/cpu.max", cpu_max) && ok;
  if (!memory_max.empty()) ok = WriteFileString(path + "/memory.max", memory_max) && ok;
  return ok;
}

bool CgroupRemove(const std::string& path) { return rmdir(path.c_str()) == 0; }

long CgroupProcCount(const std::string& path) {
  std::ifstream f(path + "/cgroup.procs");
  if (!f.good()) return -1;
  long count = 0;
  std::string line;
  while (std::getline(f, line))
    if (!line.empty()) ++count;
  return count;
}

// ---------------- EventLoop ----------------

EventLoop::EventLoop() {
  epfd_ = epoll_create1(EPOLL_CLOEXEC);
  wakefd_ = eventfd(0, EFD_CLOEXEC | EFD_NONBLOCK);
  epoll_event ev{};
  ev.events = EPOLLIN;
  ev.data.fd = wakefd_;
  epoll_ctl(epfd_, EPOLL_CTL_ADD, wakefd_, &ev);
}

EventLoop::~EventLoop() {
  std::lock_guard<std::mutex> lock(mu_);
  for (auto& [pid, e] : procs_) {
    if (e.pidfd >= 0) close(e.pidfd);
    if (e.ready_fd >= 0) close(e.ready_fd);
  }
  close(wakefd_);
  close(epfd_);
}

void EventLoop::AddProcess(int64_t pid, int pidfd, int ready_fd, uint64_t token) {
  std::lock_guard<std::mutex> lock(mu_);
  Entry e{pid, token, pidfd, ready_fd};
  procs_[pid] = e;
  epoll_event ev{};
  ev.events = EPOLLIN;
  if (pidfd >= 0) {
    ev.data.fd = pidfd;
    epoll_ctl(epfd_, EPOLL_CTL_ADD, pidfd, &ev);
    fd_to_pid_[pidfd] = pid;
  }
  if (ready_fd >= 0) {
    ev.data.fd = ready_fd;
    epoll_ctl(epfd_, EPOLL_CTL_ADD, ready_fd, &ev);
    fd_to_pid_[ready_fd] = pid;
  }
}

void EventLoop::RemoveProcess(int64_t pid) {
  std::lock_guard<std::mutex> lock(mu_);
  auto it = procs_.find(pid);
  if (it == procs_.end()) return;
  Entry& e = it->second;
  for (int fd : {e.pidfd, e.ready_fd}) {
    if (fd >= 0) {
      epoll_ctl(epfd_, EPOLL_CTL_DEL, fd, nullptr);
      fd_to_pid_.erase(fd);
      close(fd);
    }
  }
  procs_.erase(it);
}

std::vector<Event> EventLoop::Poll(int timeout_ms) {
  epoll_event evs[64];
  int n = epoll_wait(epfd_, evs, 64, timeout_ms);
  std::vector<Event> out;
  if (n <= 0) return out;

  std::lock_guard<std::mutex> lock(mu_);
  // Two passes: readiness pipes first, then pidfd exits. When a fast process
  // writes READY and exits within one epoll batch, the exit handler closes
  // the ready fd — draining ready events first keeps them from being lost.
  std::vector<int> order;
  order.reserve(static_cast<size_t>(n));
  for (int i = 0; i < n; ++i) {
    int fd = evs[i].data.fd;
    auto pit = fd_to_pid_.find(fd);
    bool is_pidfd = false;
    if (pit != fd_to_pid_.end()) {
      auto it = procs_.find(pit->second);
      if (it != procs_.end() && fd == it->second.pidfd) is_pidfd = true;
    }
    if (is_pidfd) order.push_back(i); else order.insert(order.begin(), i);
  }
  for (int idx : order) {
    int fd = evs[idx].data.fd;
    if (fd == wakefd_) {
      uint64_t junk;
      while (read(wakefd_, &junk, sizeof(junk)) > 0) {}
      continue;
    }
    auto pit = fd_to_pid_.find(fd);
    if (pit == fd_to_pid_.end()) continue;
    auto it = procs_.find(pit->second);
    if (it == procs_.end()) continue;
    Entry& e = it->second;

    if (fd == e.pidfd) {
      // Final drain of the readiness pipe before tearing the entry down.
      if (e.ready_fd >= 0) {
        char rbuf[256];
        ssize_t rr = read(e.ready_fd, rbuf, sizeof(rbuf));
        if (rr > 0) {
          Event rev;
          rev.type = Event::kReady;
          rev.pid = e.pid;
          rev.token = e.token;
          rev.data.assign(rbuf, static_cast<size_t>(rr));
          out.push_back(rev);
        }
      }
      // Process exited — reap it.
      int status = 0;
      pid_t r = waitpid(static_cast<pid_t>(e.pid), &status, WNOHANG);
      Event ev;
      ev.type = Event::kExited;
      ev.pid = e.pid;
      ev.token = e.token;
      if (r == e.pid) {
        if (WIFEXITED(status)) ev.exit_code = WEXITSTATUS(status);
        else if (WIFSIGNALED(status)) ev.exit_code = 128 + WTERMSIG(status);
      }
      out.push_back(ev);
      for (int f : {e.pidfd, e.ready_fd}) {
        if (f >= 0) {
          epoll_ctl(epfd_, EPOLL_CTL_DEL, f, nullptr);
          fd_to_pid_.erase(f);
          close(f);
        }
      }
      procs_.erase(it);
    } else if (fd == e.ready_fd) {
      char buf[256];
      ssize_t r = read(fd, buf, sizeof(buf));
      Event ev;
      ev.pid = e.pid;
      ev.token = e.token;
      if (r > 0) {
        ev.type = Event::kReady;
        ev.data.assign(buf, static_cast<size_t>(r));
        out.push_back(ev);
      } else if (r == 0) {
        ev.type = Event::kReadyClosed;
        out.push_back(ev);
        epoll_ctl(epfd_, EPOLL_CTL_DEL, fd, nullptr);
        fd_to_pid_.erase(fd);
        close(fd);
        e.ready_fd = -1;
      }
    }
  }
  return out;
}

void EventLoop::Wake() {
  uint64_t one = 1;
  WriteAll(wakefd_, reinterpret_cast<const char*>(&one), sizeof(one));
}

size_t EventLoop::TrackedCount() const {
  std::lock_guard<std::mutex> lock(mu_);
  return procs_.size();
}

}  // namespace amdvk
