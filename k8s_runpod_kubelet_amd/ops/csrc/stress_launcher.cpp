// Concurrency stress harness for the native launcher + event loop.
//
// Race-detection coverage the reference lacks entirely (no
// -race in its CI; known data races in its provider). Built with
// -fsanitize=thread by tests/test_launcher.py::test_native_tsan_stress and
// by CI: N launcher threads spawn short-lived processes and remove some of
// them mid-flight while one poller thread drains the epoll loop — every
// spawned process must produce exactly one exit event (or have been
// explicitly removed), with TSan watching the shared maps.
//
// Usage: stress_launcher [n_threads] [procs_per_thread]

#include "launcher.h"

#include <sys/wait.h>

#include <atomic>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <mutex>
#include <set>
#include <thread>
#include <vector>

using namespace amdvk;

int main(int argc, char** argv) {
  const int n_threads = argc > 1 ? atoi(argv[1]) : 8;
  const int per_thread = argc > 2 ? atoi(argv[2]) : 25;
  const int total = n_threads * per_thread;

  EventLoop loop;
  std::atomic<int> exited{0};
  std::atomic<int> ready{0};
  std::atomic<int> removed{0};
  std::atomic<bool> done{false};
  std::mutex mu;
  std::set<int64_t> seen_exit;

  std::thread poller([&] {
    while (!done.load() || loop.TrackedCount() > 0) {
      for (const Event& ev : loop.Poll(50)) {
        if (ev.type == Event::kExited) {
          std::lock_guard<std::mutex> lock(mu);
          if (!seen_exit.insert(ev.pid).second) {
            fprintf(stderr, "FAIL: duplicate exit for pid %ld\n",
                    static_cast<long>(ev.pid));
            _exit(2);
          }
          exited.fetch_add(1);
        } else if (ev.type == Event::kReady) {
          ready.fetch_add(1);
        }
      }
    }
  });

  std::vector<std::thread> workers;
  for (int t = 0; t < n_threads; ++t) {
    workers.emplace_back([&, t] {
      for (int i = 0; i < per_thread; ++i) {
        LaunchSpec spec;
        // bash: multi-digit fd redirection after expansion is supported.
        spec.argv = {"/bin/bash", "-c",
                     "echo READY >&${AMDVK_READY_FD}; exit 0"};
        spec.env = {"PATH=/usr/bin:/bin"};
        spec.stdout_path = "/dev/null";
        spec.new_session = true;
        spec.ready_pipe = true;
        LaunchResult res = LaunchProcess(spec);
        if (!res.error.empty()) {
          fprintf(stderr, "FAIL: launch: %s\n", res.error.c_str());
          _exit(3);
        }
        loop.AddProcess(res.pid, res.pidfd, res.ready_fd,
                        static_cast<uint64_t>(res.pid));
        // Every 5th process is force-removed mid-flight (the pod
        // force-delete path) — its events may or may not arrive first;
        // either way nothing must crash or double-fire.
        if (i % 5 == 4) {
          loop.RemoveProcess(res.pid);
          removed.fetch_add(1);
          // Reap it ourselves so no zombie outlives the harness.
          SignalProcess(res.pid, 9, false);
        }
      }
    });
  }
  for (auto& w : workers) w.join();
  done.store(true);
  poller.join();

  // Reap force-removed children (their pidfds were closed unread).
  int status = 0;
  while (wait(&status) > 0) {}

  const int accounted = exited.load() + removed.load();
  if (accounted < total) {
    fprintf(stderr, "FAIL: %d/%d processes accounted (exited=%d removed=%d)\n",
            accounted, total, exited.load(), removed.load());
    return 4;
  }

  // Phase 2: the OCI-rootfs code paths under concurrency. A broken rootfs
  // spec must fail cleanly (never exec on the host, never leak fds or
  // crash) from many threads at once, and the mount-namespace capability
  // probe must be thread-safe — TSan/ASan watch both.
  std::atomic<int> rootfs_errors{0};
  std::vector<std::thread> phase2;
  for (int t = 0; t < 4; ++t) {
    phase2.emplace_back([&] {
      for (int i = 0; i < 10; ++i) {
        (void)ProbeMountNamespace();
        LaunchSpec spec;
        spec.argv = {"/bin/true"};
        spec.env = {"PATH=/usr/bin:/bin"};
        spec.ready_pipe = false;
        spec.rootfs = "/amdvk-stress-nonexistent-rootfs";
        MountSpec m;
        m.src = "/amdvk-stress-nonexistent-src";
        m.dst = "/amdvk-stress-nonexistent-rootfs/x";
        m.flags = 0x1000;  // MS_BIND
        spec.mounts.push_back(m);
        LaunchResult res = LaunchProcess(spec);
        if (res.error.empty()) {
          fprintf(stderr, "FAIL: broken rootfs spec launched anyway "
                          "(pid %ld)\n", static_cast<long>(res.pid));
          _exit(5);
        }
        rootfs_errors.fetch_add(1);
      }
    });
  }
  for (auto& w : phase2) w.join();
  while (wait(&status) > 0) {}  // reap any phase-2 failed children

  printf("ok: %d spawned, %d exit events, %d ready events, %d removed, "
         "%d rootfs-failures-contained\n",
         total, exited.load(), ready.load(), removed.load(),
         rootfs_errors.load());
  return 0;
}
