#pragma once

#include <cstdint>
#include <map>
#include <mutex>
#include <string>
#include <vector>

namespace amdvk {

// One mount performed by the child inside its own mount namespace, in
// order, before pivot_root. dst is the absolute (rootfs-prefixed) target;
// all target files/dirs are pre-created by the caller — the child only
// issues mount(2) (it runs with CLONE_VM and may not allocate).
struct MountSpec {
  std::string src;
  std::string dst;
  std::string fstype;  // "" = bind mount (caller sets MS_BIND in flags)
  std::string data;    // e.g. overlay "lowerdir=..,upperdir=..,workdir=.."
  unsigned long flags = 0;
  bool readonly = false;  // bind remount MS_RDONLY after mounting
};

struct LaunchSpec {
  std::vector<std::string> argv;
  std::vector<std::string> env;  // "KEY=VALUE" entries (full environment)
  std::string cwd;
  std::string stdout_path;
  std::string stderr_path;
  std::string cgroup_dir;  // if non-empty, child is placed here before exec
  bool new_session = true;
  bool ready_pipe = true;  // create readiness pipe, exported as AMDVK_READY_FD
  // Pod securityContext runAsUser/runAsGroup: credentials dropped in the
  // child just before exec (-1 = inherit). Requires the clone3 fast path
  // (argv[0] with a '/'): posix_spawn has no setuid file action.
  int64_t uid = -1;
  int64_t gid = -1;
  // Container-like isolation (clone3 path; needs CAP_SYS_ADMIN — retried
  // without when the kernel refuses): own PID namespace (workload is pid 1,
  // descendants die with it) and, when hostname is non-empty, own UTS
  // namespace with that hostname (k8s pod-hostname semantics).
  bool new_pid_ns = false;
  std::string hostname;
  // Container-image execution (OCI rootfs). Non-empty rootfs + mount mode:
  // child gets CLONE_NEWNS, performs `mounts` in order, pivot_roots into
  // rootfs. chroot_only is the degraded mode for hosts without
  // CAP_SYS_ADMIN-in-sandbox: plain chroot(rootfs) (caller prepared a
  // fully-populated private rootfs copy; no mounts possible). A failed
  // rootfs setup NEVER execs on the host filesystem.
  std::string rootfs;
  std::vector<MountSpec> mounts;
  bool chroot_only = false;
  // kubectl-exec into a live image container: join the mount (+UTS)
  // namespace of this pid before exec (nsenter semantics). Mutually
  // exclusive with rootfs/mounts. -1 = off.
  int64_t setns_pid = -1;
};

struct LaunchResult {
  int64_t pid = -1;
  int pidfd = -1;
  int ready_fd = -1;  // parent read end of the readiness pipe (-1 if disabled)
  // Costs measured on the native side (Python-side wall timing overstates
  // them by ~100 ms whenever the launching thread loses the GIL to the
  // already-running child). Split so the cgroup.procs migration — which
  // serializes on the kernel's cgroup_mutex against concurrent teardown —
  // is attributable separately from posix_spawnp itself.
  int64_t spawn_ns = 0;   // posix_spawnp only
  int64_t cgroup_ns = 0;  // cgroup.procs migration write
  std::string error;
};

LaunchResult LaunchProcess(const LaunchSpec& spec);

// Capability probe: can this process create a mount namespace and mount in
// it? (The gpurun sandbox drops CAP_SYS_ADMIN.) Forks a throwaway child —
// call once and cache.
bool ProbeMountNamespace();

// pidfd for an already-running process (adoption-on-restart path).
int OpenPidfd(int64_t pid);

// Send a signal to a pid or (negative) to its process group.
int SignalProcess(int64_t pid, int sig, bool whole_group);

// cgroup v2 helpers (best-effort: return false without throwing when the
// controller files are absent/unwritable — e.g. unprivileged test runs).
bool CgroupCreate(const std::string& path, const std::string& cpu_max,
                  const std::string& memory_max);
bool CgroupRemove(const std::string& path);
long CgroupProcCount(const std::string& path);

// Attach a cgroup-v2 eBPF device filter to the pod cgroup: devices with
// major == denied_major are only accessible at the listed minors; all other
// majors pass. Used to make GPU binding *enforced* (deny other GPUs'
// /dev/dri/renderD<minor> nodes — ROCm cannot acquire a KFD VM without the
// render node). Best-effort: false when the kernel/permissions refuse.
bool CgroupAttachDeviceFilter(const std::string& cgroup_dir, int denied_major,
                              const std::vector<int>& allowed_minors);

struct Event {
  enum Type { kExited = 0, kReady = 1, kReadyClosed = 2 };
  Type type;
  int64_t pid = -1;
  uint64_t token = 0;
  int exit_code = -1;    // kExited: exit status, or 128+signal
  std::string data;      // kReady: bytes read from the readiness pipe
};

// epoll loop over pidfds + readiness pipes. One instance per runtime; a
// Python watcher thread calls Poll() (GIL released while blocked) and turns
// events into provider status updates — this is what replaces the
// reference's 10 s/30 s cloud-status polling (kubelet.go:292-303, :713-731)
// with sub-millisecond push.
class EventLoop {
 public:
  EventLoop();
  ~EventLoop();

  // Register a process; both fds are owned by the loop afterwards.
  void AddProcess(int64_t pid, int pidfd, int ready_fd, uint64_t token);
  // Stop tracking (e.g. pod force-removed); closes owned fds.
  void RemoveProcess(int64_t pid);
  std::vector<Event> Poll(int timeout_ms);
  void Wake();
  size_t TrackedCount() const;

 private:
  struct Entry {
    int64_t pid;
    uint64_t token;
    int pidfd;
    int ready_fd;
  };
  int epfd_;
  int wakefd_;
  std::map<int64_t, Entry> procs_;      // pid -> entry
  std::map<int, int64_t> fd_to_pid_;    // registered fd -> pid
  mutable std::mutex mu_;
};

}  // namespace amdvk
