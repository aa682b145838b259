// pybind11 bindings for the native probe + launcher (_native extension).

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "launcher.h"
#include "probe.h"

namespace py = pybind11;
using namespace amdvk;

PYBIND11_MODULE(_native, m) {
  m.doc() = "MI355X virtual-kubelet native components: KFD/DRM GPU probe + "
            "pidfd/epoll process launcher";

  py::class_<XgmiLink>(m, "XgmiLink")
      .def_readonly("peer_kfd_node", &XgmiLink::peer_kfd_node)
      .def_readonly("peer_gpu_index", &XgmiLink::peer_gpu_index)
      .def_readonly("weight", &XgmiLink::weight)
      .def_readonly("min_bandwidth_mbs", &XgmiLink::min_bandwidth_mbs)
      .def_readonly("max_bandwidth_mbs", &XgmiLink::max_bandwidth_mbs);

  py::class_<GpuInfo>(m, "GpuInfo")
      .def_readonly("index", &GpuInfo::index)
      .def_readonly("kfd_node", &GpuInfo::kfd_node)
      .def_readonly("render_minor", &GpuInfo::render_minor)
      .def_readonly("gpu_id", &GpuInfo::gpu_id)
      .def_readonly("unique_id", &GpuInfo::unique_id)
      .def_readonly("gfx_target_version", &GpuInfo::gfx_target_version)
      .def_readonly("device_id", &GpuInfo::device_id)
      .def_readonly("location_id", &GpuInfo::location_id)
      .def_readonly("cu_count", &GpuInfo::cu_count)
      .def_readonly("max_engine_clk_mhz", &GpuInfo::max_engine_clk_mhz)
      .def_readonly("vram_total_bytes", &GpuInfo::vram_total_bytes)
      .def_readonly("vram_used_bytes", &GpuInfo::vram_used_bytes)
      .def_readonly("busy_percent", &GpuInfo::busy_percent)
      .def_readonly("temperature_mc", &GpuInfo::temperature_mc)
      .def_readonly("ras_uncorrectable", &GpuInfo::ras_uncorrectable)
      .def_readonly("healthy", &GpuInfo::healthy)
      .def_readonly("xgmi_links", &GpuInfo::xgmi_links);

  py::class_<GpuDynamic>(m, "GpuDynamic")
      .def_readonly("vram_used_bytes", &GpuDynamic::vram_used_bytes)
      .def_readonly("vram_total_bytes", &GpuDynamic::vram_total_bytes)
      .def_readonly("busy_percent", &GpuDynamic::busy_percent)
      .def_readonly("temperature_mc", &GpuDynamic::temperature_mc);

  m.def("enumerate_gpus", &EnumerateGpus, py::arg("sysfs_root") = "/sys",
        py::call_guard<py::gil_scoped_release>());
  m.def("read_gpu_dynamic", &ReadGpuDynamic, py::arg("sysfs_root"),
        py::arg("render_minor"), py::call_guard<py::gil_scoped_release>());

  m.def(
      "launch_process",
      [](const std::vector<std::string>& argv, const std::vector<std::string>& env,
         const std::string& cwd, const std::string& stdout_path,
         const std::string& stderr_path, const std::string& cgroup_dir,
         bool new_session, bool ready_pipe, int64_t uid, int64_t gid,
         bool new_pid_ns, const std::string& hostname,
         const std::string& rootfs, bool chroot_only,
         const std::vector<std::tuple<std::string, std::string, std::string,
                                      std::string, uint64_t, bool>>& mounts,
         int64_t setns_pid) {
        LaunchSpec spec;
        spec.argv = argv;
        spec.env = env;
        spec.cwd = cwd;
        spec.stdout_path = stdout_path;
        spec.stderr_path = stderr_path;
        spec.cgroup_dir = cgroup_dir;
        spec.new_session = new_session;
        spec.ready_pipe = ready_pipe;
        spec.uid = uid;
        spec.gid = gid;
        spec.new_pid_ns = new_pid_ns;
        spec.hostname = hostname;
        spec.rootfs = rootfs;
        spec.chroot_only = chroot_only;
        spec.setns_pid = setns_pid;
        for (const auto& m_ : mounts) {
          MountSpec ms;
          ms.src = std::get<0>(m_);
          ms.dst = std::get<1>(m_);
          ms.fstype = std::get<2>(m_);
          ms.data = std::get<3>(m_);
          ms.flags = static_cast<unsigned long>(std::get<4>(m_));
          ms.readonly = std::get<5>(m_);
          spec.mounts.push_back(std::move(ms));
        }
        LaunchResult res;
        {
          py::gil_scoped_release release;
          res = LaunchProcess(spec);
        }
        if (!res.error.empty()) throw std::runtime_error(res.error);
        return py::make_tuple(res.pid, res.pidfd, res.ready_fd,
                              res.spawn_ns * 1e-9, res.cgroup_ns * 1e-9);
      },
      py::arg("argv"), py::arg("env"), py::arg("cwd") = "",
      py::arg("stdout_path") = "", py::arg("stderr_path") = "",
      py::arg("cgroup_dir") = "", py::arg("new_session") = true,
      py::arg("ready_pipe") = true, py::arg("uid") = -1, py::arg("gid") = -1,
      py::arg("new_pid_ns") = false, py::arg("hostname") = "",
      py::arg("rootfs") = "", py::arg("chroot_only") = false,
      py::arg("mounts") =
          std::vector<std::tuple<std::string, std::string, std::string,
                                 std::string, uint64_t, bool>>{},
      py::arg("setns_pid") = -1);

  // mount-namespace capability probe (false inside sandboxes that drop
  // CAP_SYS_ADMIN): decides mountns-vs-chroot image isolation once.
  m.def("probe_mount_namespace", &ProbeMountNamespace,
        py::call_guard<py::gil_scoped_release>());

  m.def("open_pidfd", &OpenPidfd, py::arg("pid"));
  m.def("signal_process", &SignalProcess, py::arg("pid"), py::arg("sig"),
        py::arg("whole_group") = false);
  m.def("cgroup_create", &CgroupCreate, py::arg("path"), py::arg("cpu_max") = "",
        py::arg("memory_max") = "");
  m.def("cgroup_remove", &CgroupRemove, py::arg("path"));
  m.def("cgroup_proc_count", &CgroupProcCount, py::arg("path"));
  m.def("cgroup_attach_device_filter", &CgroupAttachDeviceFilter,
        py::arg("cgroup_dir"), py::arg("denied_major"),
        py::arg("allowed_minors"));

  py::class_<Event>(m, "Event")
      .def_property_readonly("type",
                             [](const Event& e) {
                               switch (e.type) {
                                 case Event::kExited: return "exited";
                                 case Event::kReady: return "ready";
                                 case Event::kReadyClosed: return "ready_closed";
                               }
                               return "unknown";
                             })
      .def_readonly("pid", &Event::pid)
      .def_readonly("token", &Event::token)
      .def_readonly("exit_code", &Event::exit_code)
      .def_readonly("data", &Event::data);

  py::class_<EventLoop>(m, "EventLoop")
      .def(py::init<>())
      .def("add_process", &EventLoop::AddProcess, py::arg("pid"), py::arg("pidfd"),
           py::arg("ready_fd"), py::arg("token"))
      .def("remove_process", &EventLoop::RemoveProcess, py::arg("pid"))
      .def("poll", &EventLoop::Poll, py::arg("timeout_ms"),
           py::call_guard<py::gil_scoped_release>())
      .def("wake", &EventLoop::Wake)
      .def("tracked_count", &EventLoop::TrackedCount);
}
