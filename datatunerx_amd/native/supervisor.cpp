// Native runtime core: GPU gang inventory + training-process supervisor.
//
// This is the MI355X replacement for the reference's dispatch layer
// (KubeRay RayJob -> RayCluster pods): instead of
// delegating to an external operator, the controllers gang-allocate GPUs
// from this inventory and launch one trainer process per GPU directly
// (fork/exec with torchrun-style env), supervising them via waitpid.
// Exposed to the Python controllers via pybind11 as
// datatunerx_amd.native._dtx_native.
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <csignal>
#include <cstring>
#include <fcntl.h>
#include <map>
#include <mutex>
#include <string>
#include <sys/stat.h>
#include <sys/types.h>
#include <sys/wait.h>
#include <unistd.h>
#include <vector>

namespace py = pybind11;

// ------------------------------------------------------------ inventory
// Gang allocation: all-or-nothing GPU sets (the Experiment scheduler
// packs concurrent jobs onto one 8-GPU node; admission = enough free
// GPUs, queueing otherwise — replaces unbounded RayJob creation).
class GpuInventory {
 public:
  explicit GpuInventory(int n_gpus) : free_(n_gpus, true) {}

  // returns empty vector if the gang doesn't fit (caller requeues)
  std::vector<int> allocate(int count, const std::string& owner) {
    std::lock_guard<std::mutex> g(mu_);
    std::vector<int> picked;
    for (int i = 0; i < (int)free_.size() && (int)picked.size() < count; ++i)
      if (free_[i]) picked.push_back(i);
    if ((int)picked.size() < count) return {};
    for (int id : picked) {
      free_[id] = false;
      owners_[id] = owner;
    }
    return picked;
  }

  void release(const std::vector<int>& ids) {
    std::lock_guard<std::mutex> g(mu_);
    for (int id : ids)
      if (id >= 0 && id < (int)free_.size()) {
        free_[id] = true;
        owners_.erase(id);
      }
  }

  void release_owner(const std::string& owner) {
    std::lock_guard<std::mutex> g(mu_);
    for (auto it = owners_.begin(); it != owners_.end();) {
      if (it->second == owner) {
        free_[it->first] = true;
        it = owners_.erase(it);
      } else {
        ++it;
      }
    }
  }

  int n_free() const {
    std::lock_guard<std::mutex> g(mu_);
    int n = 0;
    for (bool f : free_) n += f;
    return n;
  }

  int n_total() const { return (int)free_.size(); }

  std::map<int, std::string> owners() const {
    std::lock_guard<std::mutex> g(mu_);
    return owners_;
  }

 private:
  mutable std::mutex mu_;
  std::vector<bool> free_;
  std::map<int, std::string> owners_;
};

// ------------------------------------------------------------ processes
class ProcessSupervisor {
 public:
  // Spawn argv[0] with args, extra env vars, stdout+stderr -> log_path.
  // Returns pid (or -1). The child gets its own process group so a gang
  // can be killed together.
  long spawn(const std::vector<std::string>& argv,
             const std::map<std::string, std::string>& env,
             const std::string& log_path, const std::string& cwd) {
    pid_t pid = fork();
    if (pid < 0) return -1;
    if (pid == 0) {
      setpgid(0, 0);
      if (!cwd.empty()) { if (chdir(cwd.c_str()) != 0) _exit(126); }
      if (!log_path.empty()) {
        int fd = open(log_path.c_str(), O_WRONLY | O_CREAT | O_APPEND, 0644);
        if (fd >= 0) {
          dup2(fd, 1);
          dup2(fd, 2);
          close(fd);
        }
      }
      for (const auto& kv : env)
        setenv(kv.first.c_str(), kv.second.c_str(), 1);
      std::vector<char*> cargv;
      for (const auto& a : argv) cargv.push_back(const_cast<char*>(a.c_str()));
      cargv.push_back(nullptr);
      execvp(cargv[0], cargv.data());
      _exit(127);
    }
    std::lock_guard<std::mutex> g(mu_);
    running_[pid] = true;
    return (long)pid;
  }

  // -1 = still running; >=0 exit code; -2 killed by signal; -3 unknown pid
  int poll(long pid) {
    int status = 0;
    pid_t r = waitpid((pid_t)pid, &status, WNOHANG);
    if (r == 0) return -1;
    if (r < 0) {
      std::lock_guard<std::mutex> g(mu_);
      auto it = results_.find(pid);
      return it == results_.end() ? -3 : it->second;
    }
    int code = WIFEXITED(status) ? WEXITSTATUS(status) : -2;
    std::lock_guard<std::mutex> g(mu_);
    results_[pid] = code;
    running_.erase(pid);
    return code;
  }

  // SIGTERM the process group; escalate with SIGKILL via kill_group.
  void terminate(long pid) { ::kill(-(pid_t)pid, SIGTERM); }
  void kill_group(long pid) { ::kill(-(pid_t)pid, SIGKILL); }

  // SIGTERM the group and reap (blocking up to timeout_ms; SIGKILL on
  // expiry). Prevents zombies for processes the controllers stop.
  int stop_and_reap(long pid, int timeout_ms) {
    ::kill(-(pid_t)pid, SIGTERM);
    const int step_ms = 20;
    for (int waited = 0; waited < timeout_ms; waited += step_ms) {
      int code = poll(pid);
      if (code != -1) return code;
      usleep(step_ms * 1000);
    }
    ::kill(-(pid_t)pid, SIGKILL);
    for (int i = 0; i < 100; ++i) {
      int code = poll(pid);
      if (code != -1) return code;
      usleep(step_ms * 1000);
    }
    return -1;
  }

  bool alive(long pid) { return poll(pid) == -1; }

 private:
  std::mutex mu_;
  std::map<long, bool> running_;
  std::map<long, int> results_;
};

PYBIND11_MODULE(_dtx_native, m) {
  m.doc() = "datatunerx_amd native runtime: gang GPU inventory + process "
            "supervisor (C++)";
  py::class_<GpuInventory>(m, "GpuInventory")
      .def(py::init<int>())
      .def("allocate", &GpuInventory::allocate)
      .def("release", &GpuInventory::release)
      .def("release_owner", &GpuInventory::release_owner)
      .def("n_free", &GpuInventory::n_free)
      .def("n_total", &GpuInventory::n_total)
      .def("owners", &GpuInventory::owners);
  py::class_<ProcessSupervisor>(m, "ProcessSupervisor")
      .def(py::init<>())
      .def("spawn", &ProcessSupervisor::spawn, py::arg("argv"),
           py::arg("env"), py::arg("log_path") = "", py::arg("cwd") = "")
      .def("poll", &ProcessSupervisor::poll)
      .def("terminate", &ProcessSupervisor::terminate)
      .def("kill_group", &ProcessSupervisor::kill_group)
      .def("stop_and_reap", &ProcessSupervisor::stop_and_reap,
           py::arg("pid"), py::arg("timeout_ms") = 3000)
      .def("alive", &ProcessSupervisor::alive);
}
