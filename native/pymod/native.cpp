// clawker_amd._native — in-process native helpers.
//
// GpuSampler: zero-spawn amdgpu telemetry. The reference's stats pane
// shells out to `docker stats` (internal/cmd/container/stats/stats.go
// streamStats); at 8-sandbox concurrency on an MI355X node, spawning
// rocm-smi per sample would dominate the monitor loop (SURVEY.md §7 names
// this as new GPU work). Instead we keep per-metric sysfs fds open and
// pread() them each tick: one sample across 8 GPUs costs microseconds.
//
// Metrics per GPU (amdgpu sysfs + hwmon):
//   gpu_busy_percent, mem_info_vram_used/total, hwmon temp*_input (edge +
//   junction + mem), power1_average (uW), freq1/freq2 (sclk/mclk Hz).

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <dirent.h>
#include <fcntl.h>
#include <string.h>
#include <unistd.h>

#include <map>
#include <string>
#include <vector>

namespace py = pybind11;

namespace {

struct Metric {
  int fd = -1;
  double scale = 1.0;
};

class GpuSampler {
 public:
  explicit GpuSampler(const std::vector<int>& render_minors,
                      const std::string& drm_class = "/sys/class/drm") {
    for (int minor : render_minors) {
      std::string dev = drm_class + "/renderD" + std::to_string(minor) + "/device";
      Gpu g;
      g.minor = minor;
      add(g, "busy_pct", dev + "/gpu_busy_percent", 1.0);
      add(g, "vram_used", dev + "/mem_info_vram_used", 1.0);
      add(g, "vram_total", dev + "/mem_info_vram_total", 1.0);
      add(g, "gtt_used", dev + "/mem_info_gtt_used", 1.0);
      // hwmon subdir name varies; scan once
      std::string hw = find_hwmon(dev + "/hwmon");
      if (!hw.empty()) {
        add(g, "temp_edge_c", hw + "/temp1_input", 1e-3);
        add(g, "temp_junction_c", hw + "/temp2_input", 1e-3);
        add(g, "temp_mem_c", hw + "/temp3_input", 1e-3);
        add(g, "power_w", hw + "/power1_average", 1e-6);
        if (!has(g, "power_w")) add(g, "power_w", hw + "/power1_input", 1e-6);
        add(g, "sclk_mhz", hw + "/freq1_input", 1e-6);
        add(g, "mclk_mhz", hw + "/freq2_input", 1e-6);
      }
      gpus_.push_back(std::move(g));
    }
  }

  ~GpuSampler() {
    for (auto& g : gpus_)
      for (auto& kv : g.metrics)
        if (kv.second.fd >= 0) close(kv.second.fd);
  }

  // one sample across all GPUs: list of {name: value} dicts
  std::vector<std::map<std::string, double>> sample() {
    std::vector<std::map<std::string, double>> out;
    char buf[64];
    for (auto& g : gpus_) {
      std::map<std::string, double> row;
      row["minor"] = g.minor;
      for (auto& kv : g.metrics) {
        if (kv.second.fd < 0) continue;
        ssize_t n = pread(kv.second.fd, buf, sizeof buf - 1, 0);
        if (n <= 0) continue;
        buf[n] = 0;
        row[kv.first] = strtod(buf, nullptr) * kv.second.scale;
      }
      out.push_back(std::move(row));
    }
    return out;
  }

  size_t num_gpus() const { return gpus_.size(); }

 private:
  struct Gpu {
    int minor;
    std::map<std::string, Metric> metrics;
  };

  static void add(Gpu& g, const char* name, const std::string& path, double scale) {
    int fd = open(path.c_str(), O_RDONLY | O_CLOEXEC);
    if (fd >= 0) g.metrics[name] = Metric{fd, scale};
  }

  static bool has(const Gpu& g, const char* name) { return g.metrics.count(name) > 0; }

  static std::string find_hwmon(const std::string& dir) {
    DIR* d = opendir(dir.c_str());
    if (!d) return "";
    std::string out;
    while (dirent* e = readdir(d)) {
      if (strncmp(e->d_name, "hwmon", 5) == 0) {
        out = dir + "/" + e->d_name;
        break;
      }
    }
    closedir(d);
    return out;
  }

  std::vector<Gpu> gpus_;
};

// fast multi-file reader for cgroup stats (same fd-cache trick)
class FileSampler {
 public:
  explicit FileSampler(const std::vector<std::string>& paths) {
    for (const auto& p : paths) {
      int fd = open(p.c_str(), O_RDONLY | O_CLOEXEC);
      fds_.push_back(fd);
      paths_.push_back(p);
    }
  }
  ~FileSampler() {
    for (int fd : fds_)
      if (fd >= 0) close(fd);
  }
  std::map<std::string, std::string> sample() {
    std::map<std::string, std::string> out;
    char buf[4096];
    for (size_t i = 0; i < fds_.size(); i++) {
      if (fds_[i] < 0) continue;
      ssize_t n = pread(fds_[i], buf, sizeof buf - 1, 0);
      if (n <= 0) continue;
      buf[n] = 0;
      out[paths_[i]] = buf;
    }
    return out;
  }

 private:
  std::vector<int> fds_;
  std::vector<std::string> paths_;
};

}  // namespace

PYBIND11_MODULE(_native, m) {
  m.doc() = "clawker-amd native helpers (zero-spawn amdgpu/cgroup samplers)";
  py::class_<GpuSampler>(m, "GpuSampler")
      .def(py::init<const std::vector<int>&, const std::string&>(),
           py::arg("render_minors"), py::arg("drm_class") = "/sys/class/drm")
      .def("sample", &GpuSampler::sample)
      .def_property_readonly("num_gpus", &GpuSampler::num_gpus);
  py::class_<FileSampler>(m, "FileSampler")
      .def(py::init<const std::vector<std::string>&>(), py::arg("paths"))
      .def("sample", &FileSampler::sample);
}
