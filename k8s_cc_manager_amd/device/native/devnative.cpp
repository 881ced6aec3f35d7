// Native device library: direct sysfs-PCI and /dev/kfd access.
//
// The AMD-native replacement for the register-poking layer the
// reference borrows from gpu-admin-tools (its find_gpus() walks
// /sys/bus/pci sysfs from Python; /root/reference/main.py:144-155).
// Here the device plumbing is C++:
//
//   pci_scan()        — enumerate AMD accelerators/GPUs from PCI config
//                       space (vendor 0x1002, class 0x0380xx/0x0300xx),
//                       reading each device's config header directly;
//   pci_config_read() — raw config-space bytes for capability checks;
//   pci_reset()       — sysfs function-level reset (privileged path);
//   kfd_version()     — AMDKFD_IOC_GET_VERSION ioctl on /dev/kfd: the
//                       compute stack's liveness signal (a GPU can be
//                       on the bus but absent from KFD after a failed
//                       reset);
//   kfd_topology()    — parse KFD topology nodes (gfx target, CU count,
//                       xGMI links) for capability gating.

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <dirent.h>
#include <fcntl.h>
#include <sys/ioctl.h>
#include <sys/stat.h>
#include <unistd.h>

#include <cstdint>
#include <cstring>
#include <fstream>
#include <sstream>
#include <string>
#include <vector>

namespace py = pybind11;

namespace {

// AMDKFD_IOC_GET_VERSION: _IOR('K', 0x01, struct {u32 major; u32 minor;})
struct KfdVersionArgs {
  uint32_t major;
  uint32_t minor;
};
constexpr unsigned long kKfdGetVersion = 0x80084B01UL;

std::string read_text(const std::string& path) {
  std::ifstream f(path);
  if (!f) return "";
  std::stringstream ss;
  ss << f.rdbuf();
  std::string s = ss.str();
  while (!s.empty() && (s.back() == '\n' || s.back() == ' ')) s.pop_back();
  return s;
}

long read_hex(const std::string& path, long fallback = -1) {
  std::string s = read_text(path);
  if (s.empty()) return fallback;
  return strtol(s.c_str(), nullptr, 0);
}

}  // namespace

// ---------------------------------------------------------------------------
// PCI
// ---------------------------------------------------------------------------

static std::vector<py::dict> pci_scan(const std::string& root) {
  std::vector<py::dict> out;
  DIR* dir = opendir(root.c_str());
  if (!dir) return out;
  struct dirent* ent;
  while ((ent = readdir(dir)) != nullptr) {
    std::string bdf = ent->d_name;
    if (bdf == "." || bdf == "..") continue;
    std::string base = root + "/" + bdf;
    long vendor = read_hex(base + "/vendor");
    if (vendor != 0x1002) continue;  // AMD only — no dual-vendor path
    long cls = read_hex(base + "/class");
    long kind = (cls >> 16) & 0xff;  // base class
    // 0x03 display (incl. 0x0380 processing accelerators on MI355X),
    // 0x12 (future accelerator class) accepted defensively.
    if (kind != 0x03 && kind != 0x12) continue;
    py::dict d;
    d["bdf"] = bdf;
    d["vendor"] = vendor;
    d["device"] = read_hex(base + "/device");
    d["class"] = cls;
    d["numa_node"] = read_hex(base + "/numa_node", -1);
    d["driver"] = [&]() -> std::string {
      char buf[512];
      ssize_t n = readlink((base + "/driver").c_str(), buf, sizeof(buf) - 1);
      if (n <= 0) return "";
      buf[n] = 0;
      std::string s(buf);
      auto pos = s.rfind('/');
      return pos == std::string::npos ? s : s.substr(pos + 1);
    }();
    d["has_reset"] = access((base + "/reset").c_str(), F_OK) == 0;
    out.push_back(std::move(d));
  }
  closedir(dir);
  return out;
}

static py::bytes pci_config_read(const std::string& bdf, size_t offset,
                                 size_t size, const std::string& root) {
  std::string path = root + "/" + bdf + "/config";
  int fd = open(path.c_str(), O_RDONLY);
  if (fd < 0) throw std::runtime_error("open " + path + ": " + strerror(errno));
  std::vector<char> buf(size, 0);
  ssize_t n = pread(fd, buf.data(), size, (off_t)offset);
  close(fd);
  if (n < 0) throw std::runtime_error("pread " + path + ": " + strerror(errno));
  return py::bytes(buf.data(), (size_t)n);
}

static void pci_reset(const std::string& bdf, const std::string& root) {
  std::string path = root + "/" + bdf + "/reset";
  int fd = open(path.c_str(), O_WRONLY);
  if (fd < 0) throw std::runtime_error("open " + path + ": " + strerror(errno));
  ssize_t n = write(fd, "1", 1);
  int err = errno;
  close(fd);
  if (n != 1)
    throw std::runtime_error("FLR write " + path + ": " + strerror(err));
}

// ---------------------------------------------------------------------------
// KFD
// ---------------------------------------------------------------------------

static py::object kfd_version(const std::string& dev_path) {
  int fd = open(dev_path.c_str(), O_RDWR | O_CLOEXEC);
  if (fd < 0) return py::none();
  KfdVersionArgs args{0, 0};
  int rc = ioctl(fd, kKfdGetVersion, &args);
  close(fd);
  if (rc != 0) return py::none();
  return py::make_tuple(args.major, args.minor);
}

static std::vector<py::dict> kfd_topology(const std::string& root) {
  std::vector<py::dict> out;
  DIR* dir = opendir(root.c_str());
  if (!dir) return out;
  struct dirent* ent;
  while ((ent = readdir(dir)) != nullptr) {
    std::string node = ent->d_name;
    if (node == "." || node == "..") continue;
    std::string props_path = root + "/" + node + "/properties";
    std::ifstream f(props_path);
    if (!f) continue;
    py::dict d;
    d["node"] = atoi(node.c_str());
    std::string key;
    long long value;
    int io_links = 0;
    while (f >> key >> value) {
      if (key == "simd_count" || key == "gfx_target_version" ||
          key == "cpu_cores_count" || key == "array_count" ||
          key == "simd_per_cu" || key == "location_id" ||
          key == "domain" || key == "vendor_id" || key == "device_id" ||
          key == "io_links_count")
        d[key.c_str()] = (long)value;
      if (key == "io_links_count") io_links = (int)value;
    }
    d["io_links_count"] = io_links;
    // GPUs have simd_count > 0; CPU nodes are 0
    long simd = d.contains("simd_count") ? d["simd_count"].cast<long>() : 0;
    d["is_gpu"] = simd > 0;
    if (simd > 0 && d.contains("simd_per_cu")) {
      long spc = d["simd_per_cu"].cast<long>();
      d["cu_count"] = spc > 0 ? simd / spc : 0;
    }
    out.push_back(std::move(d));
  }
  closedir(dir);
  return out;
}

PYBIND11_MODULE(_devnative, m) {
  m.doc() = "AMD CC manager native device library (PCI sysfs + /dev/kfd)";
  m.def("pci_scan", &pci_scan, py::arg("root") = "/sys/bus/pci/devices");
  m.def("pci_config_read", &pci_config_read, py::arg("bdf"),
        py::arg("offset") = 0, py::arg("size") = 64,
        py::arg("root") = "/sys/bus/pci/devices");
  m.def("pci_reset", &pci_reset, py::arg("bdf"),
        py::arg("root") = "/sys/bus/pci/devices");
  m.def("kfd_version", &kfd_version, py::arg("dev_path") = "/dev/kfd");
  m.def("kfd_topology", &kfd_topology,
        py::arg("root") = "/sys/class/kfd/kfd/topology/nodes");
}
