// _sysfs — one-pass PCI sysfs scanner and Allocate-path revalidator.
//
// The Allocate RPC's latency budget is dominated by synchronous sysfs
// re-reads (reference does several os.ReadFile/Readlink per group
// member per request, generic_device_plugin.go:383-410; SURVEY.md §3.2
// calls this out as where Allocate latency lives).  This extension
// collapses those into openat/readlinkat syscalls from C with one
// Python call per RPC, and gives discovery a single-pass scanner.

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <dirent.h>
#include <fcntl.h>
#include <sys/stat.h>
#include <sys/types.h>
#include <unistd.h>

#include <algorithm>
#include <cstring>
#include <string>
#include <vector>

namespace py = pybind11;

namespace {

// Read a small file relative to dirfd; returns false when unreadable.
bool read_small(int dirfd, const char *rel, std::string *out) {
  int fd = openat(dirfd, rel, O_RDONLY | O_CLOEXEC);
  if (fd < 0) return false;
  char buf[256];
  ssize_t n = read(fd, buf, sizeof buf - 1);
  close(fd);
  if (n <= 0) return false;
  buf[n] = 0;
  while (n > 0 && (buf[n - 1] == '\n' || buf[n - 1] == ' ')) buf[--n] = 0;
  out->assign(buf, n);
  return true;
}

// 0x1002\n -> 1002 (same positional slice as the reference's data[2:],
// device_plugin.go:300).
bool read_id(int dirfd, const char *rel, std::string *out) {
  std::string raw;
  if (!read_small(dirfd, rel, &raw) || raw.size() < 2) return false;
  *out = raw.substr(2);
  return true;
}

bool link_basename(int dirfd, const char *rel, std::string *out) {
  char buf[512];
  ssize_t n = readlinkat(dirfd, rel, buf, sizeof buf - 1);
  if (n <= 0) return false;
  buf[n] = 0;
  const char *slash = strrchr(buf, '/');
  *out = slash ? slash + 1 : buf;
  return true;
}

struct PciRecord {
  std::string addr, device, driver, iommu_group, physfn, vfio_dev;
  long numa_node = 0;
};

// Scan base_path for functions of `vendor`; one readdir pass, openat
// reads relative to each device dir.
py::list scan_pci(const std::string &base_path, const std::string &vendor) {
  py::list out;
  int base = open(base_path.c_str(),
                  O_RDONLY | O_DIRECTORY | O_CLOEXEC);
  if (base < 0)
    throw std::runtime_error("cannot open " + base_path);
  DIR *dir = fdopendir(dup(base));
  if (!dir) {
    close(base);
    throw std::runtime_error("cannot read " + base_path);
  }
  std::vector<std::string> names;
  while (struct dirent *de = readdir(dir)) {
    if (de->d_name[0] == '.') continue;
    names.emplace_back(de->d_name);
  }
  closedir(dir);
  std::sort(names.begin(), names.end());

  for (const auto &name : names) {
    int devfd = openat(base, name.c_str(),
                       O_RDONLY | O_DIRECTORY | O_CLOEXEC);
    if (devfd < 0) continue;
    std::string v;
    if (!read_id(devfd, "vendor", &v) || v != vendor) {
      close(devfd);
      continue;
    }
    PciRecord r;
    r.addr = name;
    bool ok = link_basename(devfd, "driver", &r.driver);
    ok = ok && link_basename(devfd, "iommu_group", &r.iommu_group);
    ok = ok && read_id(devfd, "device", &r.device);
    std::string numa;
    if (read_small(devfd, "numa_node", &numa)) {
      r.numa_node = strtol(numa.c_str(), nullptr, 10);
      if (r.numa_node < 0) r.numa_node = 0;
    }
    link_basename(devfd, "physfn", &r.physfn);
    // iommufd cdev name, when present
    int vd = openat(devfd, "vfio-dev", O_RDONLY | O_DIRECTORY | O_CLOEXEC);
    if (vd >= 0) {
      DIR *vdir = fdopendir(vd);
      if (vdir) {
        while (struct dirent *de = readdir(vdir)) {
          if (strncmp(de->d_name, "vfio", 4) == 0) {
            r.vfio_dev = de->d_name;
            break;
          }
        }
        closedir(vdir);
      } else {
        close(vd);
      }
    }
    close(devfd);
    if (!ok) continue;
    py::dict d;
    d["addr"] = r.addr;
    d["device"] = r.device;
    d["driver"] = r.driver;
    d["iommu_group"] = r.iommu_group;
    d["numa_node"] = r.numa_node;
    d["physfn"] = r.physfn;
    d["vfio_dev"] = r.vfio_dev;
    out.append(d);
  }
  close(base);
  return out;
}

// TOCTOU revalidation for Allocate: for each (addr, expected_group),
// confirm the iommu_group link and vendor are unchanged
// (reference: generic_device_plugin.go:389-398).  Returns
// (failed_addr, reason, vfio_devs): failed_addr "" when all pass;
// reason "changed" (group/vendor drift — invalid request) or "no_cdev"
// (iommufd cdev missing — internal error, matching the reference's
// error split, generic_device_plugin.go:403-409).
py::tuple revalidate(const std::string &base_path,
                     const std::vector<std::pair<std::string, std::string>>
                         &addr_groups,
                     const std::string &vendor, bool want_vfio_dev) {
  int base = open(base_path.c_str(), O_RDONLY | O_DIRECTORY | O_CLOEXEC);
  if (base < 0)
    return py::make_tuple(addr_groups.empty()
                              ? std::string("")
                              : addr_groups.front().first,
                          std::string("changed"), py::list());
  py::list vfio_devs;
  for (const auto &ag : addr_groups) {
    int devfd = openat(base, ag.first.c_str(),
                       O_RDONLY | O_DIRECTORY | O_CLOEXEC);
    if (devfd < 0) {
      close(base);
      return py::make_tuple(ag.first, std::string("changed"),
                            py::list());
    }
    std::string group, v;
    bool ok = link_basename(devfd, "iommu_group", &group) &&
              group == ag.second && read_id(devfd, "vendor", &v) &&
              v == vendor;
    if (!ok) {
      close(devfd);
      close(base);
      return py::make_tuple(ag.first, std::string("changed"),
                            py::list());
    }
    std::string cdev;
    if (want_vfio_dev) {
      bool found = false;
      int vd = openat(devfd, "vfio-dev",
                      O_RDONLY | O_DIRECTORY | O_CLOEXEC);
      if (vd >= 0) {
        DIR *vdir = fdopendir(vd);
        if (vdir) {
          while (struct dirent *de = readdir(vdir)) {
            if (strncmp(de->d_name, "vfio", 4) == 0) {
              cdev = de->d_name;
              found = true;
              break;
            }
          }
          closedir(vdir);
        } else {
          close(vd);
        }
      }
      if (!found) {
        close(devfd);
        close(base);
        return py::make_tuple(ag.first, std::string("no_cdev"),
                              py::list());
      }
      vfio_devs.append(cdev);
    }
    close(devfd);
  }
  close(base);
  return py::make_tuple(std::string(""), std::string(""), vfio_devs);
}

}  // namespace

PYBIND11_MODULE(_sysfs, m) {
  m.doc() = "one-pass PCI sysfs scanner / Allocate revalidator";
  m.def("scan_pci", &scan_pci, py::arg("base_path"), py::arg("vendor"));
  m.def("revalidate", &revalidate, py::arg("base_path"),
        py::arg("addr_groups"), py::arg("vendor"),
        py::arg("want_vfio_dev"));
}
