#include "common/linux.h"

#include <dirent.h>
#include <limits.h>
#include <stdlib.h>
#include <unistd.h>

#include <cstring>
#include <fstream>
#include <sstream>

namespace glooamd {

std::vector<std::string> listInterfaces() {
  std::vector<std::string> out;
  DIR* d = opendir("/sys/class/net");
  if (d == nullptr) {
    return out;
  }
  struct dirent* e;
  while ((e = readdir(d)) != nullptr) {
    if (e->d_name[0] != '.') {
      out.emplace_back(e->d_name);
    }
  }
  closedir(d);
  return out;
}

int getInterfaceSpeedByName(const std::string& name) {
  std::ifstream f("/sys/class/net/" + name + "/speed");
  int speed = -1;
  if (f.good()) {
    f >> speed;
  }
  return f.fail() ? -1 : speed;
}

static std::string devicePath(const std::string& path) {
  char resolved[PATH_MAX];
  if (realpath(path.c_str(), resolved) == nullptr) {
    return "";
  }
  return std::string(resolved);
}

std::string interfaceToBusID(const std::string& name) {
  std::string p = devicePath("/sys/class/net/" + name + "/device");
  if (p.empty()) {
    return "";
  }
  auto slash = p.rfind('/');
  return slash == std::string::npos ? p : p.substr(slash + 1);
}

int pciDistance(const std::string& busID1, const std::string& busID2) {
  auto path = [](const std::string& busID) {
    return devicePath("/sys/bus/pci/devices/" + busID);
  };
  std::string p1 = path(busID1);
  std::string p2 = path(busID2);
  if (p1.empty() || p2.empty()) {
    return INT_MAX;
  }
  auto split = [](const std::string& p) {
    std::vector<std::string> parts;
    std::stringstream ss(p);
    std::string item;
    while (std::getline(ss, item, '/')) {
      if (!item.empty()) {
        parts.push_back(item);
      }
    }
    return parts;
  };
  auto a = split(p1);
  auto b = split(p2);
  size_t common = 0;
  while (common < a.size() && common < b.size() && a[common] == b[common]) {
    common++;
  }
  return static_cast<int>((a.size() - common) + (b.size() - common));
}

} // namespace glooamd
