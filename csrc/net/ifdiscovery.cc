#include "ifdiscovery.h"

#include <ifaddrs.h>
#include <limits.h>
#include <net/if.h>
#include <stdlib.h>
#include <string.h>

#include <cstdio>
#include <fstream>

#include "baguanet/log.h"

namespace baguanet {

int if_speed_mbps(const std::string& name) {
  std::ifstream f("/sys/class/net/" + name + "/speed");
  int speed = 0;
  if (f >> speed && speed > 0) return speed;
  return 10000;
}

static std::vector<std::string> split_csv(const std::string& s) {
  std::vector<std::string> out;
  size_t pos = 0;
  while (pos <= s.size()) {
    size_t c = s.find(',', pos);
    if (c == std::string::npos) c = s.size();
    if (c > pos) out.push_back(s.substr(pos, c - pos));
    pos = c + 1;
  }
  return out;
}

bool ifname_matches(const std::string& name, const std::string& spec,
                    bool is_loopback) {
  if (spec.empty()) return !is_loopback;
  if (spec[0] == '^') {
    for (auto& p : split_csv(spec.substr(1)))
      if (name.compare(0, p.size(), p) == 0) return false;
    return true;  // caller still applies the loopback opt-in rule
  }
  bool exact = spec[0] == '=';
  for (auto& p : split_csv(exact ? spec.substr(1) : spec)) {
    if (exact ? (name == p) : (name.compare(0, p.size(), p) == 0)) return true;
  }
  return false;
}

std::vector<NetIf> find_interfaces() {
  const char* env = getenv("NCCL_SOCKET_IFNAME");
  std::string spec = env ? env : "^docker,lo";
  int family = -1;
  if (const char* f = getenv("NCCL_SOCKET_FAMILY")) family = atoi(f);

  std::vector<NetIf> out;
  struct ifaddrs* ifa0 = nullptr;
  if (getifaddrs(&ifa0) != 0) {
    BNET_WARN("getifaddrs failed: %s", strerror(errno));
    return out;
  }
  for (struct ifaddrs* ifa = ifa0; ifa; ifa = ifa->ifa_next) {
    if (!ifa->ifa_addr) continue;
    int af = ifa->ifa_addr->sa_family;
    if (af != AF_INET && af != AF_INET6) continue;
    if (family != -1 && af != family) continue;
    bool loop = (ifa->ifa_flags & IFF_LOOPBACK) != 0;
    std::string name = ifa->ifa_name;
    if (!ifname_matches(name, spec, loop)) continue;
    // loopback only when the spec names it explicitly (not via "^..." pass)
    if (loop && (spec.empty() || spec[0] == '^')) continue;
    bool dup = false;
    for (auto& d : out)
      if (d.name == name) { dup = true; break; }
    if (dup) continue;  // one address per interface (reference utils.rs:65-71)

    NetIf d;
    d.name = name;
    memcpy(&d.addr, ifa->ifa_addr,
           af == AF_INET ? sizeof(sockaddr_in) : sizeof(sockaddr_in6));
    char buf[PATH_MAX];
    std::string dev = "/sys/class/net/" + name + "/device";
    if (realpath(dev.c_str(), buf)) d.pci_path = buf;
    d.speed_mbps = if_speed_mbps(name);
    out.push_back(std::move(d));
  }
  freeifaddrs(ifa0);

  // Fallback: a bare container may only have loopback — better to expose it
  // than to report zero devices (stock NCCL does the same).
  if (out.empty()) {
    struct ifaddrs* i2 = nullptr;
    if (getifaddrs(&i2) == 0) {
      for (struct ifaddrs* ifa = i2; ifa; ifa = ifa->ifa_next) {
        if (!ifa->ifa_addr || ifa->ifa_addr->sa_family != AF_INET) continue;
        if (!(ifa->ifa_flags & IFF_LOOPBACK)) continue;
        NetIf d;
        d.name = ifa->ifa_name;
        memcpy(&d.addr, ifa->ifa_addr, sizeof(sockaddr_in));
        d.speed_mbps = 10000;
        out.push_back(std::move(d));
        break;
      }
      freeifaddrs(i2);
    }
  }
  return out;
}

}  // namespace baguanet
