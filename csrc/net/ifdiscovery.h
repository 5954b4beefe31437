// NIC discovery — the R5 equivalent (reference src/utils.rs:7-130).
//
// Semantics implemented (matching NCCL convention, which the reference
// approximates):
//   NCCL_SOCKET_IFNAME:  "^a,b"  exclude prefixes a,b
//                        "=a,b"  exact-match a or b
//                        "a,b"   prefix-match a or b
//                        unset   default "^docker,lo" (reference utils.rs:37)
//   NCCL_SOCKET_FAMILY:  AF_INET / AF_INET6 numeric filter, -1 = any
//                        (reference utils.rs:33-36)
// Unlike the reference (utils.rs:57-62, which always drops loopback), `lo`
// is usable when explicitly named — required for single-node loopback tests
// and matches stock NCCL behavior.

#pragma once

#include <netinet/in.h>
#include <sys/socket.h>

#include <string>
#include <vector>

namespace baguanet {

struct NetIf {
  std::string name;
  sockaddr_storage addr{};  // AF_INET or AF_INET6, port 0
  std::string pci_path;     // canonicalized /sys/class/net/<if>/device
  int speed_mbps = 10000;   // /sys/class/net/<if>/speed, default 10 Gbps
                            // (reference utils.rs:7-13)
};

// Enumerate + filter interfaces per the env semantics above.
std::vector<NetIf> find_interfaces();

// Exposed for unit tests.
bool ifname_matches(const std::string& name, const std::string& spec,
                    bool is_loopback);
int if_speed_mbps(const std::string& name);

}  // namespace baguanet
