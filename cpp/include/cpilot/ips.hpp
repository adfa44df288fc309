// Advertise-IP selection from interface specs: "eth0", "eth0:inet6",
// "eth0[1]", CIDR ("10.0.0.0/16"), "inet"/"inet6" wildcards (skip
// loopback), "static:<ip>". Interfaces are sorted by name then IP bytes so
// selection is deterministic. Default spec list: ["eth0:inet", "inet"].
// Parity: /root/reference/config/services/ips.go:31-310.
#pragma once

#include <string>
#include <vector>

namespace cpilot {

struct InterfaceIP {
  std::string name;
  std::string ip;     // printable
  bool ipv6 = false;
  unsigned char bytes[16] = {0};  // 16-byte normalized form for sorting/CIDR
};

// Enumerate host interface IPs, sorted by (name, ip-bytes). Overridable for
// tests via injection.
std::vector<InterfaceIP> getInterfaceIPs();

// Pick the first IP matching the spec list; empty list uses the default.
// Returns false + err if no spec matches or a spec fails to parse.
bool getIP(const std::vector<std::string>& specs,
           const std::vector<InterfaceIP>& ifaceIPs, std::string* out,
           std::string* err);

// Convenience: getIP over the live host interfaces.
bool getIP(const std::vector<std::string>& specs, std::string* out,
           std::string* err);

// Service-name validation: ^[a-z][a-zA-Z0-9-]+$
// (config/services/names.go:13-21)
bool validateServiceName(const std::string& name, std::string* err);

}  // namespace cpilot
