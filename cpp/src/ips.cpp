#include "cpilot/ips.hpp"

#include <arpa/inet.h>
#include <ifaddrs.h>
#include <netinet/in.h>
#include <sys/socket.h>

#include <algorithm>
#include <cctype>
#include <cstring>
#include <regex>

namespace cpilot {

namespace {

bool parseIPBytes(const std::string& ip, unsigned char out[16], bool* ipv6) {
  struct in_addr a4;
  if (inet_pton(AF_INET, ip.c_str(), &a4) == 1) {
    // IPv4-mapped 16-byte form (::ffff:a.b.c.d) like Go's To16()
    memset(out, 0, 16);
    out[10] = 0xff;
    out[11] = 0xff;
    memcpy(out + 12, &a4, 4);
    *ipv6 = false;
    return true;
  }
  struct in6_addr a6;
  if (inet_pton(AF_INET6, ip.c_str(), &a6) == 1) {
    memcpy(out, &a6, 16);
    *ipv6 = true;
    return true;
  }
  return false;
}

bool isLoopback(const InterfaceIP& iip) {
  if (!iip.ipv6) return iip.bytes[12] == 127;
  static const unsigned char lo[16] = {0, 0, 0, 0, 0, 0, 0, 0,
                                       0, 0, 0, 0, 0, 0, 0, 1};
  return memcmp(iip.bytes, lo, 16) == 0;
}

struct Spec {
  enum class Kind { Inet, Index, Cidr, Static } kind = Kind::Inet;
  std::string name;  // "*" for wildcard
  bool ipv6 = false;
  int index = 0;
  unsigned char netBytes[16] = {0};
  int prefixLen = 0;
  bool netIsV6 = false;
  std::string staticIP;

  bool match(int idx, const InterfaceIP& iip) const {
    switch (kind) {
      case Kind::Static:
        return false;
      case Kind::Inet:
        if (name != "*" && name != iip.name) return false;
        if (name == "*" && isLoopback(iip)) return false;
        return ipv6 == iip.ipv6;
      case Kind::Index:
        return name == iip.name && index == idx;
      case Kind::Cidr: {
        if (netIsV6 != iip.ipv6) return false;
        // netBytes holds the normalized 16-byte network (v4 prefixes are
        // stored v4-mapped with prefixLen offset by 96)
        for (int b = 0; b < prefixLen; b++) {
          int byteIdx = b / 8, bit = 7 - (b % 8);
          if (((iip.bytes[byteIdx] >> bit) & 1) !=
              ((netBytes[byteIdx] >> bit) & 1))
            return false;
        }
        return true;
      }
    }
    return false;
  }
};

bool parseSpec(const std::string& spec, Spec* out, std::string* err) {
  if (spec == "inet") {
    out->kind = Spec::Kind::Inet;
    out->name = "*";
    out->ipv6 = false;
    return true;
  }
  if (spec == "inet6") {
    out->kind = Spec::Kind::Inet;
    out->name = "*";
    out->ipv6 = true;
    return true;
  }
  if (spec.rfind("static:", 0) == 0) {
    std::string ip = spec.substr(7);
    unsigned char bytes[16];
    bool v6;
    if (!parseIPBytes(ip, bytes, &v6)) {
      *err = "Unable to parse static ip " + ip + " in " + spec;
      return false;
    }
    out->kind = Spec::Kind::Static;
    out->staticIP = ip;
    return true;
  }
  static const std::regex ifaceRe(
      R"(^(\w+)(?:(?:\[(\d+)\])|(?::(inet6?)))?$)");
  std::smatch m;
  if (std::regex_match(spec, m, ifaceRe)) {
    out->name = m[1];
    if (m[2].matched) {
      out->kind = Spec::Kind::Index;
      out->index = atoi(m[2].str().c_str());
      return true;
    }
    out->kind = Spec::Kind::Inet;
    out->ipv6 = (m[3].matched && m[3].str() == "inet6");
    return true;
  }
  // CIDR
  size_t slash = spec.find('/');
  if (slash != std::string::npos) {
    std::string ip = spec.substr(0, slash);
    int prefix = atoi(spec.substr(slash + 1).c_str());
    unsigned char bytes[16];
    bool v6;
    if (parseIPBytes(ip, bytes, &v6)) {
      out->kind = Spec::Kind::Cidr;
      memcpy(out->netBytes, bytes, 16);
      out->netIsV6 = v6;
      out->prefixLen = v6 ? prefix : prefix + 96;
      // mask the network bytes
      for (int b = out->prefixLen; b < 128; b++)
        out->netBytes[b / 8] &= ~(1 << (7 - (b % 8)));
      // restore the v4-mapped prefix for v4 networks
      if (!v6) {
        memset(out->netBytes, 0, 10);
        out->netBytes[10] = 0xff;
        out->netBytes[11] = 0xff;
      }
      return true;
    }
  }
  *err = "Unable to parse interface spec: " + spec;
  return false;
}

}  // namespace

std::vector<InterfaceIP> getInterfaceIPs() {
  std::vector<InterfaceIP> out;
  struct ifaddrs* ifaddr = nullptr;
  if (getifaddrs(&ifaddr) != 0) return out;
  for (struct ifaddrs* ifa = ifaddr; ifa; ifa = ifa->ifa_next) {
    if (!ifa->ifa_addr) continue;
    InterfaceIP iip;
    iip.name = ifa->ifa_name;
    char buf[INET6_ADDRSTRLEN] = {0};
    if (ifa->ifa_addr->sa_family == AF_INET) {
      auto* sin = (struct sockaddr_in*)ifa->ifa_addr;
      inet_ntop(AF_INET, &sin->sin_addr, buf, sizeof(buf));
    } else if (ifa->ifa_addr->sa_family == AF_INET6) {
      auto* sin6 = (struct sockaddr_in6*)ifa->ifa_addr;
      inet_ntop(AF_INET6, &sin6->sin6_addr, buf, sizeof(buf));
    } else {
      continue;
    }
    iip.ip = buf;
    // strip scope id from link-local (fe80::1%eth0)
    size_t pct = iip.ip.find('%');
    if (pct != std::string::npos) iip.ip = iip.ip.substr(0, pct);
    if (!parseIPBytes(iip.ip, iip.bytes, &iip.ipv6)) continue;
    out.push_back(iip);
  }
  freeifaddrs(ifaddr);
  std::stable_sort(out.begin(), out.end(),
                   [](const InterfaceIP& a, const InterfaceIP& b) {
                     if (a.name != b.name) return a.name < b.name;
                     return memcmp(a.bytes, b.bytes, 16) < 0;
                   });
  return out;
}

bool getIP(const std::vector<std::string>& specList,
           const std::vector<InterfaceIP>& ifaceIPs, std::string* out,
           std::string* err) {
  std::vector<std::string> specs = specList;
  if (specs.empty()) specs = {"eth0:inet", "inet"};

  std::vector<Spec> parsed;
  std::vector<std::string> errors;
  for (auto& s : specs) {
    Spec spec;
    std::string perr;
    if (!parseSpec(s, &spec, &perr)) {
      errors.push_back(perr);
      continue;
    }
    parsed.push_back(spec);
  }
  if (!errors.empty()) {
    std::string joined;
    for (size_t i = 0; i < errors.size(); i++) {
      if (i) joined += "\n";
      joined += errors[i];
    }
    *err = joined;
    return false;
  }

  for (auto& spec : parsed) {
    if (spec.kind == Spec::Kind::Static) {
      *out = spec.staticIP;
      return true;
    }
    int index = 0;
    std::string iface;
    for (auto& iip : ifaceIPs) {
      if (iface != iip.name) {
        index = 0;
        iface = iip.name;
      } else {
        index++;
      }
      if (spec.match(index, iip)) {
        *out = iip.ip;
        return true;
      }
    }
  }
  *err = "none of the interface specifications were able to match";
  return false;
}

bool getIP(const std::vector<std::string>& specs, std::string* out,
           std::string* err) {
  return getIP(specs, getInterfaceIPs(), out, err);
}

bool validateServiceName(const std::string& name, std::string* err) {
  if (name.empty()) {
    *err = "'name' must not be blank";
    return false;
  }
  static const std::regex re(R"(^[a-z][a-zA-Z0-9\-]+$)");
  if (!std::regex_match(name, re)) {
    *err =
        "service names must be alphanumeric with dashes to comply with "
        "service discovery";
    return false;
  }
  return true;
}

}  // namespace cpilot
