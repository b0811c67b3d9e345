// Linux topology helpers: NIC speed, PCI bus IDs, PCI distance.
// Capability parity with reference gloo/common/linux.cc:126-220
// (interfaceToBusID, getInterfaceSpeedByName, pciDistance) — used to
// pick the NIC/GPU pairing closest in the PCI tree.
#pragma once

#include <string>
#include <vector>

namespace glooamd {

// Network interfaces on this host.
std::vector<std::string> listInterfaces();

// Link speed in Mb/s from /sys/class/net/<name>/speed (-1 if unknown).
int getInterfaceSpeedByName(const std::string& name);

// PCI bus id ("0000:c1:00.0") for a network interface ("" if virtual).
std::string interfaceToBusID(const std::string& name);

// Number of differing path components between two PCI devices' sysfs
// paths — smaller means topologically closer.
int pciDistance(const std::string& busID1, const std::string& busID2);

} // namespace glooamd
