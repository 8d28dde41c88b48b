// Kubelet DevicePlugin v1beta1 message encode/decode (wire-level).
//
// Message/field numbers follow the stable upstream API
// (k8s.io/kubelet/pkg/apis/deviceplugin/v1beta1/api.proto); the reference
// stack's device plugin speaks exactly this protocol to kubelet
// (/root/reference/README.md:105-126). Service paths:
//   /v1beta1.Registration/Register
//   /v1beta1.DevicePlugin/{GetDevicePluginOptions,ListAndWatch,
//                          GetPreferredAllocation,Allocate,PreStartContainer}

#pragma once

#include <map>
#include <string>
#include <vector>

#include "../grpc/proto.h"

namespace k3samd::dp {

inline constexpr char kVersion[] = "v1beta1";
inline constexpr char kRegisterPath[] = "/v1beta1.Registration/Register";
inline constexpr char kOptionsPath[] =
    "/v1beta1.DevicePlugin/GetDevicePluginOptions";
inline constexpr char kListAndWatchPath[] =
    "/v1beta1.DevicePlugin/ListAndWatch";
inline constexpr char kPreferredPath[] =
    "/v1beta1.DevicePlugin/GetPreferredAllocation";
inline constexpr char kAllocatePath[] = "/v1beta1.DevicePlugin/Allocate";
inline constexpr char kPreStartPath[] =
    "/v1beta1.DevicePlugin/PreStartContainer";

inline constexpr char kHealthy[] = "Healthy";
inline constexpr char kUnhealthy[] = "Unhealthy";

struct DevicePluginOptions {
  bool pre_start_required = false;
  bool get_preferred_allocation_available = true;

  std::string encode() const {
    std::string out;
    pb::put_bool(out, 1, pre_start_required);
    pb::put_bool(out, 2, get_preferred_allocation_available);
    return out;
  }
};

struct Device {
  std::string id;       // field 1
  std::string health;   // field 2
  int numa_node = -1;   // field 3: TopologyInfo{ nodes[]{ ID } }

  std::string encode() const {
    std::string out;
    pb::put_string(out, 1, id);
    pb::put_string(out, 2, health);
    if (numa_node >= 0) {
      std::string numa, topo;
      pb::put_uint(numa, 1, (uint64_t)numa_node);
      pb::put_bytes(topo, 1, numa);
      pb::put_bytes(out, 3, topo);
    }
    return out;
  }
};

inline std::string encode_list_and_watch(const std::vector<Device>& devs) {
  std::string out;
  for (const auto& d : devs) pb::put_bytes(out, 1, d.encode());
  return out;
}

struct RegisterRequest {
  std::string version = kVersion;  // 1
  std::string endpoint;            // 2: socket basename
  std::string resource_name;       // 3
  DevicePluginOptions options;     // 4

  std::string encode() const {
    std::string out;
    pb::put_string(out, 1, version);
    pb::put_string(out, 2, endpoint);
    pb::put_string(out, 3, resource_name);
    pb::put_bytes(out, 4, options.encode());
    return out;
  }

  static bool decode(std::string_view buf, RegisterRequest& r) {
    pb::Reader rd(buf);
    int f, w;
    uint64_t u;
    std::string_view d;
    while (rd.next(f, w, u, d)) {
      if (f == 1 && w == pb::kLenDelim) r.version.assign(d);
      if (f == 2 && w == pb::kLenDelim) r.endpoint.assign(d);
      if (f == 3 && w == pb::kLenDelim) r.resource_name.assign(d);
    }
    return rd.ok();
  }
};

// AllocateRequest: container_requests(1) -> devices_ids(1)
inline bool decode_allocate_request(std::string_view buf,
                                    std::vector<std::vector<std::string>>& out) {
  pb::Reader rd(buf);
  int f, w;
  uint64_t u;
  std::string_view d;
  while (rd.next(f, w, u, d)) {
    if (f == 1 && w == pb::kLenDelim) {
      std::vector<std::string> ids;
      pb::Reader cr(d);
      int f2, w2;
      uint64_t u2;
      std::string_view d2;
      while (cr.next(f2, w2, u2, d2))
        if (f2 == 1 && w2 == pb::kLenDelim) ids.emplace_back(d2);
      if (!cr.ok()) return false;
      out.push_back(std::move(ids));
    }
  }
  return rd.ok();
}

struct Mount {
  std::string container_path, host_path;
  bool read_only = false;
  std::string encode() const {
    std::string out;
    pb::put_string(out, 1, container_path);
    pb::put_string(out, 2, host_path);
    pb::put_bool(out, 3, read_only);
    return out;
  }
};

struct DeviceSpec {
  std::string container_path, host_path, permissions;
  std::string encode() const {
    std::string out;
    pb::put_string(out, 1, container_path);
    pb::put_string(out, 2, host_path);
    pb::put_string(out, 3, permissions);
    return out;
  }
};

struct ContainerAllocateResponse {
  std::map<std::string, std::string> envs;         // 1
  std::vector<Mount> mounts;                       // 2
  std::vector<DeviceSpec> devices;                 // 3
  std::map<std::string, std::string> annotations;  // 4
  std::vector<std::string> cdi_devices;            // 5: CDIDevice{name=1}

  std::string encode() const {
    std::string out;
    for (auto& [k, v] : envs) pb::put_map_entry(out, 1, k, v);
    for (auto& m : mounts) pb::put_bytes(out, 2, m.encode());
    for (auto& d : devices) pb::put_bytes(out, 3, d.encode());
    for (auto& [k, v] : annotations) pb::put_map_entry(out, 4, k, v);
    for (auto& n : cdi_devices) {
      std::string cd;
      pb::put_string(cd, 1, n);
      pb::put_bytes(out, 5, cd);
    }
    return out;
  }
};

inline std::string encode_allocate_response(
    const std::vector<ContainerAllocateResponse>& crs) {
  std::string out;
  for (auto& cr : crs) pb::put_bytes(out, 1, cr.encode());
  return out;
}

// PreferredAllocationRequest:
//   container_requests(1) -> available_deviceIDs(1), must_include(2), size(3)
struct PreferredRequest {
  std::vector<std::string> available;
  std::vector<std::string> must_include;
  int size = 0;
};

inline bool decode_preferred_request(std::string_view buf,
                                     std::vector<PreferredRequest>& out) {
  pb::Reader rd(buf);
  int f, w;
  uint64_t u;
  std::string_view d;
  while (rd.next(f, w, u, d)) {
    if (f == 1 && w == pb::kLenDelim) {
      PreferredRequest pr;
      pb::Reader cr(d);
      int f2, w2;
      uint64_t u2;
      std::string_view d2;
      while (cr.next(f2, w2, u2, d2)) {
        if (f2 == 1 && w2 == pb::kLenDelim) pr.available.emplace_back(d2);
        if (f2 == 2 && w2 == pb::kLenDelim) pr.must_include.emplace_back(d2);
        if (f2 == 3 && w2 == pb::kVarint) pr.size = (int)u2;
      }
      if (!cr.ok()) return false;
      out.push_back(std::move(pr));
    }
  }
  return rd.ok();
}

inline std::string encode_preferred_response(
    const std::vector<std::vector<std::string>>& per_container) {
  std::string out;
  for (auto& ids : per_container) {
    std::string cr;
    for (auto& id : ids) pb::put_string(cr, 1, id);
    pb::put_bytes(out, 1, cr);
  }
  return out;
}

}  // namespace k3samd::dp
