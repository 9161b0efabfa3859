// partitiond: first-party C++ device layer over libamd_smi.so.
//
// This is the MI355X-native replacement for the reference's NVML surface
// (go-nvml cgo bindings used at internal/controller/instaslice_daemonset.go:29;
// full call-site table in SURVEY.md §2.2). One class, 13 verbs, exposed two
// ways: a pybind11 module (partitiond_pybind.cpp, in-process for the node
// agent) and a standalone JSON-over-stdio daemon (partitiond_main.cpp, for
// privilege separation: partition *sets* need root, the control plane does
// not).
//
// Design rule (north star): enumeration state is owned by the caller and
// refreshed only after a mode change — unlike the reference, which re-runs
// nvml.Init on every reconcile (instaslice_daemonset.go:112).

#pragma once

#include <amd_smi/amdsmi.h>

#include <cstdint>
#include <cstring>
#include <mutex>
#include <stdexcept>
#include <string>
#include <vector>

namespace partitiond {

struct SmiException : std::runtime_error {
  amdsmi_status_t status;
  SmiException(amdsmi_status_t st, const std::string& what)
      : std::runtime_error(what + " (amdsmi status " + std::to_string(st) + ")"),
        status(st) {}
};

inline void check(amdsmi_status_t st, const char* what) {
  if (st != AMDSMI_STATUS_SUCCESS) throw SmiException(st, what);
}

// One enumerated processor (== one schedulable partition device; a GPU in
// CPX shows up as 8 of these). Grouping into physical packages is policy and
// lives in Python (smi/native.py) where it is unit-testable.
struct ProcessorInfo {
  uint32_t index = 0;             // enumeration order (HIP device order)
  std::string uuid;               // amdsmi_get_gpu_device_uuid
  std::string asic_name;          // market name
  std::string asic_serial;       // physical-package grouping key
  uint64_t vram_total_mb = 0;
  uint64_t bdf = 0;               // raw bdf union (domain:48|bus:8|dev:5|fn:3)
  uint32_t node_id = 0xFFFFFFFF;          // kfd node
  uint32_t partition_id = 0xFFFFFFFF;     // kfd current_partition_id
  std::string compute_partition;  // "SPX"... or "" if unsupported
  std::string memory_partition;   // "NPS1"... or "" if unsupported
  uint32_t num_compute_units = 0;
};

struct ProfileInfo {
  std::string profile_type;   // "SPX" | "DPX" | "TPX" | "QPX" | "CPX"
  uint32_t num_partitions = 0;
  uint32_t profile_index = 0;
  std::vector<std::string> memory_caps;  // "NPS1"...
};

struct Metrics {
  double gfx_activity_pct = -1;
  double umc_activity_pct = -1;
  double vram_used_mb = -1;
  double socket_power_w = -1;
};

inline const char* compute_mode_name(amdsmi_compute_partition_type_t t) {
  switch (t) {
    case AMDSMI_COMPUTE_PARTITION_SPX: return "SPX";
    case AMDSMI_COMPUTE_PARTITION_DPX: return "DPX";
    case AMDSMI_COMPUTE_PARTITION_TPX: return "TPX";
    case AMDSMI_COMPUTE_PARTITION_QPX: return "QPX";
    case AMDSMI_COMPUTE_PARTITION_CPX: return "CPX";
    default: return "UNKNOWN";
  }
}

inline amdsmi_compute_partition_type_t compute_mode_from_name(const std::string& s) {
  if (s == "SPX") return AMDSMI_COMPUTE_PARTITION_SPX;
  if (s == "DPX") return AMDSMI_COMPUTE_PARTITION_DPX;
  if (s == "TPX") return AMDSMI_COMPUTE_PARTITION_TPX;
  if (s == "QPX") return AMDSMI_COMPUTE_PARTITION_QPX;
  if (s == "CPX") return AMDSMI_COMPUTE_PARTITION_CPX;
  throw std::invalid_argument("unknown compute mode: " + s);
}

inline amdsmi_memory_partition_type_t memory_mode_from_name(const std::string& s) {
  if (s == "NPS1") return AMDSMI_MEMORY_PARTITION_NPS1;
  if (s == "NPS2") return AMDSMI_MEMORY_PARTITION_NPS2;
  if (s == "NPS4") return AMDSMI_MEMORY_PARTITION_NPS4;
  if (s == "NPS8") return AMDSMI_MEMORY_PARTITION_NPS8;
  throw std::invalid_argument("unknown memory mode: " + s);
}

class Partitiond {
 public:
  Partitiond() = default;
  ~Partitiond() {
    try { shutdown(); } catch (...) {}
  }

  void init() {
    std::lock_guard<std::mutex> lk(mu_);
    if (initialized_) return;
    check(amdsmi_init(AMDSMI_INIT_AMD_GPUS), "amdsmi_init");
    initialized_ = true;
  }

  void shutdown() {
    std::lock_guard<std::mutex> lk(mu_);
    if (!initialized_) return;
    amdsmi_shut_down();
    initialized_ = false;
    handles_.clear();
  }

  // Re-walk sockets/processors. Called once at agent boot and after each
  // mode set (a CPX flip changes the processor population).
  std::vector<ProcessorInfo> enumerate() {
    std::lock_guard<std::mutex> lk(mu_);
    require_init();
    refresh_handles_locked();
    std::vector<ProcessorInfo> out;
    out.reserve(handles_.size());
    for (uint32_t i = 0; i < handles_.size(); ++i) {
      out.push_back(query_locked(i));
    }
    return out;
  }

  std::string get_compute_partition(uint32_t index) {
    std::lock_guard<std::mutex> lk(mu_);
    char buf[16] = {0};
    check(amdsmi_get_gpu_compute_partition(handle_locked(index), buf, sizeof(buf)),
          "amdsmi_get_gpu_compute_partition");
    return buf;
  }

  // Whole-GPU mode set (amdsmi.h:5799). Requires the device idle and root.
  void set_compute_partition(uint32_t index, const std::string& mode) {
    std::lock_guard<std::mutex> lk(mu_);
    check(amdsmi_set_gpu_compute_partition(handle_locked(index),
                                           compute_mode_from_name(mode)),
          "amdsmi_set_gpu_compute_partition");
    // handle population may have changed; force re-walk on next enumerate
    stale_ = true;
  }

  std::string get_memory_partition(uint32_t index) {
    std::lock_guard<std::mutex> lk(mu_);
    char buf[16] = {0};
    check(amdsmi_get_gpu_memory_partition(handle_locked(index), buf, sizeof(buf)),
          "amdsmi_get_gpu_memory_partition");
    return buf;
  }

  // NOTE: on bare metal this requires an amdgpu driver reload to complete
  // (amdsmi.h:5861) — the agent treats memory mode as sticky.
  void set_memory_partition(uint32_t index, const std::string& mode) {
    std::lock_guard<std::mutex> lk(mu_);
    check(amdsmi_set_gpu_memory_partition(handle_locked(index),
                                          memory_mode_from_name(mode)),
          "amdsmi_set_gpu_memory_partition");
    stale_ = true;
  }

  // The MODE-based memory set (amdsmi.h:5920) — distinct entry point from
  // the type-based one above; probed separately so the platform status
  // matrix covers every write variant.
  void set_memory_partition_mode(uint32_t index, const std::string& mode) {
    std::lock_guard<std::mutex> lk(mu_);
    check(amdsmi_set_gpu_memory_partition_mode(handle_locked(index),
                                               memory_mode_from_name(mode)),
          "amdsmi_set_gpu_memory_partition_mode");
    stale_ = true;
  }

  // Accelerator-partition profile catalog (amdsmi.h:5950); the MI355X analog
  // of GetGpuInstanceProfileInfo+PossiblePlacements discovery
  // (instaslice_daemonset.go:613-658).
  std::vector<ProfileInfo> get_profile_config(uint32_t index) {
    std::lock_guard<std::mutex> lk(mu_);
    amdsmi_accelerator_partition_profile_config_t cfg;
    std::memset(&cfg, 0, sizeof(cfg));
    check(amdsmi_get_gpu_accelerator_partition_profile_config(handle_locked(index), &cfg),
          "amdsmi_get_gpu_accelerator_partition_profile_config");
    std::vector<ProfileInfo> out;
    for (uint32_t i = 0; i < cfg.num_profiles && i < AMDSMI_MAX_ACCELERATOR_PROFILE; ++i) {
      const auto& p = cfg.profiles[i];
      ProfileInfo pi;
      switch (p.profile_type) {
        case AMDSMI_ACCELERATOR_PARTITION_SPX: pi.profile_type = "SPX"; break;
        case AMDSMI_ACCELERATOR_PARTITION_DPX: pi.profile_type = "DPX"; break;
        case AMDSMI_ACCELERATOR_PARTITION_TPX: pi.profile_type = "TPX"; break;
        case AMDSMI_ACCELERATOR_PARTITION_QPX: pi.profile_type = "QPX"; break;
        case AMDSMI_ACCELERATOR_PARTITION_CPX: pi.profile_type = "CPX"; break;
        default: pi.profile_type = "UNKNOWN"; break;
      }
      pi.num_partitions = p.num_partitions;
      pi.profile_index = p.profile_index;
      if (p.memory_caps.nps_flags.nps1_cap) pi.memory_caps.push_back("NPS1");
      if (p.memory_caps.nps_flags.nps2_cap) pi.memory_caps.push_back("NPS2");
      if (p.memory_caps.nps_flags.nps4_cap) pi.memory_caps.push_back("NPS4");
      if (p.memory_caps.nps_flags.nps8_cap) pi.memory_caps.push_back("NPS8");
      out.push_back(std::move(pi));
    }
    return out;
  }

  // Currently-active accelerator partition profile (amdsmi.h:5974) — the
  // read side of set_accelerator_profile, used by the write-path probe to
  // verify a flip actually landed.
  ProfileInfo get_current_profile(uint32_t index) {
    std::lock_guard<std::mutex> lk(mu_);
    amdsmi_accelerator_partition_profile_t prof;
    std::memset(&prof, 0, sizeof(prof));
    uint32_t partition_ids[AMDSMI_MAX_ACCELERATOR_PARTITIONS] = {0};
    check(amdsmi_get_gpu_accelerator_partition_profile(handle_locked(index), &prof,
                                                       partition_ids),
          "amdsmi_get_gpu_accelerator_partition_profile");
    ProfileInfo pi;
    switch (prof.profile_type) {
      case AMDSMI_ACCELERATOR_PARTITION_SPX: pi.profile_type = "SPX"; break;
      case AMDSMI_ACCELERATOR_PARTITION_DPX: pi.profile_type = "DPX"; break;
      case AMDSMI_ACCELERATOR_PARTITION_TPX: pi.profile_type = "TPX"; break;
      case AMDSMI_ACCELERATOR_PARTITION_QPX: pi.profile_type = "QPX"; break;
      case AMDSMI_ACCELERATOR_PARTITION_CPX: pi.profile_type = "CPX"; break;
      default: pi.profile_type = "UNKNOWN"; break;
    }
    pi.num_partitions = prof.num_partitions;
    pi.profile_index = prof.profile_index;
    if (prof.memory_caps.nps_flags.nps1_cap) pi.memory_caps.push_back("NPS1");
    if (prof.memory_caps.nps_flags.nps2_cap) pi.memory_caps.push_back("NPS2");
    if (prof.memory_caps.nps_flags.nps4_cap) pi.memory_caps.push_back("NPS4");
    if (prof.memory_caps.nps_flags.nps8_cap) pi.memory_caps.push_back("NPS8");
    return pi;
  }

  // Set partition layout by catalog profile_index — the forward-looking API
  // (amdsmi.h:5994); set_compute_partition is the portable one.
  void set_accelerator_profile(uint32_t index, uint32_t profile_index) {
    std::lock_guard<std::mutex> lk(mu_);
    check(amdsmi_set_gpu_accelerator_partition_profile(handle_locked(index), profile_index),
          "amdsmi_set_gpu_accelerator_partition_profile");
    stale_ = true;
  }

  // Pairwise link topology (amdsmi_topo_get_link_type/_weight, amdsmi.h:5635,
  // :5574): the placement layer uses it for xGMI-hop-aware gang scoring
  // (SURVEY.md §5 — 8 MI355X GPUs are fully connected by 7 p2p links each;
  // partial meshes and PCIe-bridged boards score worse).
  struct LinkInfo {
    uint32_t src = 0, dst = 0;
    uint64_t hops = 0;
    uint64_t weight = 0;
    std::string type;  // "XGMI" | "PCIE" | "INTERNAL" | "UNKNOWN"
  };

  std::vector<LinkInfo> get_link_topology() {
    std::lock_guard<std::mutex> lk(mu_);
    require_init();
    refresh_handles_locked();
    std::vector<LinkInfo> out;
    for (uint32_t i = 0; i < handles_.size(); ++i) {
      for (uint32_t j = 0; j < handles_.size(); ++j) {
        if (i == j) continue;
        LinkInfo li;
        li.src = i;
        li.dst = j;
        uint64_t hops = 0;
        amdsmi_link_type_t t = AMDSMI_LINK_TYPE_UNKNOWN;
        if (amdsmi_topo_get_link_type(handles_[i], handles_[j], &hops, &t) !=
            AMDSMI_STATUS_SUCCESS)
          continue;
        li.hops = hops;
        switch (t) {
          case AMDSMI_LINK_TYPE_XGMI: li.type = "XGMI"; break;
          case AMDSMI_LINK_TYPE_PCIE: li.type = "PCIE"; break;
          case AMDSMI_LINK_TYPE_INTERNAL: li.type = "INTERNAL"; break;
          default: li.type = "UNKNOWN"; break;
        }
        uint64_t w = 0;
        if (amdsmi_topo_get_link_weight(handles_[i], handles_[j], &w) ==
            AMDSMI_STATUS_SUCCESS)
          li.weight = w;
        out.push_back(std::move(li));
      }
    }
    return out;
  }

  // Counters captured around every reconfigure (north-star observability).
  Metrics get_metrics(uint32_t index) {
    std::lock_guard<std::mutex> lk(mu_);
    Metrics m;
    amdsmi_processor_handle h = handle_locked(index);
    amdsmi_engine_usage_t usage;
    if (amdsmi_get_gpu_activity(h, &usage) == AMDSMI_STATUS_SUCCESS) {
      if (usage.gfx_activity != UINT32_MAX) m.gfx_activity_pct = usage.gfx_activity;
      if (usage.umc_activity != UINT32_MAX) m.umc_activity_pct = usage.umc_activity;
    }
    uint64_t used = 0;
    if (amdsmi_get_gpu_memory_usage(h, AMDSMI_MEM_TYPE_VRAM, &used) ==
        AMDSMI_STATUS_SUCCESS) {
      m.vram_used_mb = static_cast<double>(used) / (1024.0 * 1024.0);
    }
    amdsmi_power_info_t power;
    if (amdsmi_get_power_info(h, &power) == AMDSMI_STATUS_SUCCESS) {
      if (power.socket_power != UINT64_MAX && power.socket_power != 0)
        m.socket_power_w = static_cast<double>(power.socket_power);
      else if (power.current_socket_power != UINT32_MAX)
        m.socket_power_w = power.current_socket_power;
    }
    return m;
  }

  size_t num_processors() {
    std::lock_guard<std::mutex> lk(mu_);
    require_init();
    refresh_handles_locked();
    return handles_.size();
  }

 private:
  void require_init() {
    if (!initialized_) throw std::runtime_error("partitiond: not initialized");
  }

  void refresh_handles_locked() {
    if (!handles_.empty() && !stale_) return;
    handles_.clear();
    uint32_t socket_count = 0;
    check(amdsmi_get_socket_handles(&socket_count, nullptr), "amdsmi_get_socket_handles");
    std::vector<amdsmi_socket_handle> sockets(socket_count);
    check(amdsmi_get_socket_handles(&socket_count, sockets.data()),
          "amdsmi_get_socket_handles");
    for (auto sock : sockets) {
      uint32_t dev_count = 0;
      check(amdsmi_get_processor_handles(sock, &dev_count, nullptr),
            "amdsmi_get_processor_handles");
      std::vector<amdsmi_processor_handle> procs(dev_count);
      check(amdsmi_get_processor_handles(sock, &dev_count, procs.data()),
            "amdsmi_get_processor_handles");
      for (auto p : procs) handles_.push_back(p);
    }
    stale_ = false;
  }

  amdsmi_processor_handle handle_locked(uint32_t index) {
    require_init();
    refresh_handles_locked();
    if (index >= handles_.size())
      throw std::out_of_range("processor index " + std::to_string(index) +
                              " >= " + std::to_string(handles_.size()));
    return handles_[index];
  }

  ProcessorInfo query_locked(uint32_t index) {
    ProcessorInfo info;
    info.index = index;
    amdsmi_processor_handle h = handles_[index];

    unsigned int uuid_len = AMDSMI_GPU_UUID_SIZE + 1;
    char uuid[AMDSMI_GPU_UUID_SIZE + 1] = {0};
    if (amdsmi_get_gpu_device_uuid(h, &uuid_len, uuid) == AMDSMI_STATUS_SUCCESS)
      info.uuid = uuid;

    amdsmi_asic_info_t asic;
    std::memset(&asic, 0, sizeof(asic));
    if (amdsmi_get_gpu_asic_info(h, &asic) == AMDSMI_STATUS_SUCCESS) {
      info.asic_name = asic.market_name;
      info.asic_serial = asic.asic_serial;
      if (asic.num_of_compute_units != UINT32_MAX)
        info.num_compute_units = asic.num_of_compute_units;
    }

    uint64_t total = 0;
    if (amdsmi_get_gpu_memory_total(h, AMDSMI_MEM_TYPE_VRAM, &total) ==
        AMDSMI_STATUS_SUCCESS)
      info.vram_total_mb = total / (1024ull * 1024ull);

    amdsmi_bdf_t bdf;
    if (amdsmi_get_gpu_device_bdf(h, &bdf) == AMDSMI_STATUS_SUCCESS)
      info.bdf = bdf.as_uint;

    amdsmi_kfd_info_t kfd;
    std::memset(&kfd, 0, sizeof(kfd));
    if (amdsmi_get_gpu_kfd_info(h, &kfd) == AMDSMI_STATUS_SUCCESS) {
      info.node_id = kfd.node_id;
      info.partition_id = kfd.current_partition_id;
    }

    char buf[16] = {0};
    if (amdsmi_get_gpu_compute_partition(h, buf, sizeof(buf)) == AMDSMI_STATUS_SUCCESS)
      info.compute_partition = buf;
    char mbuf[16] = {0};
    if (amdsmi_get_gpu_memory_partition(h, mbuf, sizeof(mbuf)) == AMDSMI_STATUS_SUCCESS)
      info.memory_partition = mbuf;
    return info;
  }

  std::mutex mu_;
  bool initialized_ = false;
  bool stale_ = false;
  std::vector<amdsmi_processor_handle> handles_;
};

}  // namespace partitiond
