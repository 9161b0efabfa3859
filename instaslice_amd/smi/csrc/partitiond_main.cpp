// partitiond: standalone device daemon, JSON-lines over stdio.
//
// Runs the 13-verb device layer (partitiond_core.hpp) in its own (root)
// process so the control plane can stay unprivileged — amdsmi partition sets
// require root (amdsmi.h:5791 'function requires root access'). Protocol:
// one request per line, `verb [args...]`; one JSON object per line back.
//
//   init | shutdown | enumerate | num
//   get_compute <idx> | set_compute <idx> <SPX|DPX|QPX|CPX>
//   get_memory <idx>  | set_memory <idx> <NPS1|NPS2|NPS4>
//   profiles <idx>    | set_profile <idx> <profile_index>
//   current_profile <idx> | topology | metrics <idx> | quit
//
// Build: see build_native.py (amdclang++ -lamd_smi).

#include <iostream>
#include <sstream>
#include <string>
#include <vector>

#include "partitiond_core.hpp"

using namespace partitiond;

static std::string json_escape(const std::string& s) {
  std::string out;
  for (char c : s) {
    if (c == '"' || c == '\\') { out += '\\'; out += c; }
    else if (c == '\n') out += "\\n";
    else if (static_cast<unsigned char>(c) >= 0x20) out += c;
  }
  return out;
}

static void reply_ok(const std::string& body = "null") {
  std::cout << "{\"ok\": true, \"result\": " << body << "}" << std::endl;
}

static void reply_err(const std::string& msg, int status = -1) {
  std::cout << "{\"ok\": false, \"error\": \"" << json_escape(msg)
            << "\", \"status\": " << status << "}" << std::endl;
}

static std::string processor_json(const ProcessorInfo& p) {
  std::ostringstream o;
  o << "{\"index\": " << p.index
    << ", \"uuid\": \"" << json_escape(p.uuid) << "\""
    << ", \"asic_name\": \"" << json_escape(p.asic_name) << "\""
    << ", \"asic_serial\": \"" << json_escape(p.asic_serial) << "\""
    << ", \"vram_total_mb\": " << p.vram_total_mb
    << ", \"bdf\": " << p.bdf
    << ", \"node_id\": " << p.node_id
    << ", \"partition_id\": " << p.partition_id
    << ", \"compute_partition\": \"" << json_escape(p.compute_partition) << "\""
    << ", \"memory_partition\": \"" << json_escape(p.memory_partition) << "\""
    << ", \"num_compute_units\": " << p.num_compute_units << "}";
  return o.str();
}

int main() {
  Partitiond d;
  std::string line;
  while (std::getline(std::cin, line)) {
    std::istringstream iss(line);
    std::string verb;
    iss >> verb;
    if (verb.empty()) continue;
    try {
      if (verb == "quit") { reply_ok(); break; }
      else if (verb == "init") { d.init(); reply_ok(); }
      else if (verb == "shutdown") { d.shutdown(); reply_ok(); }
      else if (verb == "num") { reply_ok(std::to_string(d.num_processors())); }
      else if (verb == "enumerate") {
        std::ostringstream o;
        o << "[";
        bool first = true;
        for (const auto& p : d.enumerate()) {
          if (!first) o << ", ";
          first = false;
          o << processor_json(p);
        }
        o << "]";
        reply_ok(o.str());
      } else if (verb == "get_compute" || verb == "get_memory") {
        uint32_t idx; iss >> idx;
        std::string v = (verb == "get_compute") ? d.get_compute_partition(idx)
                                                : d.get_memory_partition(idx);
        reply_ok("\"" + json_escape(v) + "\"");
      } else if (verb == "set_compute" || verb == "set_memory" ||
                 verb == "set_memory_mode") {
        uint32_t idx; std::string mode; iss >> idx >> mode;
        if (verb == "set_compute") d.set_compute_partition(idx, mode);
        else if (verb == "set_memory") d.set_memory_partition(idx, mode);
        else d.set_memory_partition_mode(idx, mode);
        reply_ok();
      } else if (verb == "set_profile") {
        uint32_t idx, prof; iss >> idx >> prof;
        d.set_accelerator_profile(idx, prof);
        reply_ok();
      } else if (verb == "current_profile") {
        uint32_t idx; iss >> idx;
        ProfileInfo pr = d.get_current_profile(idx);
        std::ostringstream o;
        o << "{\"profile_type\": \"" << pr.profile_type
          << "\", \"num_partitions\": " << pr.num_partitions
          << ", \"profile_index\": " << pr.profile_index << "}";
        reply_ok(o.str());
      } else if (verb == "profiles") {
        uint32_t idx; iss >> idx;
        std::ostringstream o;
        o << "[";
        bool first = true;
        for (const auto& pr : d.get_profile_config(idx)) {
          if (!first) o << ", ";
          first = false;
          o << "{\"profile_type\": \"" << pr.profile_type
            << "\", \"num_partitions\": " << pr.num_partitions
            << ", \"profile_index\": " << pr.profile_index
            << ", \"memory_caps\": [";
          for (size_t i = 0; i < pr.memory_caps.size(); ++i) {
            if (i) o << ", ";
            o << "\"" << pr.memory_caps[i] << "\"";
          }
          o << "]}";
        }
        o << "]";
        reply_ok(o.str());
      } else if (verb == "topology") {
        std::ostringstream o;
        o << "[";
        bool first = true;
        for (const auto& li : d.get_link_topology()) {
          if (!first) o << ", ";
          first = false;
          o << "{\"src\": " << li.src << ", \"dst\": " << li.dst
            << ", \"hops\": " << li.hops << ", \"weight\": " << li.weight
            << ", \"type\": \"" << li.type << "\"}";
        }
        o << "]";
        reply_ok(o.str());
      } else if (verb == "metrics") {
        uint32_t idx; iss >> idx;
        Metrics mt = d.get_metrics(idx);
        std::ostringstream o;
        o << "{\"gfx_activity_pct\": " << mt.gfx_activity_pct
          << ", \"umc_activity_pct\": " << mt.umc_activity_pct
          << ", \"vram_used_mb\": " << mt.vram_used_mb
          << ", \"socket_power_w\": " << mt.socket_power_w << "}";
        reply_ok(o.str());
      } else {
        reply_err("unknown verb: " + verb);
      }
    } catch (const SmiException& e) {
      reply_err(e.what(), static_cast<int>(e.status));
    } catch (const std::exception& e) {
      reply_err(e.what());
    }
  }
  return 0;
}
