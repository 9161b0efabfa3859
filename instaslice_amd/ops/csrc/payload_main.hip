// instaslice-payload: standalone workload binary run *inside* partitions.
//
// The MI355X analog of the reference's cuda-vectoradd sample container
// (samples/test-pod.yaml:12): the node agent / e2e tests exec this binary
// with ROCR_VISIBLE_DEVICES set from the pod's ConfigMap, so it exercises
// exactly the device set a real pod would see. Prints one JSON line.
//
//   instaslice-payload info
//   instaslice-payload vecadd [N]
//   instaslice-payload membw [BYTES] [ITERS]
//   instaslice-payload busy [MS]
//   instaslice-payload census
//   instaslice-payload serve      (persistent worker: commands on stdin,
//                                  one JSON line per command on stdout —
//                                  HIP init is paid once, so a warm pool
//                                  can run a kernel per pod lifecycle at
//                                  production rate; NOTE the live process
//                                  holds the device, which blocks
//                                  partition mode flips: serve is for
//                                  static-partitioning phases only)
//
// Build: see build_native.py (hipcc --offload-arch=gfx950).

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <iostream>
#include <sstream>
#include <string>

#include "payload_core.hpp"

using namespace payload;

int main(int argc, char** argv) {
  const char* cmd = argc > 1 ? argv[1] : "info";
  try {
    if (std::strcmp(cmd, "info") == 0) {
      int n = device_count();
      std::printf("{\"ok\": true, \"devices\": %d", n);
      if (n > 0) {
        DeviceInfo i = get_device_info(0);
        std::printf(
            ", \"name\": \"%s\", \"gcn_arch\": \"%s\", \"cu_count\": %d, "
            "\"total_mem_gb\": %.1f, \"xcd_count_visible\": %d",
            i.name.c_str(), i.gcn_arch.c_str(), i.cu_count, i.total_mem_gb,
            i.xcd_count_visible);
      }
      std::printf("}\n");
    } else if (std::strcmp(cmd, "vecadd") == 0) {
      size_t n = argc > 2 ? std::strtoull(argv[2], nullptr, 10) : (1 << 24);
      double err = run_vecadd(n);
      std::printf("{\"ok\": %s, \"cmd\": \"vecadd\", \"n\": %zu, \"max_err\": %g}\n",
                  err == 0.0 ? "true" : "false", n, err);
      return err == 0.0 ? 0 : 1;
    } else if (std::strcmp(cmd, "membw") == 0) {
      size_t bytes = argc > 2 ? std::strtoull(argv[2], nullptr, 10) : (size_t{1} << 30);
      int iters = argc > 3 ? std::atoi(argv[3]) : 10;
      int blocks = argc > 4 ? std::atoi(argv[4]) : 0;
      bool nt = argc > 5 && std::atoi(argv[5]) != 0;
      double gbs = run_membw(bytes, iters, 0, blocks, nt);
      std::printf(
          "{\"ok\": true, \"cmd\": \"membw\", \"bytes\": %zu, \"iters\": %d, "
          "\"blocks\": %d, \"nontemporal\": %s, \"gb_per_s\": %.1f}\n",
          bytes, iters, blocks, nt ? "true" : "false", gbs);
    } else if (std::strcmp(cmd, "busy") == 0) {
      double ms = argc > 2 ? std::atof(argv[2]) : 100.0;
      run_busy(ms);
      std::printf("{\"ok\": true, \"cmd\": \"busy\", \"ms\": %.1f}\n", ms);
    } else if (std::strcmp(cmd, "census") == 0) {
      auto c = run_xcd_census();
      std::printf("{\"ok\": true, \"cmd\": \"census\", \"per_xcd\": [");
      for (int i = 0; i < 8; ++i) std::printf("%s%u", i ? ", " : "", c[i]);
      int xcds = 0;
      for (unsigned v : c)
        if (v) ++xcds;
      std::printf("], \"xcds_visible\": %d}\n", xcds);
    } else if (std::strcmp(cmd, "serve") == 0) {
      // warm-pool worker: one line per request, one JSON line per reply
      //   vecadd <n> | busy <ms> | ping | quit
      std::string line;
      while (std::getline(std::cin, line)) {
        std::istringstream iss(line);
        std::string verb;
        iss >> verb;
        if (verb.empty()) continue;
        try {
          if (verb == "quit") { std::printf("{\"ok\": true}\n"); break; }
          if (verb == "ping") {
            std::printf("{\"ok\": true, \"cmd\": \"ping\"}\n");
          } else if (verb == "vecadd") {
            size_t n = 1 << 20;
            iss >> n;
            double err = run_vecadd(n);
            std::printf(
                "{\"ok\": %s, \"cmd\": \"vecadd\", \"n\": %zu, \"max_err\": %g}\n",
                err == 0.0 ? "true" : "false", n, err);
          } else if (verb == "busy") {
            double ms = 10.0;
            iss >> ms;
            run_busy(ms);
            std::printf("{\"ok\": true, \"cmd\": \"busy\", \"ms\": %.1f}\n", ms);
          } else {
            std::printf("{\"ok\": false, \"error\": \"unknown verb\"}\n");
          }
        } catch (const std::exception& e) {
          std::printf("{\"ok\": false, \"error\": \"%s\"}\n", e.what());
        }
        std::fflush(stdout);
      }
    } else {
      std::fprintf(stderr, "unknown command: %s\n", cmd);
      return 2;
    }
  } catch (const std::exception& e) {
    std::printf("{\"ok\": false, \"error\": \"%s\"}\n", e.what());
    return 1;
  }
  return 0;
}
