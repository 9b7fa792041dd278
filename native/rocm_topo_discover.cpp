// rocm-topo-discover: emit the scheduler's physicalCells YAML for this node
// from measured ROCm SMI facts (GPU count, VRAM bytes, xGMI connectivity,
// per-link weight and bandwidth).
//
// Replaces the reference's hand-transcribed cellTypes YAML from
// `nvidia-smi topo --matrix` (reference doc/user-manual.md:44-72) with
// measured CDNA4 topology: MI355X -> xGMI pair -> quad -> 8-GPU node, with
// per-GPU HBM capacity and the per-link xGMI table (gbps, healthy) as
// first-class cell attributes the scheduler consumes directly
// (topo/discover.py -> core xgmiLinks).
//
// The pair/quad GROUPING is derived from the measured link-weight matrix:
// each GPU pairs with its strongest-linked unpaired peer, pairs merge into
// quads by strongest inter-pair weight. On a fully-symmetric mesh (the
// MI355X node: 7 equal links per GPU) this reduces to the positional 0..7
// grouping; on asymmetric platforms (doubled links between adjacent pairs)
// the measured grouping wins.
//
// Build: hipcc -O2 native/rocm_topo_discover.cpp -I/opt/rocm/include \
//        -L/opt/rocm/lib -lrocm_smi64 -o native/rocm-topo-discover
#include <rocm_smi/rocm_smi.h>
#include <unistd.h>

#include <algorithm>
#include <cstdio>
#include <cstring>
#include <string>
#include <vector>

static const char* kIndent = "  ";

int main(int argc, char** argv) {
  const char* nodeNameArg = nullptr;
  for (int i = 1; i < argc - 1; i++) {
    if (!strcmp(argv[i], "--node-name")) nodeNameArg = argv[i + 1];
  }
  char hostname[256] = "unknown-node";
  if (nodeNameArg) {
    snprintf(hostname, sizeof(hostname), "%s", nodeNameArg);
  } else {
    gethostname(hostname, sizeof(hostname) - 1);
  }

  rsmi_status_t st = rsmi_init(0);
  if (st != RSMI_STATUS_SUCCESS) {
    fprintf(stderr, "rsmi_init failed (%d): no AMD GPUs visible?\n", (int)st);
    return 2;
  }
  uint32_t n = 0;
  rsmi_num_monitor_devices(&n);
  if (n == 0) {
    fprintf(stderr, "no GPUs found\n");
    rsmi_shut_down();
    return 3;
  }

  std::vector<uint64_t> vram(n, 0);
  for (uint32_t i = 0; i < n; i++) {
    rsmi_dev_memory_total_get(i, RSMI_MEM_TYPE_VRAM, &vram[i]);
  }
  // xGMI connectivity + per-link weight + min/max bandwidth (MB/s)
  std::vector<std::vector<bool>> xgmi(n, std::vector<bool>(n, false));
  std::vector<std::vector<uint64_t>> weight(n, std::vector<uint64_t>(n, 0));
  std::vector<std::vector<uint64_t>> bwMax(n, std::vector<uint64_t>(n, 0));
  uint32_t xgmiLinks = 0;
  for (uint32_t i = 0; i < n; i++) {
    for (uint32_t j = 0; j < n; j++) {
      if (i == j) continue;
      uint64_t hops = 0;
      RSMI_IO_LINK_TYPE type = RSMI_IOLINK_TYPE_UNDEFINED;
      if (rsmi_topo_get_link_type(i, j, &hops, &type) == RSMI_STATUS_SUCCESS &&
          type == RSMI_IOLINK_TYPE_XGMI) {
        xgmi[i][j] = true;
        xgmiLinks++;
        uint64_t w = 0;
        if (rsmi_topo_get_link_weight(i, j, &w) == RSMI_STATUS_SUCCESS) weight[i][j] = w;
        uint64_t bmin = 0, bmax = 0;
        if (rsmi_minmax_bandwidth_get(i, j, &bmin, &bmax) == RSMI_STATUS_SUCCESS) {
          bwMax[i][j] = bmax;
        }
      }
    }
  }
  bool fullMesh = true;
  for (uint32_t i = 0; i < n && fullMesh; i++) {
    for (uint32_t j = 0; j < n; j++) {
      if (i != j && !xgmi[i][j]) {
        fullMesh = false;
        break;
      }
    }
  }
  // link "closeness" for grouping: higher bandwidth, then LOWER weight
  // (rsmi link weight is a distance-like cost), then positional adjacency
  auto closeness = [&](uint32_t i, uint32_t j) -> double {
    double c = (double)bwMax[i][j] * 1e6;
    if (weight[i][j] > 0) c += 1e5 / (double)weight[i][j];
    return c;
  };
  bool symmetric = true;
  for (uint32_t i = 0; i < n && symmetric; i++) {
    for (uint32_t j = 0; j < n && symmetric; j++) {
      if (i == j || !xgmi[i][j]) continue;
      for (uint32_t k = 0; k < n; k++) {
        if (k == i || k == j || !xgmi[i][k]) continue;
        if (closeness(i, j) != closeness(i, k)) {
          symmetric = false;
          break;
        }
      }
    }
  }

  // Grouping: measured (greedy strongest-peer matching) unless the mesh is
  // fully symmetric, where positional order is canonical.
  std::vector<uint32_t> order(n);
  for (uint32_t i = 0; i < n; i++) order[i] = i;
  if (!symmetric && n >= 4 && n % 4 == 0) {
    std::vector<bool> used(n, false);
    std::vector<std::pair<uint32_t, uint32_t>> pairs;
    for (uint32_t i = 0; i < n; i++) {
      if (used[i]) continue;
      used[i] = true;
      int best = -1;
      for (uint32_t j = 0; j < n; j++) {
        if (used[j] || !xgmi[i][j]) continue;
        if (best < 0 || closeness(i, j) > closeness(i, (uint32_t)best)) best = (int)j;
      }
      if (best < 0) {  // disconnected: fall back to positional
        for (uint32_t j = 0; j < n; j++) {
          if (!used[j]) {
            best = (int)j;
            break;
          }
        }
      }
      used[best] = true;
      pairs.emplace_back(i, (uint32_t)best);
    }
    // merge pairs into quads by strongest inter-pair closeness
    std::vector<bool> pUsed(pairs.size(), false);
    std::vector<uint32_t> grouped;
    for (size_t a = 0; a < pairs.size(); a++) {
      if (pUsed[a]) continue;
      pUsed[a] = true;
      int best = -1;
      double bestC = -1;
      for (size_t b = 0; b < pairs.size(); b++) {
        if (pUsed[b]) continue;
        double c = closeness(pairs[a].first, pairs[b].first) +
                   closeness(pairs[a].first, pairs[b].second) +
                   closeness(pairs[a].second, pairs[b].first) +
                   closeness(pairs[a].second, pairs[b].second);
        if (c > bestC) {
          bestC = c;
          best = (int)b;
        }
      }
      grouped.push_back(pairs[a].first);
      grouped.push_back(pairs[a].second);
      if (best >= 0) {
        pUsed[best] = true;
        grouped.push_back(pairs[best].first);
        grouped.push_back(pairs[best].second);
      }
    }
    if (grouped.size() == n) order = grouped;
  }

  printf("# rocm-topo-discover: node %s, %u GPUs, xGMI links %u, fullMesh=%s, symmetric=%s\n",
         hostname, n, xgmiLinks, fullMesh ? "true" : "false", symmetric ? "true" : "false");
  printf("nodeName: %s\n", hostname);
  printf("numGpus: %u\n", n);
  printf("gpus:\n");
  for (uint32_t i = 0; i < n; i++) {
    printf("- index: %u\n", i);
    printf("%svramBytes: %llu\n", kIndent, (unsigned long long)vram[i]);
    printf("%sxgmiPeers: [", kIndent);
    bool first = true;
    for (uint32_t j = 0; j < n; j++) {
      if (i != j && xgmi[i][j]) {
        printf("%s%u", first ? "" : ", ", j);
        first = false;
      }
    }
    printf("]\n");
  }

  // physicalCells fragment for the scheduler config. Leaf cells carry the
  // measured hbmBytes; the node cell carries the measured per-link xGMI
  // table (gbps from the SMI max-bandwidth query; healthy = link present).
  printf("physicalCells:\n");
  auto printLinks = [&](const char* pad) {
    printf("%sxgmiLinks:\n", pad);
    for (uint32_t i = 0; i < n; i++) {
      for (uint32_t j = i + 1; j < n; j++) {
        if (!xgmi[i][j] && !xgmi[j][i]) continue;
        double gbps = (double)std::max(bwMax[i][j], bwMax[j][i]) / 1e3;  // MB/s -> GB/s
        bool healthy = xgmi[i][j] && xgmi[j][i];
        printf("%s- {a: %u, b: %u, gbps: %.1f, healthy: %s}\n", pad, i, j, gbps,
               healthy ? "true" : "false");
      }
    }
  };
  if (n == 8) {
    printf("- cellType: MI355X-NODE\n");
    printf("%scellAddress: %s\n", kIndent, hostname);
    printLinks(kIndent);
    printf("%scellChildren:\n", kIndent);
    for (int q = 0; q < 2; q++) {
      printf("%s- cellChildren:  # MI355X-QUAD\n", kIndent);
      for (int p = 0; p < 2; p++) {
        printf("%s%s- cellChildren:  # MI355X-PAIR\n", kIndent, kIndent);
        for (int g = 0; g < 2; g++) {
          uint32_t idx = order[q * 4 + p * 2 + g];
          printf("%s%s%s- cellAddress: %u\n", kIndent, kIndent, kIndent, idx);
          printf("%s%s%s%shbmBytes: %llu\n", kIndent, kIndent, kIndent, kIndent,
                 (unsigned long long)vram[idx]);
        }
      }
    }
  } else {
    // generic fallback: flat node with n leaves (cellTypes must define
    // MI355X-NODE-<n> with childCellNumber n)
    printf("- cellType: MI355X-NODE-%u\n", n);
    printf("%scellAddress: %s\n", kIndent, hostname);
    printLinks(kIndent);
    printf("%scellChildren:\n", kIndent);
    for (uint32_t i = 0; i < n; i++) {
      printf("%s- cellAddress: %u\n", kIndent, order[i]);
      printf("%s%shbmBytes: %llu\n", kIndent, kIndent, (unsigned long long)vram[order[i]]);
    }
  }
  rsmi_shut_down();
  return 0;
}
