// Standalone scheduling-core benchmark/profiling driver (no Python).
// Builds an N-node MI355X cluster spec, replays a mixed 1/2/4-GPU guaranteed
// workload with churn, and reports per-decision latency percentiles.
// Compile:
//   g++ -O2 [-pg] -std=c++17 -o schedbench core/sched_bench_main.cpp \
//       core/{cells,topo_sched,build,alloc,algorithm,debug}.cpp
// Used to chase (and hold) the round-2 target: 128-node filter p50 < 20 us;
// results in profiles/sched_scaling_r02.md.
#include <algorithm>
#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <random>
#include <vector>

#include "core.hpp"

using namespace hived;

static ClusterSpec makeSpec(int nodes) {
  ClusterSpec spec;
  spec.cellTypes["MI355X"] = {"", 0, false};
  spec.cellTypes["MI355X-PAIR"] = {"MI355X", 2, false};
  spec.cellTypes["MI355X-QUAD"] = {"MI355X-PAIR", 2, false};
  CellTypeSpec node;
  node.child = "MI355X-QUAD";
  node.childCount = 2;
  node.isNode = true;
  spec.cellTypes["MI355X-NODE"] = node;
  for (int i = 0; i < nodes; i++) {
    PhysCellSpec n;
    n.type = "MI355X-NODE";
    n.address = "node" + std::to_string(i + 1);
    int g = 0;
    for (int q = 0; q < 2; q++) {
      PhysCellSpec quad;
      for (int p = 0; p < 2; p++) {
        PhysCellSpec pair;
        for (int l = 0; l < 2; l++) {
          PhysCellSpec leaf;
          leaf.address = std::to_string(g++);
          pair.children.push_back(leaf);
        }
        quad.children.push_back(pair);
      }
      n.children.push_back(quad);
    }
    spec.physicalCells.push_back(n);
  }
  VCSpec vc1, vc2;
  VirtCellSpec v;
  v.typePath = "MI355X-NODE";
  v.number = nodes / 2;
  vc1.virtualCells.push_back(v);
  v.number = nodes - nodes / 2;
  vc2.virtualCells.push_back(v);
  spec.virtualClusters["VC1"] = vc1;
  spec.virtualClusters["VC2"] = vc2;
  return spec;
}

int main(int argc, char** argv) {
  int nodes = argc > 1 ? atoi(argv[1]) : 128;
  int nreq = argc > 2 ? atoi(argv[2]) : 2000;
  int badLinks = argc > 3 ? atoi(argv[3]) : 0;  // degraded links to scatter
  auto tb0 = std::chrono::steady_clock::now();
  HivedCore core(makeSpec(nodes));
  for (int i = 0; i < nodes; i++) core.setNodeHealthy("node" + std::to_string(i + 1), true);
  {
    std::mt19937 lrng(7);
    for (int i = 0; i < badLinks; i++) {
      int n = (int)(lrng() % nodes) + 1;
      int a = (int)(lrng() % 8), b = (int)(lrng() % 8);
      if (a == b) b = (a + 1) % 8;
      core.setXgmiLinkHealthy("node" + std::to_string(n), a, b, false, 12.0);
    }
  }
  auto tb1 = std::chrono::steady_clock::now();

  std::mt19937 rng(0);
  std::set<std::string> suggested;  // ignoreSuggested=true path
  std::vector<double> lat;
  lat.reserve(nreq);
  struct LivePod {
    PodSpec spec;
    BindInfo info;
    std::string key;
  };
  std::vector<LivePod> live;
  int cells[] = {1, 2, 4};
  for (int i = 0; i < nreq; i++) {
    PodSpec s;
    s.vc = (rng() % 2) ? "VC1" : "VC2";
    s.priority = (int)(rng() % 2);
    s.leafCellNumber = cells[rng() % 3];
    s.groupName = "g" + std::to_string(i);
    s.groupPodNums[s.leafCellNumber] = 1;
    std::string key = "ns/p" + std::to_string(i);
    auto t0 = std::chrono::steady_clock::now();
    ScheduleResult r = core.schedule(s, key, suggested, Phase::Filtering);
    if (r.kind == ScheduleResult::Kind::Bind) {
      core.addAllocatedPod(s, r.bindInfo, key);
    }
    auto t1 = std::chrono::steady_clock::now();
    lat.push_back(std::chrono::duration<double, std::micro>(t1 - t0).count());
    if (r.kind == ScheduleResult::Kind::Bind) {
      live.push_back({s, r.bindInfo, key});
    }
    if ((int)live.size() > nodes * 4) {
      core.deleteAllocatedPod(live.front().spec, live.front().info, live.front().key);
      live.erase(live.begin());
    }
  }
  std::sort(lat.begin(), lat.end());
  printf("nodes=%d reqs=%d badLinks=%d build=%.1fms p50=%.1fus p90=%.1fus p99=%.1fus\n", nodes, nreq,
         badLinks,
         std::chrono::duration<double, std::milli>(tb1 - tb0).count(), lat[lat.size() / 2],
         lat[(size_t)(lat.size() * 0.90)], lat[(size_t)(lat.size() * 0.99)]);
  return 0;
}
