// Topology-aware placement engine.
//
// Semantics parity: pkg/algorithm/topology_aware_scheduler.go (node sorting,
// two-pass preemption, gang fit). The intra-node search is redesigned: instead
// of backtracking over leaf-cell combinations with LCA pruning (reference
// l.309-387), a best-fit recursive descent over the cell tree directly yields
// an LCA-minimal, buddy-packed placement in O(depth * fanout) per pod — on the
// fixed MI355X chain (leaf->pair->quad->node) this is a handful of comparisons,
// which is what makes microsecond-scale Schedule() latency possible.
#include <climits>

#include "core.hpp"

namespace hived {

TopoScheduler::TopoScheduler(const ChainCellList& ccl, std::map<int, int> levelLeafNum,
                             bool crossPriorityPack)
    : levelLeafNum_(std::move(levelLeafNum)), crossPriorityPack_(crossPriorityPack) {
  // Collect the "cluster view": node-level cells, or cells without a node-level
  // ancestor (e.g. a VC that owns a below-node cell such as a quad).
  int top = ccl.top();
  int nodeLevel = top;
  for (int l = 1; l <= top; l++) {
    if (!ccl.at(l).empty() && ccl.at(l)[0]->atOrAboveNode) {
      nodeLevel = l;
      break;
    }
  }
  std::unordered_set<Cell*> seen;
  for (int l = std::min(nodeLevel, top); l >= 1; l--) {
    for (Cell* c : ccl.at(l)) {
      Cell* v = ancestorNoHigherThanNode(c);
      if (seen.insert(v).second) viewCells_.push_back(v);
    }
  }
}

namespace {

struct PickSession {
  std::unordered_set<Cell*> taken;
};

// Leaf-granular health: a bad physical leaf (sick GPU / degraded xGMI link
// endpoint) is never available; a virtual leaf bound to a bad physical leaf
// (doomed-bad binding) is equally unavailable. This is finer than the
// reference, whose health model stops at node granularity.
bool leafHealthy(Cell* c) {
  if (c->isPhysical()) return c->healthy;
  PhysicalCell* p = static_cast<VirtualCell*>(c)->phys;
  return p == nullptr || p->healthy;
}

// (availableTotal, availableFree): healthy leaves that are free or
// lower-priority (preemptible), excluding ones already taken in this session.
std::pair<int, int> availLeaves(Cell* c, int p, PickSession& s) {
  if (c->level == kLowestLevel) {
    if (s.taken.count(c) || !leafHealthy(c)) return {0, 0};
    if (c->priority == kFreePriority) return {1, 1};
    if (c->priority < p) return {1, 0};
    return {0, 0};
  }
  int at = 0, af = 0;
  for (Cell* child : c->children) {
    auto [t, f] = availLeaves(child, p, s);
    at += t;
    af += f;
  }
  return {at, af};
}

// Lowest LCA level achievable for q leaves anywhere inside `cell`
// (INT_MAX if the subtree cannot hold q). This is the lookahead that makes
// the descent LCA-minimal: a child with MORE availability may reach a LOWER
// LCA (e.g. free = {0,1,2} in quad A vs {5,6} straddling quad B's pairs —
// A serves a pair request at pair level, B only at quad level). Verified
// against brute force in tests/test_placement_optimality.py.
int bestLCALevel(Cell* c, int q, int p, PickSession& s) {
  auto [t, f] = availLeaves(c, p, s);
  (void)f;
  if (t < q) return INT_MAX;
  int best = c->level;
  for (Cell* child : c->children) {
    int b = bestLCALevel(child, q, p, s);
    if (b < best) best = b;
  }
  return best;
}

// Pick q available leaf cells inside `cell`, minimizing the LCA level
// (affinity) and, secondarily, the number of preemptions, with buddy-style
// tight packing. Caller guarantees avail(cell) >= q.
void pickLeaves(Cell* cell, int q, int p, PickSession& s, std::vector<Cell*>& out) {
  if (cell->level == kLowestLevel) {
    s.taken.insert(cell);
    out.push_back(cell);
    return;
  }
  int n = static_cast<int>(cell->children.size());
  std::vector<std::pair<int, int>> av(n);
  for (int i = 0; i < n; i++) av[i] = availLeaves(cell->children[i], p, s);

  // A single child can hold the whole request: descend into the one that
  // reaches the lowest LCA, then needs the fewest preemptions, then is the
  // tightest fit (packing).
  int best = -1, bestLca = INT_MAX;
  for (int i = 0; i < n; i++) {
    if (av[i].first >= q) {
      int lca = bestLCALevel(cell->children[i], q, p, s);
      if (best < 0) {
        best = i;
        bestLca = lca;
        continue;
      }
      int needPreempt = std::max(0, q - av[i].second);
      int bestPreempt = std::max(0, q - av[best].second);
      if (lca < bestLca ||
          (lca == bestLca && (needPreempt < bestPreempt ||
                              (needPreempt == bestPreempt && av[i].first < av[best].first)))) {
        best = i;
        bestLca = lca;
      }
    }
  }
  if (best >= 0) {
    pickLeaves(cell->children[best], q, p, s, out);
    return;
  }
  // This cell is the LCA: drain children largest-available first so the fewest
  // subtrees are touched (keeps fragmentation low for later requests).
  std::vector<int> order(n);
  for (int i = 0; i < n; i++) order[i] = i;
  std::stable_sort(order.begin(), order.end(),
                   [&](int a, int b) { return av[a].first > av[b].first; });
  int remaining = q;
  for (int idx : order) {
    if (remaining == 0) break;
    int t = std::min(remaining, av[idx].first);
    if (t > 0) {
      pickLeaves(cell->children[idx], t, p, s, out);
      remaining -= t;
    }
  }
  if (remaining != 0) throw HivedError::Internal("pickLeaves underflow in " + cell->address);
}

std::tuple<bool, bool> healthyAndSuggested(Cell* c, const std::set<std::string>& suggestedNodes,
                                           bool ignoreSuggestedNodes) {
  PhysicalCell* pc = nullptr;
  if (c->isPhysical()) {
    pc = static_cast<PhysicalCell*>(c);
  } else {
    pc = static_cast<VirtualCell*>(c)->phys;
  }
  if (pc == nullptr) return {true, true};
  bool suggested =
      ignoreSuggestedNodes || (!pc->nodes.empty() && suggestedNodes.count(pc->nodes[0]) > 0);
  return {pc->healthy, suggested};
}

}  // namespace

bool TopoScheduler::tryScheduleAtPriority(const std::vector<int>& sortedLeafNums, int priority,
                                          const std::set<std::string>& suggestedNodes,
                                          bool ignoreSuggestedNodes, Placement<Cell>* out,
                                          std::string* failedReason) const {
  // Build and sort the cluster view: healthy > suggested > same-priority used
  // (desc, packing) > higher-priority used (asc, stay away).
  std::vector<NodeView> cv;
  cv.reserve(viewCells_.size());
  PickSession probe;  // empty: availability before any placement
  for (Cell* c : viewCells_) {
    NodeView n;
    n.c = c;
    n.usedSamePriority = c->usedAt(priority);
    n.usedHigherPriority = 0;
    for (auto& [p, num] : c->usedLeafAtPriority) {
      if (crossPriorityPack_) {
        if (p != priority) n.usedSamePriority += num;
      } else if (p > priority) {
        n.usedHigherPriority += num;
      }
    }
    // health-aware availability: bad leaves never count, so partially-bad
    // nodes stay usable for their healthy pairs/quads
    n.freeAtPriority = availLeaves(c, priority, probe).first;
    auto [healthy, suggested] = healthyAndSuggested(c, suggestedNodes, ignoreSuggestedNodes);
    n.healthy = healthy;
    n.suggested = suggested;
    cv.push_back(n);
  }
  std::stable_sort(cv.begin(), cv.end(), [](const NodeView& a, const NodeView& b) {
    if (a.healthy != b.healthy) return a.healthy;
    if (a.suggested != b.suggested) return a.suggested;
    if (a.usedSamePriority != b.usedSamePriority) return a.usedSamePriority > b.usedSamePriority;
    return a.usedHigherPriority < b.usedHigherPriority;
  });

  // Single-node preference: when ONE node can host the whole gang, use it
  // (most-packed first via the sort above). The reference's greedy fit
  // splits a gang as soon as the first packed node runs out (topology_
  // aware_scheduler.go:268-306) — acceptable on NVSwitch, but on MI355X a
  // split gang communicates over the NETWORK while a same-node gang rides
  // 7x ~153 GB/s xGMI links, so locality outranks strict packing here.
  // (Property-tested: tests/test_placement_optimality.py.)
  std::vector<int> pickedNodeIndices(sortedLeafNums.size(), -1);
  size_t podIndex = 0;
  int pickedLeafCellNum = 0;
  int totalLeafNum = 0;
  for (int q : sortedLeafNums) totalLeafNum += q;
  for (size_t nodeIndex = 0; nodeIndex < cv.size(); nodeIndex++) {
    const NodeView& n = cv[nodeIndex];
    if (n.freeAtPriority >= totalLeafNum) {
      // A non-suggested candidate only disqualifies itself from the
      // preference, not the request: keep scanning for a suggested
      // whole-gang host, and if none exists fall through to the greedy
      // split loop (which alone carries the reference's hard "must use a
      // non-suggested node" failure, findNodesForPods semantics).
      if (!n.suggested) continue;
      for (size_t i = 0; i < sortedLeafNums.size(); i++) {
        pickedNodeIndices[i] = static_cast<int>(nodeIndex);
      }
      podIndex = sortedLeafNums.size();
      break;
    }
  }
  for (size_t nodeIndex = 0; nodeIndex < cv.size() && podIndex < sortedLeafNums.size();) {
    const NodeView& n = cv[nodeIndex];
    if (n.freeAtPriority - pickedLeafCellNum >= sortedLeafNums[podIndex]) {
      // note: a partially-bad node is usable (its avail excludes bad leaves);
      // a fully-bad node has avail 0 and is never reached
      if (!n.suggested) {
        *failedReason = "have to use at least one non-suggested node " + n.c->address;
        return false;
      }
      pickedNodeIndices[podIndex] = static_cast<int>(nodeIndex);
      pickedLeafCellNum += sortedLeafNums[podIndex];
      podIndex++;
    } else {
      pickedLeafCellNum = 0;
      nodeIndex++;
    }
  }
  if (podIndex < sortedLeafNums.size()) {
    *failedReason = "insufficient capacity";
    return false;
  }

  // Pick leaf cells inside each pod's node.
  PickSession session;
  out->clear();
  for (size_t i = 0; i < sortedLeafNums.size(); i++) {
    int q = sortedLeafNums[i];
    Cell* node = cv[pickedNodeIndices[i]].c;
    std::vector<Cell*> leaves;
    leaves.reserve(q);
    pickLeaves(node, q, priority, session, leaves);
    (*out)[q].push_back(std::move(leaves));
  }
  return true;
}

bool TopoScheduler::Schedule(const std::map<int, int>& podLeafCellNums, int priority,
                             const std::set<std::string>& suggestedNodes,
                             bool ignoreSuggestedNodes, Placement<Cell>* out,
                             std::string* failedReason) const {
  std::vector<int> sortedLeafNums;
  for (auto& [leafNum, podNum] : podLeafCellNums) {
    for (int i = 0; i < podNum; i++) sortedLeafNums.push_back(leafNum);
  }
  // First try without preemption (only free cells), then allow preempting
  // lower priorities.
  if (tryScheduleAtPriority(sortedLeafNums, kOpportunisticPriority, suggestedNodes,
                            ignoreSuggestedNodes, out, failedReason)) {
    return true;
  }
  if (priority > kOpportunisticPriority) {
    return tryScheduleAtPriority(sortedLeafNums, priority, suggestedNodes, ignoreSuggestedNodes,
                                 out, failedReason);
  }
  return false;
}

bool IntraVCScheduler::schedule(const SchedulingRequest& sr, Placement<VirtualCell>* out,
                                std::string* failedReason) const {
  const TopoScheduler* scheduler = nullptr;
  if (!sr.pinnedCellId.empty()) {
    auto it = pinnedSchedulers.find(sr.pinnedCellId);
    if (it != pinnedSchedulers.end()) scheduler = &it->second;
  } else {
    auto it = nonPinnedSchedulers.find(sr.chain);
    if (it != nonPinnedSchedulers.end()) scheduler = &it->second;
  }
  Placement<Cell> generic;
  if (scheduler == nullptr ||
      !scheduler->Schedule(sr.podLeafCellNums, sr.priority, *sr.suggestedNodes,
                           sr.ignoreSuggestedNodes, &generic, failedReason)) {
    if (failedReason->empty()) *failedReason = "no scheduler for request";
    *failedReason += " when scheduling in VC " + sr.vc;
    return false;
  }
  out->clear();
  for (auto& [leafNum, pods] : generic) {
    for (auto& pod : pods) {
      std::vector<VirtualCell*> cells;
      cells.reserve(pod.size());
      for (Cell* c : pod) cells.push_back(static_cast<VirtualCell*>(c));
      (*out)[leafNum].push_back(std::move(cells));
    }
  }
  return true;
}

}  // namespace hived
