// HivedCore: the scheduling algorithm facade — Schedule dispatch, affinity-
// group lifecycle (allocated / preempting / lazy-preempted), optimistic-commit
// replay (addAllocatedPod doubles as crash recovery and reconfiguration).
// Semantics parity: pkg/algorithm/hived_algorithm.go:180-1352 and
// pkg/algorithm/utils.go:38-310; state machines per doc/design/state-machine.md.
#include <cstdio>
#include <functional>
#include <cstdlib>

#include "core.hpp"

namespace hived {

bool mapDebugRelease() {
  static int v = -1;
  if (v < 0) v = getenv("HIVED_DEBUG_MAP") ? 1 : 0;
  return v == 1;
}

namespace {

// node -> victim pod keys; overlapping Preempting groups touching the placement
struct Victims {
  std::map<std::string, std::set<std::string>> byNode;
  std::set<Group*> overlappingPreemptors;
};

Victims collectPreemptionVictims(const Placement<PhysicalCell>& placement) {
  Victims v;
  for (auto& [leafNum, pods] : placement) {
    (void)leafNum;
    for (auto& pod : pods) {
      for (PhysicalCell* c : pod) {
        if (c == nullptr) continue;
        if ((c->state == CState::Used || c->state == CState::Reserving) &&
            c->usingGroup != nullptr) {
          // gang semantics: preempting any cell of a group victimizes the
          // whole group
          for (auto& [ln, victims] : c->usingGroup->allocatedPods) {
            (void)ln;
            for (auto& p : victims) {
              if (p.present) v.byNode[p.node].insert(p.key);
            }
          }
        }
        if (c->state == CState::Reserving || c->state == CState::Reserved) {
          v.overlappingPreemptors.insert(c->reservingGroup);
        }
      }
    }
  }
  return v;
}

std::set<std::string> collectBadOrNonSuggestedNodes(const Placement<PhysicalCell>& placement,
                                                    const std::set<std::string>& suggestedNodes,
                                                    bool ignoreSuggestedNodes) {
  std::set<std::string> bad;
  for (auto& [leafNum, pods] : placement) {
    (void)leafNum;
    for (auto& pod : pods) {
      for (PhysicalCell* c : pod) {
        if (c == nullptr) continue;
        if (!c->healthy || (!ignoreSuggestedNodes && !suggestedNodes.count(c->nodes[0]))) {
          bad.insert(c->nodes[0]);
        }
      }
    }
  }
  return bad;
}

std::pair<PodPlacementInfo, std::string> retrieveMissingPodPlacement(Group* g, int leafCellNum,
                                                                     int podIndex) {
  for (auto& [ln, pods] : g->allocatedPods) {
    (void)ln;
    for (auto& p : pods) {
      if (!p.present) continue;
      for (auto& mbi : p.bindInfo.memberBindInfo) {
        if (!mbi.empty() && static_cast<int>(mbi[0].leafIndices.size()) == leafCellNum &&
            podIndex < static_cast<int>(mbi.size())) {
          return {mbi[podIndex], p.bindInfo.chain};
        }
      }
    }
  }
  throw HivedError::Internal("no allocated pod found in group " + g->name +
                             " when retrieving missing placement");
}

VirtualCell* retrieveVirtualCell(const Placement<PhysicalCell>& phys,
                                 const Placement<VirtualCell>& virt, PhysicalCell* target) {
  for (auto& [leafNum, pods] : phys) {
    auto vIt = virt.find(leafNum);
    if (vIt == virt.end()) continue;
    for (size_t podIdx = 0; podIdx < pods.size(); podIdx++) {
      for (size_t i = 0; i < pods[podIdx].size(); i++) {
        if (pods[podIdx][i] == target) return vIt->second[podIdx][i];
      }
    }
  }
  return nullptr;
}

int getNewPodIndex(const std::vector<AllocatedPod>& pods) {
  for (size_t i = 0; i < pods.size(); i++) {
    if (!pods[i].present) return static_cast<int>(i);
  }
  return -1;
}

int getAllocatedPodIndex(const BindInfo& info, int leafCellNum) {
  for (auto& mbi : info.memberBindInfo) {
    if (mbi.empty() || static_cast<int>(mbi[0].leafIndices.size()) != leafCellNum) continue;
    for (size_t podIndex = 0; podIndex < mbi.size(); podIndex++) {
      auto& pl = mbi[podIndex];
      if (pl.node == info.node && !info.isolation.empty() &&
          std::find(pl.leafIndices.begin(), pl.leafIndices.end(), info.isolation[0]) !=
              pl.leafIndices.end()) {
        return static_cast<int>(podIndex);
      }
    }
  }
  return -1;
}

bool allPodsReleased(const std::map<int, std::vector<AllocatedPod>>& allocatedPods) {
  for (auto& [ln, pods] : allocatedPods) {
    (void)ln;
    for (auto& p : pods) {
      if (p.present) return false;
    }
  }
  return true;
}

std::unique_ptr<Group> newGroup(const PodSpec& s, GState state) {
  auto g = std::make_unique<Group>();
  g->name = s.groupName;
  g->vc = s.vc;
  g->lazyPreemptionEnable = s.lazyPreemptionEnable;
  g->ignoreK8sSuggestedNodes = s.ignoreK8sSuggestedNodes;
  g->gangReleaseEnable = s.gangReleaseEnable;
  g->priority = s.priority;
  g->state = state;
  g->totalPodNums = s.groupPodNums;
  for (auto& [leafNum, podNum] : s.groupPodNums) {
    g->allocatedPods[leafNum].resize(podNum);
    g->physPlacement[leafNum].assign(podNum, std::vector<PhysicalCell*>(leafNum, nullptr));
    g->virtPlacement[leafNum].assign(podNum, std::vector<VirtualCell*>(leafNum, nullptr));
  }
  return g;
}

}  // namespace

// ---------------------------------------------------------------------------
// Schedule
// ---------------------------------------------------------------------------

ScheduleResult HivedCore::schedule(const PodSpec& s, const std::string& podKey,
                                   const std::set<std::string>& suggestedNodes, Phase phase) {
  OpGuard opGuard(this);
  scheduleCount_++;
  Placement<PhysicalCell> phys;
  Placement<VirtualCell> virt;
  bool hasVirtual = true;
  std::map<std::string, std::set<std::string>> victims;
  std::string waitReason;
  int podIndex = 0;
  bool havePlacement = false;

  auto it = groups_.find(s.groupName);
  if (it != groups_.end()) {
    havePlacement = schedulePodFromExistingGroup(it->second.get(), s, suggestedNodes, phase,
                                                 podKey, &phys, &hasVirtual, &virt, &victims,
                                                 &podIndex);
  }
  // the group may have been a preempting group deleted just above
  if (groups_.find(s.groupName) == groups_.end()) {
    schedulePodFromNewGroup(s, suggestedNodes, phase, podKey, &phys, &hasVirtual, &virt, &victims,
                            &waitReason);
    havePlacement = !phys.empty();
  }
  (void)havePlacement;
  Group* g = nullptr;
  if (auto git = groups_.find(s.groupName); git != groups_.end()) g = git->second.get();
  return generateResult(phys, hasVirtual, virt, victims, waitReason, s.leafCellNumber, podIndex, g,
                        s.groupName);
}

ScheduleResult HivedCore::generateResult(const Placement<PhysicalCell>& phys, bool hasVirtual,
                                         const Placement<VirtualCell>& virt,
                                         const std::map<std::string, std::set<std::string>>& victims,
                                         const std::string& waitReason, int currentLeafNum,
                                         int podIndex, Group* group,
                                         const std::string& groupName) {
  ScheduleResult r;
  if (phys.empty()) {
    r.kind = ScheduleResult::Kind::Wait;
    r.waitReason = waitReason.empty() ? "waiting for resources" : waitReason;
    return r;
  }
  if (!victims.empty()) {
    // K8s preempts one node per round: report victims of ONE node, chosen
    // uniformly at random (seeded) so victim churn spreads across nodes
    // under contention instead of herding on the first map key (reference
    // utils.go:82-103 randomizes for the same reason).
    r.kind = ScheduleResult::Kind::Preempt;
    auto it = victims.begin();
    std::advance(it, static_cast<long>(victimRng_() % victims.size()));
    r.victimNode = it->first;
    r.victimPodKeys.assign(it->second.begin(), it->second.end());
    return r;
  }
  r.kind = ScheduleResult::Kind::Bind;
  BindInfo& info = r.bindInfo;
  for (auto& [leafNum, pods] : phys) {
    std::vector<PodPlacementInfo> placements(pods.size());
    for (size_t pi = 0; pi < pods.size(); pi++) {
      auto& pl = placements[pi];
      pl.leafIndices.assign(leafNum, 0);
      pl.preassignedTypes.assign(leafNum, "");
      for (int li = 0; li < leafNum; li++) {
        PhysicalCell* c = pods[pi][li];
        if (c == nullptr) {
          if (group == nullptr || group->state == GState::Preempting) {
            throw HivedError::Internal("first pod in group " + groupName +
                                       " was allocated invalid resource");
          }
          // placement lost (e.g. reconfiguration): insist the original
          // decision by retrieving it from a sibling pod's bind info
          auto [retrieved, chain] = retrieveMissingPodPlacement(group, leafNum, static_cast<int>(pi));
          pl = retrieved;
          if (info.chain.empty()) info.chain = chain;
          break;
        }
        if (pl.node.empty()) pl.node = c->nodes[0];
        pl.leafIndices[li] = c->leafIndices[0];
        if (hasVirtual) {
          auto vIt = virt.find(leafNum);
          if (vIt != virt.end() && pi < vIt->second.size() &&
              li < static_cast<int>(vIt->second[pi].size()) && vIt->second[pi][li] != nullptr) {
            VirtualCell* v = vIt->second[pi][li];
            pl.preassignedTypes[li] = cellTypes_[v->chain][v->preassigned->level];
          }
        }
      }
    }
    if (leafNum == currentLeafNum) {
      if (podIndex >= static_cast<int>(placements.size())) {
        throw HivedError::BadRequest("pod index " + std::to_string(podIndex) +
                                     " out of range in group " + groupName);
      }
      info.node = placements[podIndex].node;
      info.isolation = placements[podIndex].leafIndices;
      PhysicalCell* first = pods[podIndex].empty() ? nullptr : pods[podIndex][0];
      if (first != nullptr) info.chain = first->chain;
    }
    info.memberBindInfo.push_back(std::move(placements));
  }
  return r;
}

// ---------------------------------------------------------------------------
// Existing / new group scheduling
// ---------------------------------------------------------------------------

bool HivedCore::schedulePodFromExistingGroup(
    Group* g, const PodSpec& s, const std::set<std::string>& suggestedNodes, Phase phase,
    const std::string& podKey, Placement<PhysicalCell>* phys, bool* hasVirtual,
    Placement<VirtualCell>* virt, std::map<std::string, std::set<std::string>>* victims,
    int* podIndex) {
  std::set<std::string> badNodes =
      collectBadOrNonSuggestedNodes(g->physPlacement, suggestedNodes, g->ignoreK8sSuggestedNodes);
  if (g->state == GState::Allocated) {
    *phys = g->physPlacement;
    *hasVirtual = g->hasVirtualPlacement;
    *virt = g->virtPlacement;
    auto it = g->allocatedPods.find(s.leafCellNumber);
    int idx = (it == g->allocatedPods.end()) ? -1 : getNewPodIndex(it->second);
    if (idx == -1) {
      throw HivedError::BadRequest("Requesting more pods than the configured number for " +
                                   std::to_string(s.leafCellNumber) + " leaf cells in group " +
                                   s.groupName);
    }
    *podIndex = idx;
    return true;
  }
  // groupPreempting
  if (phase == Phase::Preempting && !badNodes.empty()) {
    // placement no longer fully healthy / suggested: cancel and reschedule
    deletePreemptingGroup(g, podKey);
    return false;
  }
  *phys = g->physPlacement;
  *hasVirtual = g->hasVirtualPlacement;
  *virt = g->virtPlacement;
  Victims v = collectPreemptionVictims(g->physPlacement);
  *victims = v.byNode;
  g->preemptingPods.insert(podKey);
  return true;
}

void HivedCore::schedulePodFromNewGroup(const PodSpec& s,
                                        const std::set<std::string>& suggestedNodes, Phase phase,
                                        const std::string& podKey, Placement<PhysicalCell>* phys,
                                        bool* hasVirtual, Placement<VirtualCell>* virt,
                                        std::map<std::string, std::set<std::string>>* victims,
                                        std::string* waitReason) {
  if (!scheduleNewAffinityGroup(s, suggestedNodes, podKey, phys, hasVirtual, virt, waitReason)) {
    phys->clear();
    return;
  }
  Victims v = collectPreemptionVictims(*phys);
  *victims = v.byNode;
  if (phase == Phase::Preempting) {
    // cancel lower-priority preemptors whose reservations overlap ours
    for (Group* preemptor : v.overlappingPreemptors) {
      deletePreemptingGroup(preemptor, podKey);
    }
    if (!v.byNode.empty()) {
      // reserve cells immediately so equal-priority groups cannot contend
      createPreemptingGroup(s, *phys, *virt, podKey);
    }
  }
}

bool HivedCore::scheduleNewAffinityGroup(const PodSpec& s,
                                         const std::set<std::string>& suggestedNodes,
                                         const std::string& podKey, Placement<PhysicalCell>* phys,
                                         bool* hasVirtual, Placement<VirtualCell>* virt,
                                         std::string* failedReason) {
  SchedulingRequest sr;
  sr.vc = s.vc;
  sr.pinnedCellId = s.pinnedCellId;
  sr.priority = s.priority;
  sr.groupName = s.groupName;
  sr.podLeafCellNums = s.groupPodNums;
  sr.suggestedNodes = &suggestedNodes;
  sr.ignoreSuggestedNodes = s.ignoreK8sSuggestedNodes;
  sr.hbmBytes = s.hbmBytesPerCell;
  validateSchedulingRequest(sr, podKey);
  if (!sr.pinnedCellId.empty()) {
    sr.chain = pinnedPhysical_[s.vc][s.pinnedCellId]->chain;
    return handleSchedulingRequest(sr, phys, hasVirtual, virt, failedReason);
  }
  if (!s.leafCellType.empty()) {
    if (!cellChains_.count(s.leafCellType)) {
      throw HivedError::BadRequest("[" + podKey + "]: Pod requesting leaf cell type " +
                                   s.leafCellType + " which the whole cluster does not have");
    }
    return scheduleForLeafCellType(sr, s.leafCellType, podKey, true, phys, hasVirtual, virt,
                                   failedReason);
  }
  for (auto& [leafType, chains] : cellChains_) {
    (void)chains;
    if (scheduleForLeafCellType(sr, leafType, podKey, false, phys, hasVirtual, virt,
                                failedReason)) {
      return true;
    }
  }
  return false;
}

bool HivedCore::scheduleForLeafCellType(SchedulingRequest& sr, const std::string& leafCellType,
                                        const std::string& podKey, bool typeSpecified,
                                        Placement<PhysicalCell>* phys, bool* hasVirtual,
                                        Placement<VirtualCell>* virt, std::string* failedReason) {
  bool vcHasType = false;
  for (const std::string& chain : cellChains_[leafCellType]) {
    if (sr.priority < kMinGuaranteedPriority ||
        vcSchedulers_[sr.vc].nonPinnedFull.count(chain)) {
      vcHasType = true;
      sr.chain = chain;
      if (handleSchedulingRequest(sr, phys, hasVirtual, virt, failedReason)) return true;
    }
  }
  if (typeSpecified && sr.priority >= kMinGuaranteedPriority && !vcHasType) {
    throw HivedError::BadRequest("[" + podKey + "]: Pod requesting leaf cell type " + leafCellType +
                                 " which VC " + sr.vc + " does not have");
  }
  return false;
}

void HivedCore::validateSchedulingRequest(const SchedulingRequest& sr, const std::string& podKey) {
  std::string message;
  if (!vcSchedulers_.count(sr.vc)) {
    message = "VC " + sr.vc + " does not exist!";
  } else if (!sr.pinnedCellId.empty()) {
    if (!vcSchedulers_[sr.vc].pinned.count(sr.pinnedCellId)) {
      message = "VC " + sr.vc + " does not have pinned cell " + sr.pinnedCellId;
    } else if (sr.priority == kOpportunisticPriority) {
      message = "opportunistic pod not supported to use pinned cell " + sr.pinnedCellId;
    }
  }
  if (!message.empty()) throw HivedError::BadRequest("[" + podKey + "]: " + message);
}

bool HivedCore::handleSchedulingRequest(const SchedulingRequest& sr, Placement<PhysicalCell>* phys,
                                        bool* hasVirtual, Placement<VirtualCell>* virt,
                                        std::string* failedReason) {
  if (sr.priority >= kMinGuaranteedPriority) {
    *hasVirtual = true;
    return scheduleGuaranteedGroup(sr, phys, virt, failedReason);
  }
  *hasVirtual = false;
  virt->clear();
  return scheduleOpportunisticGroup(sr, phys, failedReason);
}

bool HivedCore::scheduleGuaranteedGroup(const SchedulingRequest& srIn, Placement<PhysicalCell>* phys,
                                        Placement<VirtualCell>* virt, std::string* failedReason) {
  SchedulingRequest sr = srIn;

  // One full placement attempt: intra-VC schedule -> lazy preemption ->
  // binding-vertex construction -> virtual->physical mapping. honorOnly
  // restricts both the descent and the mapping to link-clean choices
  // within sr.cleanWorld; reverts lazy preemption on failure.
  auto attemptOnce = [&](bool honorOnly) -> bool {
    sr.honorLinksOnly = honorOnly;
    if (!vcSchedulers_[sr.vc].schedule(sr, virt, failedReason)) return false;
    if (mapDebugRelease()) {
      for (auto& [ln, pods] : *virt) {
        for (auto& pod : pods) {
          fprintf(stderr, "[virt ln=%d honorOnly=%d]", ln, (int)honorOnly);
          for (auto* v : pod) fprintf(stderr, " %s", v->address.c_str());
          fprintf(stderr, "\n");
        }
      }
    }

    std::unordered_map<VirtualCell*, PhysicalCell*> bindings;
    auto lazyPreempted = tryLazyPreempt(*virt, sr.groupName);

    // build binding paths: trees of unbound virtual cells to bind
    std::vector<std::unique_ptr<BindingVertex>> roots;
    std::vector<BindingVertex*> preassigned;
    std::vector<std::vector<BindingVertex*>> nonPreassigned;
    std::unordered_map<VirtualCell*, BindingVertex*> vertexMap;
    for (auto& [leafNum, pods] : *virt) {
      (void)leafNum;
      for (auto& pod : pods) {
        for (VirtualCell* leaf : pod) {
          if (leaf->phys != nullptr) {
            bindings[leaf] = leaf->phys;
            continue;
          }
          std::vector<VirtualCell*> path;
          for (Cell* c = leaf; c != nullptr; c = c->parent) {
            auto* vc = static_cast<VirtualCell*>(c);
            if (vc->phys != nullptr || vertexMap.count(vc)) break;
            path.push_back(vc);
          }
          if (path.empty()) continue;
          VirtualCell* pathRoot = path.back();
          auto rootVertex = std::make_unique<BindingVertex>();
          rootVertex->cell = pathRoot;
          BindingVertex* rootPtr = rootVertex.get();
          vertexMap[pathRoot] = rootPtr;
          if (pathRoot->parent == nullptr) {
            preassigned.push_back(rootPtr);
            roots.push_back(std::move(rootVertex));
          } else if (static_cast<VirtualCell*>(pathRoot->parent)->phys != nullptr) {
            bool buddyExists = false;
            for (auto& grp : nonPreassigned) {
              if (grp[0]->cell->parent == pathRoot->parent) {
                grp.push_back(rootPtr);
                buddyExists = true;
                break;
              }
            }
            if (!buddyExists) nonPreassigned.push_back({rootPtr});
            roots.push_back(std::move(rootVertex));
          } else {
            BindingVertex* parentVertex = vertexMap.at(static_cast<VirtualCell*>(pathRoot->parent));
            parentVertex->children.push_back(std::move(rootVertex));
          }
          for (int i = static_cast<int>(path.size()) - 2; i >= 0; i--) {
            auto childVertex = std::make_unique<BindingVertex>();
            childVertex->cell = path[i];
            vertexMap[path[i]] = childVertex.get();
            vertexMap.at(static_cast<VirtualCell*>(path[i]->parent))
                ->children.push_back(std::move(childVertex));
          }
        }
      }
    }

    const std::unordered_map<VirtualCell*, PhysicalCell*> seedBindings = bindings;
    auto tryMap = [&](bool honorLinks) {
      bindings = seedBindings;  // drop partial picks from a failed attempt
      std::map<int, int> freeCellNumCopy = allVCFreeCellNum_[sr.chain];
      return mapVirtualPlacementToPhysical(preassigned, nonPreassigned,
                                           freeCellList_[sr.chain].shallowCopy(), freeCellNumCopy,
                                           *sr.suggestedNodes, sr.ignoreSuggestedNodes, bindings,
                                           sr.hbmBytes, honorLinks, sr.cleanWorld);
    };
    bool mapped;
    if (honorOnly) {
      mapped = tryMap(true);
    } else if (sr.cleanWorld != nullptr || chainHasBadLinks(sr.chain)) {
      mapped = tryMap(true) || tryMap(false);
    } else {
      mapped = tryMap(false);
    }
    if (mapped) {
      phys->clear();
      for (auto& [leafNum, pods] : *virt) {
        for (auto& pod : pods) {
          std::vector<PhysicalCell*> cells;
          cells.reserve(pod.size());
          for (VirtualCell* v : pod) cells.push_back(bindings.at(v));
          (*phys)[leafNum].push_back(std::move(cells));
        }
      }
      return true;
    }
    for (auto& [groupName, placement] : lazyPreempted) {
      revertLazyPreempt(groups_.at(groupName).get(), placement);
    }
    return false;
  };

  // When the chain carries degraded xGMI links and the gang needs >= 2
  // leaves, try every enumerated clean-shape world (different consistent
  // endpoint-avoidance choices admit different clean shapes) with
  // link-honoring-only rungs, then fall back to the unconstrained ladder
  // (capacity outranks link quality). A 1-leaf request never straddles a
  // link: no worlds (it may freely use — and by the parking preference,
  // should use — degraded endpoints).
  int gangLeaves = 0;
  for (auto& [ln, pn] : sr.podLeafCellNums) gangLeaves += ln * pn;
  if (chainHasBadLinks(sr.chain) && gangLeaves >= 2) {
    const std::set<std::string>* filter =
        sr.ignoreSuggestedNodes ? nullptr : sr.suggestedNodes;
    // Link-clean placements take precedence over dirty ones even when
    // cleanliness costs a preemption: a gang all-reducing over a degraded
    // link (~12 vs ~153 GB/s) is crippled for its whole lifetime, while
    // preemption (usually of opportunistic scavengers) is the scheduler's
    // normal business. Tier order: all-free clean worlds, then preemptive
    // clean worlds, then the unconstrained fallback ladder.
    auto tryWorlds = [&](int tier) {
      // cache worlds per (chain, tier) while no world-relevant mutation
      // happened; failed filters of waiting pods then skip the rebuild.
      // Requests with a suggested-node restriction bypass the cache (the
      // filter changes the world).
      std::vector<CleanShapeWorld> computed;
      const std::vector<CleanShapeWorld>* worldsPtr;
      if (filter == nullptr) {
        WorldCacheEntry& entry = worldCache_[{sr.chain, tier}];
        if (entry.epoch != gWorldEpochCounter) {
          entry.worlds = computeCleanShapeWorlds(fullCellList_[sr.chain], nullptr, 4, tier);
          entry.epoch = gWorldEpochCounter;
        }
        worldsPtr = &entry.worlds;
      } else {
        computed = computeCleanShapeWorlds(fullCellList_[sr.chain], filter, 4, tier);
        worldsPtr = &computed;
      }
      for (const auto& w : *worldsPtr) {
        sr.cleanWorld = &w;
        if (mapDebugRelease()) {
          fprintf(stderr, "[world] chain=%s tier=%d excl=%zu:", sr.chain.c_str(), tier,
                  w.excluded.size());
          for (auto& [l, v] : w.caps) fprintf(stderr, " L%d=%d", l, v);
          fprintf(stderr, "\n");
        }
        if (attemptOnce(true)) return true;
      }
      return false;
    };
    if (tryWorlds(kOpportunisticPriority)) return true;
    if (sr.priority > kOpportunisticPriority && tryWorlds(sr.priority)) return true;
    sr.cleanWorld = nullptr;
  }
  if (attemptOnce(false)) return true;
  *failedReason = std::string("Mapping the virtual placement would need to use at least one ") +
                  (sr.ignoreSuggestedNodes ? "bad" : "bad or non-suggested") + " node";
  return false;
}

bool HivedCore::chainHasBadLinks(const std::string& chain) {
  auto& ccl = fullCellList_[chain];
  for (Cell* c : ccl.at(ccl.top())) {
    if (static_cast<PhysicalCell*>(c)->badLinksUnder > 0) return true;
  }
  return false;
}

std::map<std::string, Placement<VirtualCell>> HivedCore::tryLazyPreempt(
    const Placement<VirtualCell>& p, const std::string& groupName) {
  std::map<std::string, Placement<VirtualCell>> preempted;
  for (auto& [leafNum, pods] : p) {
    (void)leafNum;
    for (auto& pod : pods) {
      for (VirtualCell* leaf : pod) {
        if (PhysicalCell* pc = leaf->phys) {
          if (pc->state == CState::Used && pc->usingGroup->lazyPreemptionEnable &&
              !preempted.count(pc->usingGroup->name)) {
            preempted[pc->usingGroup->name] = lazyPreemptGroup(pc->usingGroup, groupName);
          }
        }
      }
    }
  }
  return preempted;
}

bool HivedCore::scheduleOpportunisticGroup(const SchedulingRequest& sr,
                                           Placement<PhysicalCell>* phys,
                                           std::string* failedReason) {
  Placement<Cell> generic;
  if (!opportunisticSchedulers_.at(sr.chain).Schedule(sr.podLeafCellNums, kOpportunisticPriority,
                                                      *sr.suggestedNodes, sr.ignoreSuggestedNodes,
                                                      &generic, failedReason, sr.hbmBytes)) {
    *failedReason += " when scheduling in physical cluster";
    return false;
  }
  phys->clear();
  for (auto& [leafNum, pods] : generic) {
    for (auto& pod : pods) {
      std::vector<PhysicalCell*> cells;
      cells.reserve(pod.size());
      for (Cell* c : pod) cells.push_back(static_cast<PhysicalCell*>(c));
      (*phys)[leafNum].push_back(std::move(cells));
    }
  }
  return true;
}

// ---------------------------------------------------------------------------
// Pod lifecycle
// ---------------------------------------------------------------------------

void HivedCore::deleteUnallocatedPod(const PodSpec& s, const std::string& podKey) {
  OpGuard opGuard(this);
  auto it = groups_.find(s.groupName);
  if (it == groups_.end() || it->second->state != GState::Preempting) return;
  Group* g = it->second.get();
  g->preemptingPods.erase(podKey);
  if (g->preemptingPods.empty()) {
    deletePreemptingGroup(g, podKey);
  }
}

void HivedCore::addAllocatedPod(const PodSpec& s, const BindInfo& info, const std::string& podKey) {
  OpGuard opGuard(this);
  int podIndex = 0;
  auto it = groups_.find(s.groupName);
  if (it != groups_.end()) {
    Group* g = it->second.get();
    if (g->state == GState::Preempting) allocatePreemptingGroup(g, podKey);
    podIndex = getAllocatedPodIndex(info, s.leafCellNumber);
    if (podIndex == -1) return;  // placement not found in group; ignore
  } else {
    createAllocatedGroup(s, info, podKey);
    podIndex = getAllocatedPodIndex(info, s.leafCellNumber);
    if (podIndex == -1) podIndex = 0;
  }
  Group* g = groups_.at(s.groupName).get();
  auto& slots = g->allocatedPods[s.leafCellNumber];
  if (podIndex >= static_cast<int>(slots.size())) slots.resize(podIndex + 1);
  slots[podIndex] = AllocatedPod{true, podKey, info.node, info};
}

void HivedCore::deleteAllocatedPod(const PodSpec& s, const BindInfo& info,
                                   const std::string& podKey) {
  OpGuard opGuard(this);
  auto it = groups_.find(s.groupName);
  if (it == groups_.end()) return;
  Group* g = it->second.get();
  int podIndex = getAllocatedPodIndex(info, s.leafCellNumber);
  if (podIndex == -1) return;
  auto& slots = g->allocatedPods[s.leafCellNumber];
  if (podIndex < static_cast<int>(slots.size())) slots[podIndex] = AllocatedPod{};
  if (allPodsReleased(g->allocatedPods)) {
    deleteAllocatedGroup(g, podKey);
  }
}

// ---------------------------------------------------------------------------
// Group lifecycle
// ---------------------------------------------------------------------------

void HivedCore::createAllocatedGroup(const PodSpec& s, const BindInfo& info,
                                     const std::string& podKey) {
  auto group = newGroup(s, GState::Allocated);
  Group* g = group.get();
  groups_[s.groupName] = std::move(group);
  bool shouldLazyPreempt = false;
  for (auto& mbi : info.memberBindInfo) {
    if (mbi.empty()) continue;
    int leafCellNumber = static_cast<int>(mbi[0].leafIndices.size());
    if (!g->physPlacement.count(leafCellNumber)) {
      // bind info inconsistent with the group spec; tolerate by extending
      g->physPlacement[leafCellNumber].assign(mbi.size(),
                                              std::vector<PhysicalCell*>(leafCellNumber, nullptr));
      g->virtPlacement[leafCellNumber].assign(mbi.size(),
                                              std::vector<VirtualCell*>(leafCellNumber, nullptr));
      g->allocatedPods[leafCellNumber].resize(mbi.size());
    }
    for (size_t podIndex = 0; podIndex < mbi.size(); podIndex++) {
      for (size_t li = 0; li < mbi[podIndex].leafIndices.size(); li++) {
        bool lazyPreempt = false;
        bool isOpportunistic = false;
        auto [p, v] = findAllocatedLeafCell(static_cast<int>(li), mbi[podIndex], info.chain, s, g,
                                            podKey, &lazyPreempt, &isOpportunistic);
        if (p == nullptr) {
          // leaf cell address no longer in the spec: ignore this cell
          continue;
        }
        g->physPlacement[leafCellNumber][podIndex][li] = p;
        if (isOpportunistic) {
          g->hasVirtualPlacement = false;
        } else if (v != nullptr) {
          g->virtPlacement[leafCellNumber][podIndex][li] = v;
          if (inFreeCellList(p) && v->preassigned->priority > kFreePriority) {
            // binding this cell requires re-binding a preassigned cell that is
            // in use (reconfiguration shrank the VC): lazy preempt the users
            lazyPreemptCell(v->preassigned, g->name);
          }
        } else {
          shouldLazyPreempt = shouldLazyPreempt || lazyPreempt;
        }
        if (p->reservingGroup != nullptr && p->reservingGroup != g) {
          // cell event e8(i) (reference doc/design/state-machine.md): this
          // cell is Reserved by a lower-priority Preempting group, but an
          // Allocated group is now taking it (a higher-priority bind landed
          // on the vacated reservation) -> cancel that preemptor entirely,
          // releasing all its reservations; it will retry from Pending.
          // Found by fuzzing: without this, the preemptor stays Preempting
          // while its cells are Used by another group (state corruption).
          deletePreemptingGroup(p->reservingGroup, podKey);
        }
        auto [safetyOk, reason] =
            allocateLeafCell(p, g->virtPlacement[leafCellNumber][podIndex][li], s.priority, g->vc);
        p->usingGroup = g;
        setCellState(p, CState::Used);
        if (!safetyOk) {
          shouldLazyPreempt = true;
        }
        (void)reason;
      }
    }
  }
  if (shouldLazyPreempt) {
    lazyPreemptGroup(g, g->name);
  }
}

void HivedCore::deleteAllocatedGroup(Group* g, const std::string& podKey) {
  (void)podKey;
  for (auto& [leafNum, pods] : g->physPlacement) {
    (void)leafNum;
    for (auto& pod : pods) {
      for (PhysicalCell* c : pod) {
        if (c == nullptr) continue;
        c->usingGroup = nullptr;
        if (c->state == CState::Used) {
          releaseLeafCell(c, g->vc);
          setCellState(c, CState::Free);
        } else {
          // Reserving: the cell was already re-allocated to a preempting group
          setCellState(c, CState::Reserved);
        }
      }
    }
  }
  groups_.erase(g->name);
}

void HivedCore::createPreemptingGroup(const PodSpec& s, const Placement<PhysicalCell>& phys,
                                      const Placement<VirtualCell>& virt,
                                      const std::string& podKey) {
  auto group = newGroup(s, GState::Preempting);
  Group* g = group.get();
  g->physPlacement = phys;
  g->virtPlacement = virt;
  for (auto& [leafNum, pods] : phys) {
    auto vIt = virt.find(leafNum);
    for (size_t pi = 0; pi < pods.size(); pi++) {
      for (size_t li = 0; li < pods[pi].size(); li++) {
        PhysicalCell* pc = pods[pi][li];
        VirtualCell* vc = vIt->second[pi][li];
        if (pc->state == CState::Used) {
          Group* usingGroup = pc->usingGroup;
          releaseLeafCell(pc, usingGroup->vc);
          usingGroup->state = GState::BeingPreempted;
        }
        allocateLeafCell(pc, vc, s.priority, g->vc);
        pc->reservingGroup = g;
        if (pc->state == CState::Used) {
          setCellState(pc, CState::Reserving);
        } else {  // Free
          setCellState(pc, CState::Reserved);
        }
      }
    }
  }
  g->preemptingPods.insert(podKey);
  groups_[s.groupName] = std::move(group);
}

void HivedCore::deletePreemptingGroup(Group* g, const std::string& podKey) {
  (void)podKey;
  for (auto& [leafNum, pods] : g->physPlacement) {
    (void)leafNum;
    for (auto& pod : pods) {
      for (PhysicalCell* pc : pod) {
        releaseLeafCell(pc, g->vc);
        pc->reservingGroup = nullptr;
        if (pc->state == CState::Reserving) {
          setCellState(pc, CState::Used);
          // return the cell to the group being preempted
          Group* beingPreempted = pc->usingGroup;
          VirtualCell* vc = nullptr;
          if (beingPreempted->hasVirtualPlacement) {
            vc = retrieveVirtualCell(beingPreempted->physPlacement, beingPreempted->virtPlacement,
                                     pc);
          }
          allocateLeafCell(pc, vc, beingPreempted->priority, beingPreempted->vc);
        } else {  // Reserved
          setCellState(pc, CState::Free);
        }
      }
    }
  }
  // groups being preempted solely by g return to Allocated
  for (auto& [name, other] : groups_) {
    (void)name;
    if (other->state == GState::BeingPreempted) {
      bool stillPreempted = false;
      for (auto& [ln, pods] : other->physPlacement) {
        (void)ln;
        for (auto& pod : pods) {
          for (PhysicalCell* c : pod) {
            if (c != nullptr && (c->state == CState::Reserving || c->state == CState::Reserved)) {
              stillPreempted = true;
            }
          }
        }
      }
      if (!stillPreempted) other->state = GState::Allocated;
    }
  }
  groups_.erase(g->name);
}

void HivedCore::allocatePreemptingGroup(Group* g, const std::string& podKey) {
  (void)podKey;
  for (auto& [leafNum, pods] : g->physPlacement) {
    (void)leafNum;
    for (auto& pod : pods) {
      for (PhysicalCell* pc : pod) {
        pc->reservingGroup = nullptr;
        pc->usingGroup = g;
        setCellState(pc, CState::Used);
      }
    }
  }
  g->state = GState::Allocated;
  g->preemptingPods.clear();
}

Placement<VirtualCell> HivedCore::lazyPreemptGroup(Group* victim, const std::string& preemptor) {
  if (victim->hasVirtualPlacement) {
    for (auto& [leafNum, pods] : victim->virtPlacement) {
      (void)leafNum;
      for (auto& pod : pods) {
        for (VirtualCell* v : pod) {
          if (v != nullptr && v->phys != nullptr) {
            PhysicalCell* p = v->phys;
            releaseLeafCell(p, victim->vc);
            allocateLeafCell(p, nullptr, kOpportunisticPriority, victim->vc);
          }
        }
      }
    }
  }
  Placement<VirtualCell> original = victim->virtPlacement;
  victim->virtPlacement.clear();
  victim->hasVirtualPlacement = false;
  victim->lazyStatus = LazyPreemptionStatus{preemptor, ""};
  return original;
}

void HivedCore::lazyPreemptCell(VirtualCell* c, const std::string& preemptor) {
  if (c->level == kLowestLevel && c->phys != nullptr && c->phys->state == CState::Used &&
      c->phys->usingGroup != nullptr) {
    lazyPreemptGroup(c->phys->usingGroup, preemptor);
  }
  for (Cell* child : c->children) {
    lazyPreemptCell(static_cast<VirtualCell*>(child), preemptor);
  }
}

void HivedCore::revertLazyPreempt(Group* g, const Placement<VirtualCell>& virt) {
  for (auto& [leafNum, pods] : g->physPlacement) {
    auto vIt = virt.find(leafNum);
    if (vIt == virt.end()) continue;
    for (size_t pi = 0; pi < pods.size(); pi++) {
      for (size_t li = 0; li < pods[pi].size(); li++) {
        PhysicalCell* p = pods[pi][li];
        if (p == nullptr) continue;
        VirtualCell* v = vIt->second[pi][li];
        releaseLeafCell(p, g->vc);
        allocateLeafCell(p, v, g->priority, g->vc);
      }
    }
  }
  g->virtPlacement = virt;
  g->hasVirtualPlacement = true;
  g->lazyStatus.reset();
}

// ---------------------------------------------------------------------------
// Leaf-cell allocate/release
// ---------------------------------------------------------------------------

std::pair<PhysicalCell*, VirtualCell*> HivedCore::findAllocatedLeafCell(
    int index, const PodPlacementInfo& placement, const std::string& chain, const PodSpec& s,
    Group* group, const std::string& podKey, bool* lazyPreempt, bool* isOpportunistic) {
  (void)podKey;
  *lazyPreempt = false;
  *isOpportunistic = false;
  int leafIndex = placement.leafIndices[index];
  PhysicalCell* p = findPhysicalLeafCell(fullCellList_, chain, placement.node, leafIndex);
  if (p == nullptr) return {nullptr, nullptr};
  if (placement.preassignedTypes.empty()) {
    *lazyPreempt = true;
    return {p, nullptr};
  }
  if (!group->hasVirtualPlacement) {
    return {p, nullptr};
  }
  const std::string& preassignedType =
      index < static_cast<int>(placement.preassignedTypes.size()) ? placement.preassignedTypes[index]
                                                                  : std::string();
  if (preassignedType.empty()) {
    *isOpportunistic = true;
    return {p, nullptr};
  }
  int preassignedLevel = -1;
  auto typesIt = cellTypes_.find(p->chain);
  if (typesIt != cellTypes_.end()) {
    for (auto& [l, t] : typesIt->second) {
      if (t == preassignedType) preassignedLevel = l;
    }
  }
  if (preassignedLevel > 0) {
    // Ownership-conflict check (reconfiguration/recovery semantics,
    // reference hived_algorithm.go:1036-1038): the recorded placement's
    // preassigned-level cell may now sit INSIDE another preassigned's
    // domain — e.g. after a restart, an earlier-replayed pod re-took a
    // whole-node preassigned covering this pod's old quad (replay order is
    // arbitrary). Binding across that boundary would double-own the
    // subtree; fuzz-found via crash-recovery fuzzing. Lazy-preempt instead.
    PhysicalCell* pre = p;
    while (pre->parent != nullptr && pre->level < preassignedLevel) {
      pre = static_cast<PhysicalCell*>(pre->parent);
    }
    for (PhysicalCell* a = static_cast<PhysicalCell*>(pre->parent); a != nullptr;
         a = static_cast<PhysicalCell*>(a->parent)) {
      if (a->virt != nullptr) {
        *lazyPreempt = true;
        return {p, nullptr};
      }
    }
    // inverse direction: an UNBOUND preassigned candidate with a bound
    // descendant means another preassigned tree already owns part of this
    // subtree (e.g. an earlier-replayed pod bound its own quad inside) —
    // freshly binding `pre` on top would nest ownerships (fuzz-found:
    // left the sibling quad in the free list while bound)
    if (pre->virt == nullptr) {
      std::function<bool(PhysicalCell*)> anyBound = [&](PhysicalCell* c) -> bool {
        if (c->virt != nullptr) return true;
        for (Cell* ch : c->children) {
          if (anyBound(static_cast<PhysicalCell*>(ch))) return true;
        }
        return false;
      };
      for (Cell* ch : pre->children) {
        if (anyBound(static_cast<PhysicalCell*>(ch))) {
          *lazyPreempt = true;
          return {p, nullptr};
        }
      }
    }
  }
  VirtualCell* v = nullptr;
  std::string message;
  if (preassignedLevel < 0) {
    message = "preassigned cell type " + preassignedType + " not found in chain " + p->chain;
  } else if (!vcSchedulers_.count(s.vc)) {
    message = "VC " + s.vc + " not found";
  } else {
    IntraVCScheduler& vcs = vcSchedulers_[s.vc];
    const ChainCellList* vccl = nullptr;
    if (!s.pinnedCellId.empty()) {
      auto it = vcs.pinned.find(s.pinnedCellId);
      if (it != vcs.pinned.end()) vccl = &it->second;
    } else {
      auto it = vcs.nonPinnedPreassigned.find(p->chain);
      if (it != vcs.nonPinnedPreassigned.end()) vccl = &it->second;
    }
    if (vccl == nullptr) {
      message = "VC " + s.vc + " has no cell for chain " + p->chain;
    } else {
      v = mapPhysicalCellToVirtual(p, *vccl, preassignedLevel, s.priority, &message);
    }
  }
  if (v == nullptr) {
    *lazyPreempt = true;
    return {p, nullptr};
  }
  if (v->vc != s.vc) {
    // the physical cell is already bound into ANOTHER VC's virtual tree
    // (cross-VC conflict after recovery/reconfiguration): never adopt a
    // foreign binding — lazy-preempt this group instead
    *lazyPreempt = true;
    return {p, nullptr};
  }
  return {p, v};
}

std::pair<bool, std::string> HivedCore::allocateLeafCell(PhysicalCell* p, VirtualCell* v,
                                                         int priority, const std::string& vc) {
  bool safetyOk = true;
  std::string reason;
  if (v != nullptr) {
    setCellPriority(v, priority);
    updateUsedLeafCellNumAtPriority(v, priority, true);
    setCellPriority(p, priority);
    updateUsedLeafCellNumAtPriority(p, priority, true);
    VirtualCell* pac = v->preassigned;
    if (pac->phys != nullptr && pac->phys->virt == pac &&
        pac->phys->priority < kMinGuaranteedPriority &&
        vcDoomedBadCells_[vc][p->chain].contains(pac->phys, pac->phys->level)) {
      // The preassigned cell was doomed onto a bad physical cell AFTER this
      // placement was computed against a different physical target (the doom
      // machinery can fire mid-commit, e.g. inside an overlapping
      // preemptor's cancellation). If the placement is not actually ON the
      // doomed cell, undo the doomed binding so the real binding can form.
      // Found by fuzzing: without this, the leaf binds under a preassigned
      // pointing at an unrelated physical cell and the real target stays in
      // the free list carrying a guaranteed priority.
      PhysicalCell* anc = p;
      while (anc->parent != nullptr && anc != pac->phys && anc->level < pac->phys->level) {
        anc = static_cast<PhysicalCell*>(anc->parent);
      }
      if (anc != pac->phys) {
        PhysicalCell* doomed = pac->phys;
        pac->phys = nullptr;
        doomed->virt = nullptr;
        vcDoomedBadCells_[vc][p->chain].remove(doomed, doomed->level);
        allVCDoomedBadCellNum_[p->chain][doomed->level]--;
        releasePreassignedCell(doomed, vc, true);
      }
    }
    bool preassignedNewlyBound = pac->phys == nullptr;
    bool preassignedProducible = false;
    if (preassignedNewlyBound) {
      // The cell this bind will attach the preassigned to is p's ancestor at
      // the preassigned's level. Only run the accounting allocation if that
      // cell is actually producible from the free list: an earlier unbind
      // whose matching release was skipped (priority roll-up from sibling
      // subtrees kept the accounting "allocated") leaves the cell
      // allocated-but-unbound — re-allocating it here would double-count
      // and throw mid-commit (fuzz-found). Re-binding without re-allocating
      // heals that state.
      PhysicalCell* anc = p;
      while (anc->parent != nullptr && anc->level < pac->level) {
        anc = static_cast<PhysicalCell*>(anc->parent);
      }
      // authoritative membership: anc, or an unbound ancestor of it, must be
      // DIRECTLY in the free list (inFreeCellList's split-flag shortcut
      // reports stale true for an allocated-but-unbound cell)
      auto& fl = freeCellList_[p->chain];
      for (PhysicalCell* a = anc; a != nullptr && a->virt == nullptr;
           a = static_cast<PhysicalCell*>(a->parent)) {
        if (a->level <= fl.top() && fl.contains(a, a->level)) {
          preassignedProducible = true;
          break;
        }
      }
    }
    if (p->virt == nullptr) {
      // binding may already exist if the cell is bad
      bindCell(p, v);
    }
    if (preassignedNewlyBound && preassignedProducible) {
      std::tie(safetyOk, reason) = allocatePreassignedCell(pac->phys, vc, false);
    }
  } else {
    setCellPriority(p, kOpportunisticPriority);
    updateUsedLeafCellNumAtPriority(p, kOpportunisticPriority, true);
    p->otVC = vc;
  }
  return {safetyOk, reason};
}

void HivedCore::releaseLeafCell(PhysicalCell* p, const std::string& vc) {
  (void)vc;  // the owning VC is derived from the binding, not the releasing
             // group: an opportunistic pod of VC-A can sit on a leaf carrying
             // VC-B's bad-cell binding, and must not touch VC-B's accounting
  if (!p->otVC.empty() || p->virt == nullptr) {
    // opportunistic allocation: physical side only; a bad-cell binding (if
    // any) belongs to the doomed-bad machinery and stays
    p->otVC.clear();
  } else {
    VirtualCell* v = p->virt;
    const std::string& owner = v->vc;
    updateUsedLeafCellNumAtPriority(v, v->priority, false);
    setCellPriority(v, kFreePriority);
    PhysicalCell* preassignedPhysical = v->preassigned->phys;
    if (p->healthy) {
      // keep the binding if the cell is bad
      unbindCell(p);
    }
    if (preassignedPhysical != nullptr && !preassignedPhysical->pinned &&
        v->preassigned->priority < kMinGuaranteedPriority &&
        !vcDoomedBadCells_[owner][preassignedPhysical->chain].contains(
            preassignedPhysical, preassignedPhysical->level)) {
      releasePreassignedCell(preassignedPhysical, owner, false);
    } else if (mapDebugRelease()) {
      fprintf(stderr, "[rel] keep preassigned %s: phys=%d pinned=%d prio=%d doomed=%d (leaf %s)\n",
              v->preassigned->address.c_str(), preassignedPhysical != nullptr,
              preassignedPhysical != nullptr && preassignedPhysical->pinned,
              v->preassigned->priority,
              preassignedPhysical != nullptr &&
                  vcDoomedBadCells_[owner][preassignedPhysical->chain].contains(
                      preassignedPhysical, preassignedPhysical->level),
              p->address.c_str());
    }
  }
  updateUsedLeafCellNumAtPriority(p, p->priority, false);
  setCellPriority(p, kFreePriority);
}

}  // namespace hived
