// Buddy allocation, virtual->physical cell mapping, free-list split/merge,
// VC-safety accounting and bad-cell (doomed-bad) machinery.
// Semantics parity: pkg/algorithm/cell_allocation.go:34-315 and
// pkg/algorithm/hived_algorithm.go:453-653, 1354-1565.
#include <cstdio>
#include <cstdlib>

#include "core.hpp"

namespace hived {

static bool mapDebug() {
  static int v = -1;
  if (v < 0) v = getenv("HIVED_DEBUG_MAP") ? 1 : 0;
  return v == 1;
}

namespace {

// Refcounted claims with an undo journal: physical leaves (and their
// ancestors) claimed by picks of THIS mapping round. The journal makes
// backtracking EXACT — abandoned branches are fully unclaimed. That matters
// twice: stale claims used to merely bias the packing sort, but with the
// honorLinks hard constraint a stale claim on a degraded link's endpoint
// wrongly vetoes its peer on a later branch and turns findable clean
// mappings into dirty fallbacks (found by the dirty-only-when-forced
// property oracle, seed 1999-class: the 4-leaf vertex retried on another
// quad still saw its abandoned quad's claims).
struct ClaimSet {
  std::unordered_map<PhysicalCell*, int> refs;
  std::vector<PhysicalCell*> log;  // claimed leaves, in order

  bool count(PhysicalCell* c) const { return refs.count(c) > 0; }
  size_t checkpoint() const { return log.size(); }
  void claim(PhysicalCell* leaf) {
    log.push_back(leaf);
    for (PhysicalCell* a = leaf; a != nullptr; a = static_cast<PhysicalCell*>(a->parent)) {
      refs[a]++;
    }
  }
  void unwindTo(size_t cp) {
    while (log.size() > cp) {
      PhysicalCell* leaf = log.back();
      log.pop_back();
      for (PhysicalCell* a = leaf; a != nullptr; a = static_cast<PhysicalCell*>(a->parent)) {
        auto it = refs.find(a);
        if (--it->second == 0) refs.erase(it);
      }
    }
  }
};

struct AllocCtx {
  const std::set<std::string>& suggested;
  bool ignoreSuggested;
  std::unordered_map<VirtualCell*, PhysicalCell*>& bindings;
  // exact claim tracking (see ClaimSet): drives the packing sort keys AND
  // the honorLinks pick-time link constraint
  ClaimSet& claimed;
  // minimum measured HBM per leaf (0 = any): leaves below are unusable
  long long minHbm = 0;
  // when true, a leaf candidate whose degraded-link peer is already claimed
  // by this mapping round is unusable (the gang would all-reduce over the
  // sick link); the backtracking search then finds a clean mapping or fails,
  // and the caller retries with honorLinks=false (capacity over quality)
  bool honorLinks = false;
  // the clean-shape world the virtual placement was computed in (honorLinks
  // runs): its excluded leaves are unusable here too, keeping the mapper's
  // choices consistent with the shape — without this, a mapper pick of an
  // excluded endpoint forces later cross-group conflicts the per-group
  // backtracking cannot undo
  const CleanShapeWorld* world = nullptr;
};

// Number of gang leaves that will land under this vertex's physical binding
// (vertex-tree leaves are the placed level-1 cells).
int vertexLeafDemand(const BindingVertex* v) {
  if (v->children.empty()) return v->cell->totalLeaf;
  int n = 0;
  for (auto& ch : v->children) n += vertexLeafDemand(ch.get());
  return n;
}

void markClaimed(const AllocCtx& ctx, PhysicalCell* c) { ctx.claimed.claim(c); }

// honorLinks hard constraint, checked at PICK time (claims accumulate while
// the backtracking search runs): a leaf whose degraded-link peer is already
// claimed by this round would put the gang's collective on the sick link.
bool leafLinkConflict(const AllocCtx& ctx, PhysicalCell* leaf) {
  if (!ctx.honorLinks || leaf->badLinkPeers.empty()) return false;
  for (PhysicalCell* peer : leaf->badLinkPeers) {
    if (ctx.claimed.count(peer)) return true;
  }
  return false;
}

// Usable = unbound, not a known-bad single-node cell, and (unless ignoring
// suggestions) at least one node within the suggested set. Sorted by
// opportunistic usage ascending to minimize preemption of opportunistic pods.
std::vector<PhysicalCell*> getUsablePhysicalCells(const std::vector<Cell*>& candidates,
                                                  int numNeeded, const AllocCtx& ctx,
                                                  int demandLeaves = 0) {
  std::vector<PhysicalCell*> usable;
  for (Cell* cc : candidates) {
    auto* c = static_cast<PhysicalCell*>(cc);
    if (c->virt != nullptr) continue;
    if (c->nodes.size() == 1 && !c->healthy) continue;
    if (ctx.minHbm > 0 && c->level == kLowestLevel && c->hbmBytes > 0 &&
        c->hbmBytes < ctx.minHbm) {
      continue;  // leaf's measured HBM falls short of the request's demand
    }
    if (ctx.honorLinks && ctx.world != nullptr && c->level == kLowestLevel &&
        ctx.world->excluded.count(c)) {
      continue;  // endpoint the clean-shape world avoids
    }
    if (!ctx.ignoreSuggested) {
      bool anySuggested = false;
      for (auto& n : c->nodes) {
        if (ctx.suggested.count(n)) {
          anySuggested = true;
          break;
        }
      }
      if (!anySuggested) continue;
    }
    usable.push_back(c);
  }
  if (static_cast<int>(usable.size()) < numNeeded) return {};
  // Buddy packing first: prefer candidates in already-fragmented regions
  // (fewer untouched free buddies), so intact higher-level free cells — other
  // VCs' guarantees — survive. Fuzz-found: picking a pair under the last
  // intact quad instead of one in a fragmented quad stranded a quad-level
  // guarantee (totalLeft < allVCFree). Then opportunistic usage ascending,
  // to minimize preemption of opportunistic pods (reference
  // cell_allocation.go:252-315 ordering).
  auto freeBuddies = [&ctx](PhysicalCell* c) {
    if (c->parent == nullptr) return 0;
    int n = 0;
    for (Cell* b : c->parent->children) {
      auto* pb = static_cast<PhysicalCell*>(b);
      if (pb != c && pb->priority == kFreePriority && pb->virt == nullptr &&
          !ctx.claimed.count(pb)) {
        n++;
      }
    }
    return n;
  };
  // Same-parent-as-claimed is the strongest packing signal: when every
  // candidate looks equally fragmented (e.g. all buddies squatted by
  // opportunistic pods at priority -1, so freeBuddies ties at 0), the
  // group's Nth cell must still land next to its earlier picks or the
  // allocation splits one higher-level free cell per pod (fuzz-found
  // VC-safety break).
  auto parentClaimed = [&ctx](PhysicalCell* c) {
    return c->parent != nullptr && ctx.claimed.count(static_cast<PhysicalCell*>(c->parent)) > 0;
  };
  // xGMI link preference (weakest key, after the safety-driven packing keys
  // and the opportunistic-disruption key): a multi-leaf vertex avoids
  // candidates containing degraded links (its leaves would all-reduce over
  // the sick link); a single-leaf vertex PREFERS them (parking 1-GPU work on
  // degraded pairs keeps clean pairs free for gangs).
  auto linkKey = [demandLeaves](PhysicalCell* c) {
    return demandLeaves >= 2 ? c->badLinksUnder : -c->badLinksUnder;
  };
  std::stable_sort(usable.begin(), usable.end(), [&](PhysicalCell* a, PhysicalCell* b) {
    bool pa = parentClaimed(a), pb = parentClaimed(b);
    if (pa != pb) return pa;
    int fa = freeBuddies(a), fb = freeBuddies(b);
    if (fa != fb) return fa < fb;
    int oa = a->usedAt(kOpportunisticPriority), ob = b->usedAt(kOpportunisticPriority);
    if (oa != ob) return oa < ob;
    return demandLeaves != 0 && linkKey(a) < linkKey(b);
  });
  if (mapDebug()) {
    fprintf(stderr, "[sort]");
    for (auto* u : usable)
      fprintf(stderr, " %s(fb=%d,ot=%d)", u->address.c_str(), freeBuddies(u),
              u->usedAt(kOpportunisticPriority));
    fprintf(stderr, "\n");
  }
  return usable;
}

// Backtracking bipartite match of virtual cells onto candidate physical cells,
// recursing into children to preserve the intra-cell topology.
bool mapVirtualCellsToPhysical(const std::vector<BindingVertex*>& cellsIn,
                               const std::vector<Cell*>& candidatesIn, const AllocCtx& ctx,
                               bool returnPicked, std::vector<PhysicalCell*>* picked) {
  // Multi-leaf vertices map first (demand descending) so they get the
  // link-clean candidates and single-leaf vertices take the leftovers
  // (incl. degraded pairs, which suit them fine).
  std::vector<BindingVertex*> cells = cellsIn;
  int maxDemand = 0;
  {
    std::vector<int> demand(cells.size());
    for (size_t i = 0; i < cells.size(); i++) {
      demand[i] = vertexLeafDemand(cells[i]);
      maxDemand = std::max(maxDemand, demand[i]);
    }
    std::stable_sort(cells.begin(), cells.end(), [&](BindingVertex* a, BindingVertex* b) {
      return vertexLeafDemand(a) > vertexLeafDemand(b);
    });
  }
  std::vector<PhysicalCell*> candidates =
      getUsablePhysicalCells(candidatesIn, static_cast<int>(cells.size()), ctx, maxDemand);
  if (candidates.empty() && !cells.empty()) return false;
  int n = static_cast<int>(cells.size());
  int m = static_cast<int>(candidates.size());
  std::vector<int> pickedIdx(n, 0);
  std::vector<size_t> claimCp(n, 0);  // claims checkpoint before each vertex's pick
  std::vector<bool> used(m, false);
  int cellIndex = 0;
  while (cellIndex >= 0) {
    int candidateIndex;
    for (candidateIndex = pickedIdx[cellIndex]; candidateIndex < m; candidateIndex++) {
      if (used[candidateIndex]) continue;
      PhysicalCell* candidate = candidates[candidateIndex];
      size_t attemptCp = ctx.claimed.checkpoint();
      bool ok;
      if (candidate->level == kLowestLevel) {
        if (leafLinkConflict(ctx, candidate)) continue;
        ok = true;
        ctx.bindings[cells[cellIndex]->cell] = candidate;
        markClaimed(ctx, candidate);
      } else {
        std::vector<BindingVertex*> childVerts;
        childVerts.reserve(cells[cellIndex]->children.size());
        for (auto& ch : cells[cellIndex]->children) childVerts.push_back(ch.get());
        ok = mapVirtualCellsToPhysical(childVerts, candidate->children, ctx, false, nullptr);
        // abandoned branch: unwind its claims exactly, or the honorLinks
        // link constraint would see ghosts on the next branch
        if (!ok) ctx.claimed.unwindTo(attemptCp);
      }
      if (ok) {
        pickedIdx[cellIndex] = candidateIndex;
        claimCp[cellIndex] = attemptCp;
        used[candidateIndex] = true;
        if (cellIndex == n - 1) {
          if (returnPicked) {
            picked->clear();
            for (int i = 0; i < n; i++) picked->push_back(candidates[pickedIdx[i]]);
          }
          return true;
        }
        break;
      }
    }
    if (candidateIndex == m) {
      cellIndex--;
      if (cellIndex >= 0) {
        used[pickedIdx[cellIndex]] = false;
        ctx.claimed.unwindTo(claimCp[cellIndex]);
        pickedIdx[cellIndex]++;
      }
    } else {
      cellIndex++;
      if (cellIndex < n) pickedIdx[cellIndex] = 0;
    }
  }
  return false;
}

// Backtracking buddy allocation: allocate a free physical cell at the virtual
// cell's level, splitting one higher-level free cell per level on the way
// down. Backtracking handles cells that are bad or outside suggested nodes.
bool buddyAlloc(BindingVertex* cell, ChainCellList& freeList, int currentLevel,
                const AllocCtx& ctx) {
  if (mapDebug()) {
    fprintf(stderr, "[map] buddyAlloc virt level=%d at level=%d candidates:", cell->cell->level,
            currentLevel);
    for (Cell* c : freeList.at(currentLevel)) fprintf(stderr, " %s", c->address.c_str());
    fprintf(stderr, "\n");
  }
  if (currentLevel == cell->cell->level) {
    std::vector<PhysicalCell*> picked;
    if (mapVirtualCellsToPhysical({cell}, freeList.at(currentLevel), ctx, true, &picked)) {
      for (PhysicalCell* c : picked) freeList.remove(c, currentLevel);
      if (mapDebug())
        fprintf(stderr, "[map]  -> picked %s\n", picked.empty() ? "?" : picked[0]->address.c_str());
      return true;
    }
    if (mapDebug()) fprintf(stderr, "[map]  -> no usable candidate at own level\n");
    return false;
  }
  std::vector<PhysicalCell*> freeCells =
      getUsablePhysicalCells(freeList.at(currentLevel), 1, ctx, vertexLeafDemand(cell));
  if (freeCells.empty()) return false;
  for (PhysicalCell* c : freeCells) {
    auto saved = freeList.at(currentLevel - 1);
    for (Cell* child : c->children) freeList.add(child, currentLevel - 1);
    if (buddyAlloc(cell, freeList, currentLevel - 1, ctx)) {
      freeList.remove(c, currentLevel);
      return true;
    }
    freeList.at(currentLevel - 1) = saved;
  }
  return false;
}

// When buddy alloc fails due to bad / non-suggested cells, split higher-level
// free cells beyond the buddy path, but only as many as safety allows
// (splittable = free cells minus cells the VCs still need reserved).
bool safeRelaxedBuddyAlloc(BindingVertex* cell, ChainCellList& freeList,
                           std::map<int, int>& freeCellNum, int currentLevel,
                           const AllocCtx& ctx) {
  if (mapDebug()) fprintf(stderr, "[map] safeRelaxed for virt level=%d\n", currentLevel);
  int top = freeList.top();
  std::map<int, int> splittableNum;
  Cell* splittableCell = nullptr;
  for (int i = top; i > currentLevel; i--) {
    splittableNum[i] =
        static_cast<int>(freeList.at(i).size()) - (freeCellNum.count(i) ? freeCellNum[i] : 0);
    if (i < top && splittableCell != nullptr) {
      splittableNum[i] += splittableNum[i + 1] * static_cast<int>(splittableCell->children.size());
    }
    if (splittableCell == nullptr && !freeList.at(i).empty()) {
      splittableCell = freeList.at(i)[0];
    } else if (splittableCell != nullptr) {
      splittableCell = splittableCell->children[0];
    }
    if (splittableNum[i] < 0) {
      throw HivedError::Internal("VC Safety Broken: level " + std::to_string(i) +
                                 " cell is unsplittable, splittableNum=" +
                                 std::to_string(splittableNum[i]));
    }
  }
  for (int l = currentLevel + 1; l <= top; l++) {
    int cellNum = std::min(static_cast<int>(freeList.at(l).size()), splittableNum[l]);
    if (cellNum <= 0) continue;
    std::vector<Cell*> splitList;
    for (int i = 0; i < cellNum; i++) {
      Cell* c = freeList.at(l)[0];
      splitList.push_back(c);
      freeList.remove(c, l);
    }
    splittableNum[l] -= cellNum;
    for (int sl = l; sl > currentLevel; sl--) {
      std::vector<Cell*> childrenList;
      for (Cell* sc : splitList) {
        for (Cell* ch : sc->children) childrenList.push_back(ch);
      }
      splitList = std::move(childrenList);
    }
    // prepend the split cells so they are tried first
    auto& cur = freeList.at(currentLevel);
    cur.insert(cur.begin(), splitList.begin(), splitList.end());
    std::vector<PhysicalCell*> picked;
    if (mapVirtualCellsToPhysical({cell}, cur, ctx, true, &picked)) {
      for (PhysicalCell* c : picked) freeList.remove(c, currentLevel);
      if (mapDebug())
        fprintf(stderr, "[map] safeRelaxed picked %s\n",
                picked.empty() ? "?" : picked[0]->address.c_str());
      return true;
    }
  }
  return false;
}

int getLowestFreeCellLevel(const ChainCellList& freeList, int l) {
  for (; l <= freeList.top(); l++) {
    if (!freeList.at(l).empty()) return l;
  }
  throw HivedError::Internal("VC Safety Broken: free cell not found even at the highest level");
}

}  // namespace

bool HivedCore::mapVirtualPlacementToPhysical(
    std::vector<BindingVertex*>& preassigned, std::vector<std::vector<BindingVertex*>>& nonPreassigned,
    ChainCellList freeList, std::map<int, int> freeCellNum,
    const std::set<std::string>& suggestedNodes, bool ignoreSuggestedNodes,
    std::unordered_map<VirtualCell*, PhysicalCell*>& bindings, long long minHbmBytes,
    bool honorLinks, const CleanShapeWorld* world) {
  ClaimSet claimed;
  AllocCtx ctx{suggestedNodes, ignoreSuggestedNodes, bindings, claimed,
               minHbmBytes, honorLinks, honorLinks ? world : nullptr};
  // pre-claim physical leaves ALREADY bound for this placement (the caller
  // seeds `bindings` with bound virtual leaves, which skip the vertex trees)
  // so the link constraint sees the whole gang, not just the new picks
  if (honorLinks) {
    for (auto& [v, p] : bindings) {
      (void)v;
      if (p != nullptr) markClaimed(ctx, p);
    }
  }
  for (BindingVertex* c : preassigned) {
    if (!buddyAlloc(c, freeList, getLowestFreeCellLevel(freeList, c->cell->level), ctx)) {
      if (!safeRelaxedBuddyAlloc(c, freeList, freeCellNum, c->cell->level, ctx)) {
        return false;
      }
    } else {
      freeCellNum[c->cell->level]--;
    }
  }
  for (auto& cells : nonPreassigned) {
    auto* parentVirtual = static_cast<VirtualCell*>(cells[0]->cell->parent);
    if (parentVirtual == nullptr || parentVirtual->phys == nullptr) return false;
    if (!mapVirtualCellsToPhysical(cells, parentVirtual->phys->children, ctx, false, nullptr)) {
      return false;
    }
  }
  // HARD post-mapping safety guard: the mutated free-list COPY reflects the
  // post-commit physical free state for this placement. At every level, the
  // cells still producible from it must cover every OTHER VC's remaining free
  // quota (allVCFree minus what this placement's new preassigned bindings
  // consume). The packing heuristics above make placements good; this guard
  // makes them SAFE — fuzzing kept finding candidate orders (bad cells +
  // opportunistic squatters + relaxed splits) where a group's cells landed
  // across multiple higher-level free cells and stranded another VC's
  // guarantee. Rejecting here turns that into a correct "wait".
  if (!preassigned.empty()) {
    const std::string& chain = preassigned[0]->cell->chain;
    auto avfIt = allVCFreeCellNum_.find(chain);
    if (avfIt != allVCFreeCellNum_.end()) {
      std::map<int, int> newPre;
      for (BindingVertex* c : preassigned) newPre[c->cell->level]++;
      int top = freeList.top();
      int producible = 0;
      for (int l = top; l >= kLowestLevel; l--) {
        if (l < top && !fullCellList_[chain].at(l + 1).empty()) {
          producible *= static_cast<int>(fullCellList_[chain].at(l + 1)[0]->children.size());
        }
        producible += static_cast<int>(freeList.at(l).size());
        auto needIt = avfIt->second.find(l);
        int need = needIt == avfIt->second.end() ? 0 : needIt->second;
        need -= newPre.count(l) ? newPre[l] : 0;
        if (producible < need) return false;
      }
    }
  }
  return true;
}

// --- free-list split/merge -------------------------------------------------

// Remove a cell from the free list, splitting unsplit ancestors on the way up.
// Returns the highest level at which a cell was removed.
int HivedCore::removeCellFromFreeList(PhysicalCell* c) {
  ChainCellList& freeList = freeCellList_[c->chain];
  bool terminate = false;
  for (;;) {
    int l = c->level;
    auto* parent = static_cast<PhysicalCell*>(c->parent);
    if (parent != nullptr) {
      if (parent->split) {
        terminate = true;
      } else {
        for (Cell* child : parent->children) freeList.add(child, l);
        parent->split = true;
      }
    } else {
      terminate = true;
    }
    freeList.remove(c, l);
    if (terminate) return l;
    c = parent;
  }
}

// Add a cell back to the free list, merging buddies into the parent while all
// of them are free. Returns the highest level at which a cell was added.
int HivedCore::addCellToFreeList(PhysicalCell* c) {
  ChainCellList& freeList = freeCellList_[c->chain];
  bool terminate = false;
  for (;;) {
    int l = c->level;
    auto* parent = static_cast<PhysicalCell*>(c->parent);
    if (parent != nullptr) {
      bool allBuddyFree = true;
      for (Cell* buddy : parent->children) {
        if (buddy != c && !freeList.contains(buddy, l)) {
          allBuddyFree = false;
          break;
        }
      }
      if (!allBuddyFree) {
        terminate = true;
      } else {
        for (Cell* buddy : parent->children) {
          if (buddy != c) freeList.remove(buddy, l);
        }
        parent->split = false;
      }
    } else {
      terminate = true;
    }
    if (terminate) {
      freeList.add(c, l);
      return l;
    }
    c = parent;
  }
}

// --- preassigned-cell allocation + safety & doomed-bad accounting -----------

std::pair<bool, std::string> HivedCore::allocatePreassignedCell(PhysicalCell* c,
                                                                const std::string& vcn,
                                                                bool doomedBad) {
  bool safetyOk = true;
  std::string reason;
  const std::string& chain = c->chain;
  int level = c->level;
  vcFreeCellNum_[vcn][chain][level]--;
  allVCFreeCellNum_[chain][level]--;
  totalLeftCellNum_[chain][level]--;
  int splitLevelUpTo = removeCellFromFreeList(c);

  Cell* parent = c->parent;
  for (int l = level + 1; l <= splitLevelUpTo; l++) {
    totalLeftCellNum_[chain][l]--;
    if (totalLeftCellNum_[chain][l] < allVCFreeCellNum_[chain][l]) {
      safetyOk = false;
      reason = "Adding pod would lead to broken safety: cell type " + cellTypes_[chain][l] + ", " +
               std::to_string(totalLeftCellNum_[chain][l]) + " left, " +
               std::to_string(allVCFreeCellNum_[chain][l]) + " free cells in all VCs";
    }
    if (!static_cast<PhysicalCell*>(parent)->healthy) {
      // parent bad: healthy-free count unchanged; just drop it from bad frees
      badFreeCells_[chain].remove(parent, l);
    } else {
      // parent healthy: healthy-free count decreased; maybe doom VC cells
      tryBindDoomedBadCell(chain, l);
    }
    parent = parent->parent;
  }
  if (!c->healthy) {
    allocateBadCell(c);
    if (!doomedBad) tryUnbindDoomedBadCell(chain, level);
  } else {
    tryBindDoomedBadCell(chain, level);
  }
  int numToReduce = static_cast<int>(c->children.size());
  for (int l = level - 1; l >= kLowestLevel; l--) {
    totalLeftCellNum_[chain][l] -= numToReduce;
    if (totalLeftCellNum_[chain][l] < allVCFreeCellNum_[chain][l]) {
      safetyOk = false;
      reason = "Adding pod would lead to broken safety: cell type " + cellTypes_[chain][l] + ", " +
               std::to_string(totalLeftCellNum_[chain][l]) + " left, " +
               std::to_string(allVCFreeCellNum_[chain][l]) + " free cells in all VCs";
    }
    if (!doomedBad) tryBindDoomedBadCell(chain, l);
    numToReduce *= static_cast<int>(fullCellList_[chain].at(l)[0]->children.size());
  }
  return {safetyOk, reason};
}

void HivedCore::releasePreassignedCell(PhysicalCell* c, const std::string& vcn, bool doomedBad) {
  const std::string& chain = c->chain;
  int level = c->level;
  vcFreeCellNum_[vcn][chain][level]++;
  allVCFreeCellNum_[chain][level]++;
  totalLeftCellNum_[chain][level]++;
  int mergeLevelUpTo = addCellToFreeList(c);

  Cell* parent = c->parent;
  for (int l = level + 1; l <= mergeLevelUpTo; l++) {
    totalLeftCellNum_[chain][l]++;
    if (!static_cast<PhysicalCell*>(parent)->healthy) {
      badFreeCells_[chain].add(parent, l);
    } else {
      tryUnbindDoomedBadCell(chain, l);
    }
    parent = parent->parent;
  }
  if (!c->healthy) {
    releaseBadCell(c);
    if (!doomedBad) tryBindDoomedBadCell(chain, level);
  } else {
    tryUnbindDoomedBadCell(chain, level);
  }
  int numToAdd = static_cast<int>(c->children.size());
  for (int l = level - 1; l >= kLowestLevel; l--) {
    totalLeftCellNum_[chain][l] += numToAdd;
    if (!doomedBad) tryUnbindDoomedBadCell(chain, l);
    numToAdd *= static_cast<int>(fullCellList_[chain].at(l)[0]->children.size());
  }
}

// A bad free cell being allocated: bind each bad child to a virtual cell so
// the VC scheduler sees the failure.
void HivedCore::allocateBadCell(PhysicalCell* c) {
  if (badFreeCells_[c->chain].contains(c, c->level)) {
    badFreeCells_[c->chain].remove(c, c->level);
  }
  if (c->virt == nullptr) {
    VirtualCell* vc = getUnboundVirtualCell(static_cast<PhysicalCell*>(c->parent)->virt->children);
    c->virt = vc;
    vc->phys = c;
  }
  for (Cell* child : c->children) {
    auto* pc = static_cast<PhysicalCell*>(child);
    if (!pc->healthy) allocateBadCell(pc);
  }
}

void HivedCore::releaseBadCell(PhysicalCell* c) {
  badFreeCells_[c->chain].add(c, c->level);
  if (VirtualCell* vc = c->virt) {
    c->virt = nullptr;
    vc->phys = nullptr;
  }
  for (Cell* child : c->children) {
    auto* pc = static_cast<PhysicalCell*>(child);
    if (!pc->healthy) releaseBadCell(pc);
  }
}

// --- node / cell health ------------------------------------------------------

void HivedCore::setNodeHealthy(const std::string& node, bool healthy) {
  OpGuard opGuard(this);
  if (healthy) {
    if (!badNodes_.count(node)) return;
    badNodes_.erase(node);
    auto it = nodeLeafCellsStorage_.find(node);
    if (it == nodeLeafCellsStorage_.end()) return;
    for (PhysicalCell* leaf : it->second) {
      if (!badLeafMarks_.count(leaf)) setHealthyCell(leaf);
    }
  } else {
    if (badNodes_.count(node)) return;
    badNodes_.insert(node);
    auto it = nodeLeafCellsStorage_.find(node);
    if (it == nodeLeafCellsStorage_.end()) return;
    for (PhysicalCell* leaf : it->second) setBadCell(leaf);
  }
}

// Fine-grained health: one GPU (leaf cell) on a node. A degraded xGMI link is
// reported against its endpoint leaf; the pair/quad cell turns bad via the
// any-child-bad roll-up. Node-level badness (badNodes_) takes precedence: a
// leaf on a bad node cannot be marked healthy.
void HivedCore::setLeafCellHealthy(const std::string& node, int leafIndex, bool healthy) {
  OpGuard opGuard(this);
  auto it = nodeLeafCellsStorage_.find(node);
  if (it == nodeLeafCellsStorage_.end()) return;
  for (PhysicalCell* leaf : it->second) {
    if (!leaf->leafIndices.empty() && leaf->leafIndices[0] == leafIndex) {
      if (healthy) {
        badLeafMarks_.erase(leaf);
        if (!badNodes_.count(node)) setHealthyCell(leaf);
      } else {
        badLeafMarks_.insert(leaf);
        setBadCell(leaf);
      }
      return;
    }
  }
}

// First-class xGMI link health (BASELINE north star). A degraded link is
// recorded on the LINK, not the endpoint leaves: badLinksUnder is rolled up
// from the endpoints' LCA so pair/quad/node cells know a degraded link lies
// under them, while both endpoint GPUs stay schedulable for 1-GPU work and
// for placements that do not co-place the two endpoints. This extends the
// reference's leaf-only healthiness seam (cell.go:302-312) with a per-link
// dimension the reference cannot express.
void HivedCore::setXgmiLinkHealthy(const std::string& node, int a, int b, bool healthy,
                                   double gbps) {
  if (a == b) throw HivedError::BadRequest("xGMI link endpoints must differ: " + node);
  if (a > b) std::swap(a, b);
  auto it = nodeLeafCellsStorage_.find(node);
  if (it == nodeLeafCellsStorage_.end()) return;
  PhysicalCell* la = nullptr;
  PhysicalCell* lb = nullptr;
  for (PhysicalCell* leaf : it->second) {
    if (leaf->leafIndices.empty()) continue;
    if (leaf->leafIndices[0] == a) la = leaf;
    if (leaf->leafIndices[0] == b) lb = leaf;
  }
  if (la == nullptr || lb == nullptr || la->chain != lb->chain) return;
  XgmiLink& rec = xgmiLinks_[node][{a, b}];
  bool wasHealthy = rec.a == nullptr ? true : rec.healthy;
  rec.a = la;
  rec.b = lb;
  if (gbps > 0) rec.gbps = gbps;
  rec.healthy = healthy;
  if (healthy == wasHealthy) return;
  gWorldEpochCounter++;
  if (healthy) {
    auto erasePeer = [](PhysicalCell* from, PhysicalCell* peer) {
      auto& v = from->badLinkPeers;
      v.erase(std::remove(v.begin(), v.end(), peer), v.end());
    };
    erasePeer(la, lb);
    erasePeer(lb, la);
  } else {
    la->badLinkPeers.push_back(lb);
    lb->badLinkPeers.push_back(la);
  }
  // roll the count up from the endpoints' LCA (both are leaves: walk in step)
  Cell* x = la;
  Cell* y = lb;
  while (x != y && x != nullptr && y != nullptr) {
    x = x->parent;
    y = y->parent;
  }
  int delta = healthy ? -1 : 1;
  for (Cell* c = x; c != nullptr; c = c->parent) {
    static_cast<PhysicalCell*>(c)->badLinksUnder += delta;
  }
}

std::vector<std::tuple<int, int, double, bool>> HivedCore::xgmiLinks(
    const std::string& node) const {
  std::vector<std::tuple<int, int, double, bool>> out;
  auto it = xgmiLinks_.find(node);
  if (it == xgmiLinks_.end()) return out;
  for (auto& [key, l] : it->second) out.emplace_back(key.first, key.second, l.gbps, l.healthy);
  return out;
}

// A cell is bad if ANY child is bad; propagate from leaf up.
void HivedCore::setBadCell(PhysicalCell* c) {
  if (!c->healthy) return;
  gWorldEpochCounter++;
  c->healthy = false;
  if (c->parent != nullptr) setBadCell(static_cast<PhysicalCell*>(c->parent));
  if (inFreeCellList(c)) {
    addBadFreeCell(c);
  } else if (c->virt == nullptr && !c->split && c->parent != nullptr &&
             static_cast<PhysicalCell*>(c->parent)->virt != nullptr) {
    // the parent is bound to a virtual cell: bind this bad cell to one of
    // its unbound virtual children too, so the VC scheduler can see the
    // failure. The parent may be UNBOUND (a deeper ancestor holds the
    // binding) or every virtual child slot may already be taken — in either
    // case skip: the exposure binding is best-effort visibility, and
    // dereferencing a missing slot segfaulted (fuzz seed 1431).
    VirtualCell* vc =
        getUnboundVirtualCell(static_cast<PhysicalCell*>(c->parent)->virt->children);
    if (vc != nullptr) {
      c->virt = vc;
      vc->phys = c;
    }
  }
}

// A cell is healthy if ALL children are healthy; propagate from leaf up.
void HivedCore::setHealthyCell(PhysicalCell* c) {
  if (c->healthy) return;
  gWorldEpochCounter++;
  c->healthy = true;
  if (inFreeCellList(c)) {
    removeBadFreeCell(c);
  } else if (VirtualCell* vc = c->virt) {
    if (!c->pinned && c->priority < kMinGuaranteedPriority) {
      // binding existed only because the cell was bad; undo it
      c->virt = nullptr;
      vc->phys = nullptr;
      if (vc->parent == nullptr) {
        auto& doomed = vcDoomedBadCells_[vc->vc][c->chain];
        if (c->level <= doomed.top() && doomed.contains(c, c->level)) {
          // a doomed bad cell healing: undo the doom's accounting
          doomed.remove(c, c->level);
          allVCDoomedBadCellNum_[c->chain][c->level]--;
          releasePreassignedCell(c, vc->vc, true);
        }
        // else: an allocated-bad preassigned whose pods already left — its
        // accounting was released at pod-delete time while the binding was
        // kept because the cell was bad (fuzz-found: assuming "doomed" here
        // threw on the registry remove and double-released the accounting)
      }
    } else if (!c->pinned && vc->parent == nullptr &&
               vcDoomedBadCells_[vc->vc][c->chain].contains(c, c->level)) {
      // a doomed bad cell now healthy AND in real use: it is no longer
      // "doomed" — it transitions to a normally-allocated preassigned cell
      // (keeps its binding; releaseLeafCell will release it when free)
      vcDoomedBadCells_[vc->vc][c->chain].remove(c, c->level);
      allVCDoomedBadCellNum_[c->chain][c->level]--;
    }
  }
  if (c->parent == nullptr) return;
  for (Cell* buddy : c->parent->children) {
    if (!static_cast<PhysicalCell*>(buddy)->healthy) return;
  }
  setHealthyCell(static_cast<PhysicalCell*>(c->parent));
}

void HivedCore::addBadFreeCell(PhysicalCell* c) {
  const std::string& chain = c->chain;
  int level = c->level;
  badFreeCells_[chain].add(c, level);
  if (allVCFreeCellNum_[chain][level] >
      totalLeftCellNum_[chain][level] - static_cast<int>(badFreeCells_[chain].at(level).size())) {
    tryBindDoomedBadCell(chain, level);
  }
}

void HivedCore::removeBadFreeCell(PhysicalCell* c) {
  badFreeCells_[c->chain].remove(c, c->level);
  tryUnbindDoomedBadCell(c->chain, c->level);
}

// Dooming allocates the bad physical cell out of the free list. badFreeCells_
// contains SUBSUMED cells (free via an unsplit ancestor), and allocating one
// of those splits every unsplit ancestor — removing a cell from totalLeft at
// each split level — and removes the candidate's producible children below.
// Found by fuzzing: dooming a bad pair under a free quad stranded another
// VC's quad-level guarantee (totalLeft < allVCFree). A candidate is only
// eligible if no level's guarantee would break.
bool HivedCore::doomAllocationIsSafe(PhysicalCell* pc) {
  const std::string& chain = pc->chain;
  PhysicalCell* a = pc;
  while (a->parent != nullptr && !static_cast<PhysicalCell*>(a->parent)->split) {
    a = static_cast<PhysicalCell*>(a->parent);
    if (totalLeftCellNum_[chain][a->level] - 1 < allVCFreeCellNum_[chain][a->level]) return false;
  }
  int numToReduce = static_cast<int>(pc->children.size());
  for (int l = pc->level - 1; l >= kLowestLevel && numToReduce > 0; l--) {
    if (totalLeftCellNum_[chain][l] - numToReduce < allVCFreeCellNum_[chain][l]) return false;
    numToReduce *= static_cast<int>(fullCellList_[chain].at(l)[0]->children.size());
  }
  return true;
}

void HivedCore::tryBindDoomedBadCell(const std::string& chain, int level) {
  if (inOperation_) {
    pendingDoomChecks_.emplace_back(chain, level);
    return;
  }
  doBindDoomedBadCell(chain, level);
}

void HivedCore::tryUnbindDoomedBadCell(const std::string& chain, int level) {
  if (inOperation_) {
    pendingDoomChecks_.emplace_back(chain, level);
    return;
  }
  doUnbindDoomedBadCell(chain, level);
}

void HivedCore::drainDoomChecks() {
  int guard = 0;
  while (!pendingDoomChecks_.empty()) {
    if (++guard > 100000) throw HivedError::Internal("doom-check drain did not converge");
    auto [chain, level] = pendingDoomChecks_.front();
    pendingDoomChecks_.erase(pendingDoomChecks_.begin());
    // execute with the operation flag set so nested triggers re-queue
    inOperation_ = true;
    try {
      doUnbindDoomedBadCell(chain, level);
      doBindDoomedBadCell(chain, level);
    } catch (...) {
      inOperation_ = false;
      throw;
    }
    inOperation_ = false;
  }
}

// If healthy free cells < a VC's free cells at some level, some of the VC's
// cells are inevitably bad: bind bad physical cells to free virtual cells so
// users and the intra-VC scheduler can see them.
void HivedCore::doBindDoomedBadCell(const std::string& chain, int level) {
  for (auto& [vcName, vcFreeNum] : vcFreeCellNum_) {
    auto chainIt = vcFreeNum.find(chain);
    if (chainIt == vcFreeNum.end()) continue;
    while (chainIt->second[level] >
           totalLeftCellNum_[chain][level] -
               static_cast<int>(badFreeCells_[chain].at(level).size())) {
      if (badFreeCells_[chain].at(level).empty()) return;
      PhysicalCell* pc = nullptr;
      for (Cell* cand : badFreeCells_[chain].at(level)) {
        // Only cells DIRECTLY in the free list at this level are eligible:
        // a subsumed candidate (free via an unsplit ancestor) would require
        // splitting that ancestor, and tryBindDoomedBadCell can fire
        // re-entrantly inside another allocation's free-list surgery — the
        // split would mutate levels the outer operation is mid-way through
        // (fuzz-found crash: "cell not found in list when removing").
        // The ancestor itself is bad and listed at ITS level, so node-level
        // dooming still covers the VC's higher-level quota.
        if (!freeCellList_[chain].at(level).empty() &&
            freeCellList_[chain].contains(cand, level) &&
            doomAllocationIsSafe(static_cast<PhysicalCell*>(cand))) {
          pc = static_cast<PhysicalCell*>(cand);
          break;
        }
      }
      if (pc == nullptr) return;  // no directly-free, safety-preserving candidate
      if (mapDebug())
        fprintf(stderr, "[doom] bind %s level=%d for vc=%s\n", pc->address.c_str(), level,
                vcName.c_str());
      auto& pre = vcSchedulers_[vcName].nonPinnedPreassigned;
      auto preIt = pre.find(chain);
      VirtualCell* vc = nullptr;
      if (preIt != pre.end() && level <= preIt->second.top()) {
        vc = getUnboundVirtualCell(preIt->second.at(level));
      }
      if (vc == nullptr) return;  // no more free preassigned cells to doom
      pc->virt = vc;
      vc->phys = pc;
      vcDoomedBadCells_[vcName][chain].add(pc, level);
      allVCDoomedBadCellNum_[chain][level]++;
      allocatePreassignedCell(pc, vcName, true);
    }
  }
}

void HivedCore::doUnbindDoomedBadCell(const std::string& chain, int level) {
  for (auto& [vcName, vcFreeNum] : vcFreeCellNum_) {
    auto chainIt = vcFreeNum.find(chain);
    if (chainIt == vcFreeNum.end()) continue;
    auto& doomed = vcDoomedBadCells_[vcName][chain];
    if (level > doomed.top()) continue;
    while (chainIt->second[level] <
           totalLeftCellNum_[chain][level] -
               static_cast<int>(badFreeCells_[chain].at(level).size())) {
      // release only doomed cells whose binding exists purely because they
      // are bad (not in real use by a group scheduled onto healthy leaves)
      PhysicalCell* pc = nullptr;
      for (Cell* cand : doomed.at(level)) {
        if (cand->priority < kMinGuaranteedPriority) {
          pc = static_cast<PhysicalCell*>(cand);
          break;
        }
      }
      if (pc == nullptr) break;
      if (mapDebug())
        fprintf(stderr, "[doom] unbind %s level=%d vc=%s\n", pc->address.c_str(), level,
                vcName.c_str());
      pc->virt->phys = nullptr;
      pc->virt = nullptr;
      doomed.remove(pc, level);
      allVCDoomedBadCellNum_[chain][level]--;
      releasePreassignedCell(pc, vcName, true);
    }
  }
}

}  // namespace hived
