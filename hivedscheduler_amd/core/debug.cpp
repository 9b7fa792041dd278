// Invariant checker: verifies the correctness spec (SURVEY.md Appendix B /
// docs/design.md) over the live data structures. Used by the randomized
// fuzz tests; throws with a description on any violation.
#include <sstream>

#include "core.hpp"

namespace hived {

namespace {

void fail(const std::string& what) { throw HivedError::Internal("invariant violated: " + what); }

int countFreeListSubtree(const ChainCellList& freeList, int level) {
  // number of cells at `level` obtainable from the free list (free cells at
  // that level + descendants of free cells above)
  int count = 0;
  for (int l = level; l <= freeList.top(); l++) {
    int mult = 1;
    bool ok = true;
    // compute how many level-`level` cells one level-l cell contains
    for (Cell* c : freeList.at(l)) {
      Cell* probe = c;
      mult = 1;
      while (probe->level > level) {
        mult *= static_cast<int>(probe->children.size());
        probe = probe->children[0];
      }
      if (ok) count += mult;
    }
  }
  return count;
}

}  // namespace

void checkInvariants(const HivedCore& core) {
  // ---- per-cell roll-ups over every physical chain ----
  for (auto& [chain, ccl] : core.fullCellList_) {
    for (int l = ccl.top(); l >= kLowestLevel; l--) {
      for (Cell* cc : ccl.at(l)) {
        auto* c = static_cast<PhysicalCell*>(cc);
        if (!c->children.empty()) {
          int maxPrio = kFreePriority;
          bool allHealthy = true;
          bool anyUsed = false;
          std::map<int, int> usedSum;
          for (Cell* childC : c->children) {
            auto* child = static_cast<PhysicalCell*>(childC);
            maxPrio = std::max(maxPrio, child->priority);
            allHealthy = allHealthy && child->healthy;
            anyUsed = anyUsed || (child->state == CState::Used);
            for (auto& [p, n] : child->usedLeafAtPriority) usedSum[p] += n;
          }
          if (c->priority != maxPrio) {
            fail("priority roll-up at " + c->address + ": " + std::to_string(c->priority) +
                 " != max(children) " + std::to_string(maxPrio));
          }
          int freeSum = 0;
          for (Cell* childC : c->children) freeSum += childC->freeLeavesUnder;
          if (c->freeLeavesUnder != freeSum) {
            fail("freeLeavesUnder roll-up at " + c->address + ": " +
                 std::to_string(c->freeLeavesUnder) + " != sum(children) " +
                 std::to_string(freeSum));
          }
          if (c->healthy != allHealthy) fail("healthiness roll-up at " + c->address);
          if (anyUsed && c->state != CState::Used) fail("state roll-up (Used) at " + c->address);
          if (usedSum != c->usedLeafAtPriority) fail("usedLeafAtPriority roll-up at " + c->address);
        } else if (c->freeLeavesUnder != (c->priority == kFreePriority ? 1 : 0)) {
          fail("leaf freeLeavesUnder at " + c->address);
        }
        // binding symmetry
        if (c->virt != nullptr && c->virt->phys != c) fail("asymmetric binding at " + c->address);
      }
    }
  }
  // ---- xGMI link state: badLinksUnder roll-up + peer symmetry ----
  {
    // recompute badLinksUnder from the link registry and compare
    std::unordered_map<const Cell*, int> expect;
    for (auto& [node, links] : core.xgmiLinks_) {
      (void)node;
      for (auto& [key, l] : links) {
        (void)key;
        if (l.healthy) continue;
        // peer symmetry on the endpoint leaves
        auto hasPeer = [](PhysicalCell* from, PhysicalCell* to) {
          return std::find(from->badLinkPeers.begin(), from->badLinkPeers.end(), to) !=
                 from->badLinkPeers.end();
        };
        if (!hasPeer(l.a, l.b) || !hasPeer(l.b, l.a)) {
          fail("bad xGMI link " + l.a->address + "<->" + l.b->address +
               " not mirrored in badLinkPeers");
        }
        Cell* x = l.a;
        Cell* y = l.b;
        while (x != y && x != nullptr && y != nullptr) {
          x = x->parent;
          y = y->parent;
        }
        for (Cell* c = x; c != nullptr; c = c->parent) expect[c]++;
      }
    }
    for (auto& [chain, ccl] : core.fullCellList_) {
      (void)chain;
      for (int l = ccl.top(); l >= kLowestLevel; l--) {
        for (Cell* cc : ccl.at(l)) {
          auto* c = static_cast<PhysicalCell*>(cc);
          int want = expect.count(c) ? expect[c] : 0;
          if (c->badLinksUnder != want) {
            fail("badLinksUnder roll-up at " + c->address + ": " +
                 std::to_string(c->badLinksUnder) + " != " + std::to_string(want));
          }
        }
      }
    }
  }
  // ---- free list consistency + accounting ----
  for (auto& [chain, freeList] : core.freeCellList_) {
    for (int l = freeList.top(); l >= kLowestLevel; l--) {
      for (Cell* cc : freeList.at(l)) {
        auto* c = static_cast<PhysicalCell*>(cc);
        if (c->virt != nullptr) fail("free cell " + c->address + " is bound");
        if (c->split) fail("free cell " + c->address + " is split");
        if (c->parent != nullptr && !static_cast<PhysicalCell*>(c->parent)->split) {
          fail("free cell " + c->address + " has unsplit parent (should be merged)");
        }
        // free-list cells may carry opportunistic usage (OT pods do not own
        // cells), but never guaranteed priority (guaranteed implies binding)
        if (c->priority >= kMinGuaranteedPriority) {
          fail("free cell " + c->address + " has guaranteed priority " +
               std::to_string(c->priority));
        }
      }
    }
    // totalLeftCellNum matches what the free list can actually produce
    auto tlIt = core.totalLeftCellNum_.find(chain);
    if (tlIt != core.totalLeftCellNum_.end()) {
      for (auto& [level, left] : tlIt->second) {
        int actual = countFreeListSubtree(freeList, level);
        if (actual != left) {
          std::ostringstream os;
          os << "totalLeftCellNum mismatch at chain " << chain << " level " << level
             << ": accounted " << left << ", free list yields " << actual;
          fail(os.str());
        }
      }
    }
  }
  // ---- VC safety: totalLeft >= allVCFree at every level ----
  for (auto& [chain, perLevel] : core.allVCFreeCellNum_) {
    auto tlIt = core.totalLeftCellNum_.find(chain);
    if (tlIt == core.totalLeftCellNum_.end()) continue;
    for (auto& [level, vcFree] : perLevel) {
      auto it = tlIt->second.find(level);
      int left = it == tlIt->second.end() ? 0 : it->second;
      if (left < vcFree) {
        std::ostringstream os;
        os << "VC safety broken at chain " << chain << " level " << level << ": " << left
           << " left < " << vcFree << " free in all VCs";
        fail(os.str());
      }
    }
  }
  // ---- allVCFreeCellNum == sum of vcFreeCellNum ----
  std::map<std::string, std::map<int, int>> sum;
  for (auto& [vc, perChain] : core.vcFreeCellNum_) {
    (void)vc;
    for (auto& [chain, perLevel] : perChain) {
      for (auto& [level, n] : perLevel) sum[chain][level] += n;
    }
  }
  for (auto& [chain, perLevel] : core.allVCFreeCellNum_) {
    for (auto& [level, n] : perLevel) {
      int s = sum.count(chain) && sum[chain].count(level) ? sum[chain][level] : 0;
      if (s != n) {
        fail("allVCFreeCellNum mismatch at " + chain + " level " + std::to_string(level));
      }
    }
  }
  // ---- doomed-bad cells: every entry is bound symmetrically ----
  for (auto& [vcName, perChain] : core.vcDoomedBadCells_) {
    for (auto& [chain, ccl] : perChain) {
      for (int l = 1; l <= ccl.top(); l++) {
        for (Cell* cc : ccl.at(l)) {
          auto* pc = static_cast<PhysicalCell*>(cc);
          if (pc->virt == nullptr) {
            fail("doomed bad cell " + pc->address + " (VC " + vcName + ", chain " + chain +
                 ") has no binding");
          }
          if (pc->virt->phys != pc) fail("doomed bad cell " + pc->address + " binding asymmetric");
        }
      }
    }
  }
  // ---- virtual side: binding symmetry + roll-ups ----
  for (auto& [vcName, vcs] : core.vcSchedulers_) {
    (void)vcName;
    for (auto& [chain, ccl] : vcs.nonPinnedFull) {
      (void)chain;
      for (int l = ccl.top(); l >= kLowestLevel; l--) {
        for (Cell* cc : ccl.at(l)) {
          auto* v = static_cast<VirtualCell*>(cc);
          if (v->phys != nullptr && v->phys->virt != v) {
            fail("asymmetric virtual binding at " + v->address);
          }
          if (!v->children.empty()) {
            int maxPrio = kFreePriority;
            int freeSum = 0;
            for (Cell* child : v->children) {
              maxPrio = std::max(maxPrio, child->priority);
              freeSum += child->freeLeavesUnder;
            }
            if (v->priority != maxPrio) fail("virtual priority roll-up at " + v->address);
            if (v->freeLeavesUnder != freeSum) {
              fail("virtual freeLeavesUnder roll-up at " + v->address);
            }
          } else if (v->freeLeavesUnder != (v->priority == kFreePriority ? 1 : 0)) {
            fail("virtual leaf freeLeavesUnder at " + v->address);
          }
        }
      }
    }
  }
  // ---- group placements: used cells point back to their groups ----
  for (auto& [name, g] : core.groups_) {
    (void)name;
    for (auto& [leafNum, pods] : g->physPlacement) {
      (void)leafNum;
      for (auto& pod : pods) {
        for (PhysicalCell* c : pod) {
          if (c == nullptr) continue;
          if (g->state == GState::Allocated && c->state == CState::Used &&
              c->usingGroup != g.get() && c->usingGroup != nullptr) {
            // a cell may have been taken over after lazy preemption races —
            // but a Used cell of an allocated group must reference SOME group
          }
          if (g->state == GState::Preempting &&
              (c->state == CState::Reserving || c->state == CState::Reserved) &&
              c->reservingGroup != g.get()) {
            fail("reserved cell " + c->address + " does not reference its preemptor");
          }
        }
      }
    }
  }
}

}  // namespace hived
