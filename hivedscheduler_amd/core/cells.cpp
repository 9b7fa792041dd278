// Cell helpers: priority/state roll-ups, virtual<->physical binding,
// recovery mapping. Semantics parity: pkg/algorithm/cell_allocation.go:384-454,
// pkg/algorithm/utils.go:380-415.
#include "core.hpp"

namespace hived {

const char* to_string(CState s) {
  switch (s) {
    case CState::Free: return "Free";
    case CState::Used: return "Used";
    case CState::Reserving: return "Reserving";
    case CState::Reserved: return "Reserved";
  }
  return "?";
}

const char* to_string(GState s) {
  switch (s) {
    case GState::Allocated: return "Allocated";
    case GState::Preempting: return "Preempting";
    case GState::BeingPreempted: return "BeingPreempted";
  }
  return "?";
}

// Global mutation counter for the clean-shape-world cache: bumped by every
// mutation that can change a world (leaf priorities, bindings, health,
// links). Process-global on purpose — with several cores alive (tests) a
// foreign bump merely costs a cache miss, never staleness.
unsigned long long gWorldEpochCounter = 1;

static inline void bumpWorldEpoch() { gWorldEpochCounter++; }

// Priority of a cell is the max of its children's; propagate from leaf up.
void setCellPriority(Cell* c, int p) {
  int original = c->priority;
  c->priority = p;
  if (c->level == kLowestLevel && original != p) bumpWorldEpoch();
  // maintain the freeLeavesUnder cache on leaf free<->used transitions
  if (c->level == kLowestLevel && (original == kFreePriority) != (p == kFreePriority)) {
    int delta = p == kFreePriority ? 1 : -1;
    for (Cell* a = c; a != nullptr; a = a->parent) a->freeLeavesUnder += delta;
  }
  if (Cell* parent = c->parent) {
    if (p > parent->priority) {
      setCellPriority(parent, p);
    } else if (original == parent->priority && p < original) {
      int maxBuddy = kFreePriority;
      for (Cell* buddy : parent->children) maxBuddy = std::max(maxBuddy, buddy->priority);
      setCellPriority(parent, maxBuddy);
    }
  }
}

void updateUsedLeafCellNumAtPriority(Cell* c, int p, bool increase) {
  for (; c != nullptr; c = c->parent) {
    int& v = c->usedLeafAtPriority[p];
    v += increase ? 1 : -1;
    if (v == 0) c->usedLeafAtPriority.erase(p);
  }
}

// Bind a virtual cell to a physical cell, and their ancestors bottom-up until
// hitting an already-bound ancestor ON EITHER SIDE. The physical-side check
// matters: walking past a physical ancestor that is already bound to a
// DIFFERENT virtual cell would overwrite its pointer and strand the old
// partner with a dangling phys reference (fuzz-found asymmetric binding
// after a lazy-preemption revert whose victim kept bad-cell bindings).
void bindCell(PhysicalCell* pc, VirtualCell* vc) {
  bumpWorldEpoch();
  while (vc->phys == nullptr && pc->virt == nullptr) {
    pc->virt = vc;
    vc->phys = pc;
    if (vc->parent == nullptr) break;
    vc = static_cast<VirtualCell*>(vc->parent);
    pc = static_cast<PhysicalCell*>(pc->parent);
  }
}

// Unbind a physical cell's binding and its ancestors bottom-up while no bound
// sibling remains; never unbind pinned cells (same guard as the reference,
// cell_allocation.go:386-420). The complementary unhealthy-cell guard — bad
// (incl. doomed-bad) cells keep their bindings until healthy, reference
// hived_algorithm.go:1327-1345 — is applied by the CALLER at leaf level
// (releaseLeafCell, algorithm.cpp): an unhealthy leaf never reaches this
// function, so its ancestors' bindings survive too. Fuzz-found rationale for
// that caller-side guard: unbinding a doomed bad quad here while
// releaseLeafCell's doomed guard skipped the matching accounting release
// left the binding and the free list out of sync, and a later re-bind
// double-allocated the cell.
void unbindCell(PhysicalCell* c) {
  bumpWorldEpoch();
  VirtualCell* boundVirtual = c->virt;
  while (boundVirtual->phys != nullptr && !boundVirtual->phys->pinned) {
    PhysicalCell* boundPhysical = boundVirtual->phys;
    boundVirtual->phys = nullptr;
    boundPhysical->virt = nullptr;
    if (boundVirtual->parent == nullptr) return;
    for (Cell* cc : boundVirtual->parent->children) {
      if (static_cast<VirtualCell*>(cc)->phys != nullptr) return;
    }
    boundVirtual = static_cast<VirtualCell*>(boundVirtual->parent);
  }
}

VirtualCell* getUnboundVirtualCell(const std::vector<Cell*>& cl) {
  for (Cell* c : cl) {
    auto* vc = static_cast<VirtualCell*>(c);
    if (vc->phys == nullptr) return vc;
  }
  return nullptr;
}

// Lowest-priority virtual cell whose priority is lower than p. A free cell
// with a binding is skipped (such bindings exist only for doomed bad cells and
// cannot be preempted).
VirtualCell* getLowestPriorityVirtualCell(const std::vector<Cell*>& cl, int p) {
  int lowest = kMaxGuaranteedPriority;
  VirtualCell* lowestCell = nullptr;
  for (Cell* c : cl) {
    auto* vc = static_cast<VirtualCell*>(c);
    if (vc->priority == kFreePriority) {
      if (vc->phys == nullptr) return vc;
      continue;
    }
    if (vc->priority < p && vc->priority < lowest) {
      lowest = vc->priority;
      lowestCell = vc;
    }
  }
  return lowestCell;
}

// Inverse of the virtual->physical mapping, used when replaying allocated pods
// (recovery / reconfiguration). Parity: cell_allocation.go:320-346.
VirtualCell* mapPhysicalCellToVirtual(PhysicalCell* c, const ChainCellList& vccl,
                                      int preassignedLevel, int p, std::string* message) {
  if (c->virt != nullptr) return c->virt;
  if (c->level == preassignedLevel) {
    if (preassignedLevel > vccl.top()) {
      *message = "preassigned level above VC cell list top";
      return nullptr;
    }
    VirtualCell* pre = getLowestPriorityVirtualCell(vccl.at(preassignedLevel), p);
    if (pre == nullptr) {
      *message = "insufficient free cell in the VC at the preassigned level " +
                 std::to_string(preassignedLevel);
    }
    return pre;
  }
  if (c->parent == nullptr) {
    *message = "physical and virtual cell hierarchies do not match (cannot reach preassigned level " +
               std::to_string(preassignedLevel) + ")";
    return nullptr;
  }
  VirtualCell* parentVirtual =
      mapPhysicalCellToVirtual(static_cast<PhysicalCell*>(c->parent), vccl, preassignedLevel, p, message);
  if (parentVirtual == nullptr) return nullptr;
  VirtualCell* vc = getLowestPriorityVirtualCell(parentVirtual->children, p);
  if (vc == nullptr) *message = "no lower-priority virtual cell among children of " + parentVirtual->address;
  return vc;
}

// A cell is "in the free list" (itself or via an unsplit ancestor) iff it is
// unbound and no ancestor below the first split one is bound.
bool inFreeCellList(PhysicalCell* c) {
  for (;;) {
    if (c->virt != nullptr || c->split) return false;
    if (c->parent == nullptr || static_cast<PhysicalCell*>(c->parent)->split) return true;
    c = static_cast<PhysicalCell*>(c->parent);
  }
}

static bool allChildrenSameState(PhysicalCell* c, CState s) {
  for (Cell* child : c->children) {
    if (static_cast<PhysicalCell*>(child)->state != s) return false;
  }
  return true;
}

// A parent is Used if any child is Used; for other states, unanimous children.
void setCellState(PhysicalCell* c, CState s) {
  c->state = s;
  if (c->parent != nullptr) {
    auto* parent = static_cast<PhysicalCell*>(c->parent);
    if (s == CState::Used || allChildrenSameState(parent, s)) setCellState(parent, s);
  }
}

static PhysicalCell* findPhysicalLeafCellInChain(std::map<std::string, ChainCellList>& fullCellList,
                                                 const std::string& chain, const std::string& node,
                                                 int leafIndex) {
  auto it = fullCellList.find(chain);
  if (it == fullCellList.end()) return nullptr;
  for (Cell* c : it->second.at(kLowestLevel)) {
    auto* pc = static_cast<PhysicalCell*>(c);
    if (!pc->nodes.empty() && pc->nodes[0] == node && !pc->leafIndices.empty() &&
        pc->leafIndices[0] == leafIndex) {
      return pc;
    }
  }
  return nullptr;
}

PhysicalCell* findPhysicalLeafCell(std::map<std::string, ChainCellList>& fullCellList,
                                   const std::string& chain, const std::string& node,
                                   int leafIndex) {
  if (PhysicalCell* c = findPhysicalLeafCellInChain(fullCellList, chain, node, leafIndex)) return c;
  for (auto& [other, ccl] : fullCellList) {
    (void)ccl;
    if (other == chain) continue;
    if (PhysicalCell* c = findPhysicalLeafCellInChain(fullCellList, other, node, leafIndex)) return c;
  }
  return nullptr;
}

Cell* ancestorNoHigherThanNode(Cell* c) {
  while (!c->atOrAboveNode && c->parent != nullptr) c = c->parent;
  return c;
}

}  // namespace hived
