// Cluster-spec compiler: normalized spec -> physical/virtual cell trees,
// free lists, VC accounting, schedulers. Semantics parity:
// pkg/algorithm/config.go (ParseConfig) + hived_algorithm.go:108-464 (init).
#include "core.hpp"

namespace hived {

namespace {

struct ChainMeta {
  std::vector<std::string> types;  // top-down type names, types.back() = leaf
  std::vector<bool> isNode;        // aligned with types
  int topLevel() const { return static_cast<int>(types.size()); }
  // level l (1=leaf) -> index into types: types[topLevel - l]
  const std::string& typeAt(int l) const { return types[topLevel() - l]; }
  bool isNodeAt(int l) const { return isNode[topLevel() - l]; }
  // a level is at-or-above node iff the chain's node level is at or below it
  bool atOrAboveNodeAt(int l) const {
    for (int ll = l; ll >= 1; ll--) {
      if (isNodeAt(ll)) return true;
    }
    return false;
  }
};

ChainMeta buildChainMeta(const std::map<std::string, CellTypeSpec>& cellTypes,
                         const std::string& topType) {
  ChainMeta m;
  std::string t = topType;
  std::set<std::string> seen;
  for (;;) {
    if (seen.count(t)) throw HivedError::BadRequest("cellTypes contains a cycle at " + t);
    seen.insert(t);
    auto it = cellTypes.find(t);
    bool leaf = (it == cellTypes.end()) || it->second.child.empty();
    m.types.push_back(t);
    m.isNode.push_back(it != cellTypes.end() && it->second.isNode);
    if (leaf) break;
    t = it->second.child;
  }
  return m;
}

int childCountAt(const std::map<std::string, CellTypeSpec>& cellTypes, const std::string& type) {
  auto it = cellTypes.find(type);
  if (it == cellTypes.end() || it->second.child.empty()) return 0;
  return it->second.childCount;
}

}  // namespace

struct BuildContext {
  HivedCore* h;
  const ClusterSpec& spec;
  std::map<std::string, ChainMeta> chainMeta;
  std::map<std::string, std::vector<PhysicalCell*>> nodeLeafCells;  // node -> leaf cells
  std::map<std::string, PhysicalCell*> pinnedCellsById;
  // (node, (a, b, gbps, healthy)) link specs deferred until leaf maps exist
  std::vector<std::pair<std::string, std::tuple<int, int, double, bool>>> pendingLinks;

  BuildContext(HivedCore* h_, const ClusterSpec& s) : h(h_), spec(s) {}

  PhysicalCell* buildPhysicalCell(const PhysCellSpec& cs, const ChainMeta& meta,
                                  const std::string& chain, int level,
                                  const std::string& parentAddr, const std::string& nodeName) {
    auto cell = std::make_unique<PhysicalCell>();
    PhysicalCell* c = cell.get();
    h->cellStore_.push_back(std::move(cell));
    c->chain = chain;
    c->level = level;
    c->typeName = meta.typeAt(level);
    c->isNodeLevel = meta.isNodeAt(level);
    c->atOrAboveNode = meta.atOrAboveNodeAt(level);
    std::string ownNodeName = nodeName;
    if (c->isNodeLevel) ownNodeName = cs.address;
    c->address = parentAddr.empty() ? cs.address : parentAddr + "/" + cs.address;
    if (!cs.pinnedId.empty()) {
      c->pinned = true;
      c->pinnedId = cs.pinnedId;
      if (!pinnedCellsById.emplace(cs.pinnedId, c).second) {
        throw HivedError::BadRequest("pinnedCellId " + cs.pinnedId + " used by multiple physical cells");
      }
    }
    // discovery-measured xGMI link table (node-level entries): applied after
    // construction, once nodeLeafCellsStorage_ is populated
    if (!cs.xgmiLinks.empty()) {
      for (auto& l : cs.xgmiLinks) pendingLinks.emplace_back(ownNodeName, l);
    }
    if (level == kLowestLevel) {
      if (ownNodeName.empty()) {
        // chain without a node level: the leaf's own address is the node name
        ownNodeName = c->address;
      }
      int leafIndex = 0;
      try {
        leafIndex = std::stoi(cs.address);
      } catch (...) {
        throw HivedError::BadRequest("leaf cell address must be an integer device index, got '" +
                                     cs.address + "'");
      }
      c->totalLeaf = 1;
      c->freeLeavesUnder = 1;  // all cells start Free
      c->nodes = {ownNodeName};
      c->leafIndices = {leafIndex};
      // measured per-GPU capacity from discovery when provided (a GPU
      // reporting less than the nominal 288 GB is a health signal and is
      // avoided by requests carrying hbmBytesPerCell), nominal otherwise
      c->hbmBytes = cs.hbmBytes > 0 ? cs.hbmBytes : 288LL * 1024 * 1024 * 1024;
      nodeLeafCells[ownNodeName].push_back(c);
    } else {
      int expected = childCountAt(spec.cellTypes, c->typeName);
      if (static_cast<int>(cs.children.size()) != expected) {
        throw HivedError::BadRequest("cell " + c->address + " of type " + c->typeName + " has " +
                                     std::to_string(cs.children.size()) + " children, expected " +
                                     std::to_string(expected));
      }
      for (const auto& childSpec : cs.children) {
        PhysicalCell* child =
            buildPhysicalCell(childSpec, meta, chain, level - 1, c->address, ownNodeName);
        child->parent = c;
        c->children.push_back(child);
        c->totalLeaf += child->totalLeaf;
        c->freeLeavesUnder += child->freeLeavesUnder;
        c->hbmBytes += child->hbmBytes;
        for (auto& n : child->nodes) {
          if (std::find(c->nodes.begin(), c->nodes.end(), n) == c->nodes.end()) c->nodes.push_back(n);
        }
        for (int gi : child->leafIndices) c->leafIndices.push_back(gi);
      }
    }
    h->fullCellList_[chain].add(c, level);
    return c;
  }

  VirtualCell* buildVirtualCell(const std::string& vc, const std::string& chain,
                                const ChainMeta& meta, int level, VirtualCell* preassigned,
                                const std::string& addr, const std::string& pinnedId,
                                ChainCellList& fullList) {
    auto cell = std::make_unique<VirtualCell>();
    VirtualCell* c = cell.get();
    h->cellStore_.push_back(std::move(cell));
    c->vc = vc;
    c->chain = chain;
    c->level = level;
    c->typeName = meta.typeAt(level);
    c->isNodeLevel = meta.isNodeAt(level);
    c->atOrAboveNode = meta.atOrAboveNodeAt(level);
    c->address = addr;
    c->pinnedId = pinnedId;
    c->preassigned = (preassigned == nullptr) ? c : preassigned;
    if (level == kLowestLevel) {
      c->totalLeaf = 1;
      c->freeLeavesUnder = 1;  // all cells start Free
    } else {
      int childNum = childCountAt(spec.cellTypes, c->typeName);
      for (int i = 0; i < childNum; i++) {
        VirtualCell* child = buildVirtualCell(vc, chain, meta, level - 1, c->preassigned,
                                              addr + "/" + std::to_string(i), pinnedId, fullList);
        child->parent = c;
        c->children.push_back(child);
        c->totalLeaf += child->totalLeaf;
        c->freeLeavesUnder += child->freeLeavesUnder;
      }
    }
    fullList.add(c, level);
    return c;
  }

  void build() {
    // --- physical cluster ---
    for (const auto& cs : spec.physicalCells) {
      const std::string& chain = cs.type;
      if (!chainMeta.count(chain)) chainMeta.emplace(chain, buildChainMeta(spec.cellTypes, chain));
      const ChainMeta& meta = chainMeta.at(chain);
      if (!h->fullCellList_.count(chain)) {
        h->fullCellList_[chain].init(meta.topLevel());
        h->freeCellList_[chain].init(meta.topLevel());
        for (int l = 1; l <= meta.topLevel(); l++) {
          h->cellTypes_[chain][l] = meta.typeAt(l);
        }
      }
      PhysicalCell* top = buildPhysicalCell(cs, meta, chain, meta.topLevel(), "", "");
      h->freeCellList_[chain].add(top, meta.topLevel());
    }
    for (auto& [chain, meta] : chainMeta) {
      if (!h->fullCellList_.count(chain)) continue;
      h->cellChains_[meta.types.back()].push_back(chain);
      int leafNum = 1;
      for (int l = 1; l <= meta.topLevel(); l++) {
        h->leafCellNums_[chain][l] = leafNum;
        if (l < meta.topLevel()) {
          leafNum *= childCountAt(spec.cellTypes, meta.typeAt(l + 1));
        }
      }
    }

    // --- virtual clusters ---
    for (const auto& [vcName, vcSpec] : spec.virtualClusters) {
      IntraVCScheduler& vcs = h->vcSchedulers_[vcName];
      int cellCounter = 0;
      for (const auto& vcell : vcSpec.virtualCells) {
        // resolve hierarchical type path "CHAIN.CHILD..." -> (chain, level)
        std::vector<std::string> parts;
        size_t start = 0;
        const std::string& path = vcell.typePath;
        while (start <= path.size()) {
          size_t dot = path.find('.', start);
          if (dot == std::string::npos) {
            parts.push_back(path.substr(start));
            break;
          }
          parts.push_back(path.substr(start, dot - start));
          start = dot + 1;
        }
        const std::string& chain = parts[0];
        auto metaIt = chainMeta.find(chain);
        if (metaIt == chainMeta.end() || !h->fullCellList_.count(chain)) {
          throw HivedError::BadRequest("VC " + vcName + " virtual cell type " + path +
                                       ": chain " + chain + " does not exist in physical cluster");
        }
        const ChainMeta& meta = metaIt->second;
        for (size_t i = 1; i < parts.size(); i++) {
          if (i >= meta.types.size() || meta.types[i] != parts[i]) {
            throw HivedError::BadRequest("VC " + vcName + " virtual cell type path " + path +
                                         " does not match chain " + chain);
          }
        }
        int level = meta.topLevel() - static_cast<int>(parts.size()) + 1;
        if (!vcs.nonPinnedFull.count(chain)) {
          vcs.nonPinnedFull[chain].init(meta.topLevel());
          vcs.nonPinnedPreassigned[chain].init(meta.topLevel());
        }
        for (int i = 0; i < vcell.number; i++) {
          std::string addr = vcName + "/" + chain + "/" + std::to_string(cellCounter++);
          VirtualCell* pre = buildVirtualCell(vcName, chain, meta, level, nullptr, addr, "",
                                              vcs.nonPinnedFull[chain]);
          vcs.nonPinnedPreassigned[chain].add(pre, level);
        }
        h->vcFreeCellNum_[vcName][chain][level] += vcell.number;
      }
      // pinned cells: statically bound, scheduled by their own scheduler
      for (const auto& pinnedId : vcSpec.pinnedIds) {
        auto it = pinnedCellsById.find(pinnedId);
        if (it == pinnedCellsById.end()) {
          throw HivedError::BadRequest("VC " + vcName + " refers to unknown pinnedCellId " + pinnedId);
        }
        PhysicalCell* pc = it->second;
        if (h->pinnedPhysical_.count(vcName) == 0) h->pinnedPhysical_[vcName] = {};
        for (auto& [otherVc, ids] : h->pinnedPhysical_) {
          if (ids.count(pinnedId)) {
            throw HivedError::BadRequest("pinnedCellId " + pinnedId + " referred by multiple VCs");
          }
        }
        const ChainMeta& meta = chainMeta.at(pc->chain);
        ChainCellList& pinnedList = vcs.pinned[pinnedId];
        pinnedList.init(pc->level);
        std::string addr = vcName + "/" + pinnedId;
        buildVirtualCell(vcName, pc->chain, meta, pc->level, nullptr, addr, pinnedId, pinnedList);
        h->pinnedPhysical_[vcName][pinnedId] = pc;
        h->vcFreeCellNum_[vcName][pc->chain][pc->level] += 1;
      }
      for (auto& [chain, ccl] : vcs.nonPinnedFull) {
        vcs.nonPinnedSchedulers.emplace(chain,
                                        TopoScheduler(ccl, h->leafCellNums_[chain], true));
      }
      for (auto& [pid, ccl] : vcs.pinned) {
        const std::string& chain = ccl.at(kLowestLevel)[0]->chain;
        vcs.pinnedSchedulers.emplace(pid, TopoScheduler(ccl, h->leafCellNums_[chain], true));
      }
    }
    for (auto& [chain, ccl] : h->fullCellList_) {
      h->opportunisticSchedulers_.emplace(chain, TopoScheduler(ccl, h->leafCellNums_[chain], false));
    }
  }
};

void HivedCore::buildFromSpec(const ClusterSpec& spec) {
  BuildContext ctx(this, spec);
  ctx.build();
  nodeLeafCellsStorage_.clear();
  for (auto& [node, cells] : ctx.nodeLeafCells) nodeLeafCellsStorage_[node] = cells;
  // apply the discovery-measured xGMI link table (records gbps; a link
  // reported unhealthy at config time degrades its pair/quad from the start)
  for (auto& [node, l] : ctx.pendingLinks) {
    setXgmiLinkHealthy(node, std::get<0>(l), std::get<1>(l), std::get<3>(l), std::get<2>(l));
  }
}

// Validates VC quota against the physical cluster and initializes the
// accounting maps (parity: hived_algorithm.go:369-409).
void HivedCore::initCellNums() {
  // Every physical chain gets accounting structures, even with no VC quota.
  for (auto& [chain, ccl] : fullCellList_) {
    int top = ccl.top();
    badFreeCells_[chain].init(top);
    totalLeftCellNum_[chain][top] = static_cast<int>(ccl.at(top).size());
    for (int l = top; l >= kLowestLevel; l--) {
      allVCDoomedBadCellNum_[chain][l] = 0;
      allVCFreeCellNum_[chain][l] += 0;
      if (l > kLowestLevel) {
        int childNum = static_cast<int>(ccl.at(l)[0]->children.size());
        totalLeftCellNum_[chain][l - 1] = totalLeftCellNum_[chain][l] * childNum;
      }
    }
  }
  for (auto& [vc, perChain] : vcFreeCellNum_) {
    vcDoomedBadCells_[vc] = {};
    for (auto& [chain, perLevel] : perChain) {
      if (!fullCellList_.count(chain)) {
        throw HivedError::BadRequest("Illegal initial VC assignment: chain " + chain +
                                     " does not exist in physical cluster");
      }
      vcDoomedBadCells_[vc][chain].init(fullCellList_[chain].top());
      for (auto& [level, num] : perLevel) {
        allVCFreeCellNum_[chain][level] += num;
      }
    }
    (void)vc;
  }
  // Validate: the VCs' free cells fit into the physical cluster at every level.
  for (auto& [chain, chainFreeCellNum] : allVCFreeCellNum_) {
    ChainCellList& ccl = fullCellList_.at(chain);
    int top = ccl.top();
    int available = static_cast<int>(ccl.at(top).size());
    for (int l = top; l >= kLowestLevel; l--) {
      int need = chainFreeCellNum.count(l) ? chainFreeCellNum[l] : 0;
      int left = available - need;
      if (left < 0) {
        throw HivedError::BadRequest(
            "Illegal initial VC assignment: insufficient physical cells at chain " + chain +
            " level " + std::to_string(l) + ": " + std::to_string(need) + " needed, " +
            std::to_string(available) + " available");
      }
      if (l > kLowestLevel) {
        int childNum = static_cast<int>(ccl.at(l)[0]->children.size());
        available = left * childNum;
      }
    }
  }
}

// Static bindings for pinned cells; removes them from the free list.
void HivedCore::initPinnedCells() {
  for (auto& [vcName, perId] : pinnedPhysical_) {
    for (auto& [pid, pc] : perId) {
      allocatePreassignedCell(pc, vcName, false);
      ChainCellList& vlist = vcSchedulers_[vcName].pinned[pid];
      auto* pinnedVirtual = static_cast<VirtualCell*>(vlist.at(vlist.top())[0]);
      bindCell(pc, pinnedVirtual);
    }
  }
}

// All nodes start bad until the informer confirms them healthy.
void HivedCore::initBadNodes() {
  std::vector<std::string> nodes;
  for (auto& [node, cells] : nodeLeafCellsStorage_) nodes.push_back(node);
  for (auto& n : nodes) setNodeHealthy(n, false);
}

HivedCore::HivedCore(const ClusterSpec& spec) {
  buildFromSpec(spec);
  initCellNums();
  initPinnedCells();
  initBadNodes();
}

HivedCore::~HivedCore() = default;

std::vector<std::string> HivedCore::allNodes() const {
  std::vector<std::string> nodes;
  for (auto& [node, cells] : nodeLeafCellsStorage_) nodes.push_back(node);
  return nodes;
}

std::vector<std::string> HivedCore::chains() const {
  std::vector<std::string> out;
  for (auto& [chain, ccl] : fullCellList_) out.push_back(chain);
  return out;
}

std::map<int, std::string> HivedCore::chainLevelTypes(const std::string& chain) const {
  auto it = cellTypes_.find(chain);
  if (it == cellTypes_.end()) return {};
  return it->second;
}

}  // namespace hived
