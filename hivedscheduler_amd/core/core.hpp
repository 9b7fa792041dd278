// MI355X-native gang-scheduler core: cell model, buddy allocation,
// topology-aware placement, VC-safety accounting, preemption lifecycle.
//
// This is a from-scratch C++ implementation of the scheduling semantics of
// microsoft/hivedscheduler's pkg/algorithm (Go), redesigned for:
//  - microsecond-scale Schedule() latency (the headline metric): pointer-based
//    cell trees, no per-call allocations on the hot path where avoidable;
//  - the CDNA4 cell chain (MI355X -> xGMI pair -> quad -> 8-GPU node -> pool)
//    with HBM capacity and link health as first-class cell attributes;
//  - a best-fit recursive descent for intra-node placement that directly
//    yields LCA-minimal placements (instead of combination backtracking).
//
// Behavioral parity references (semantics, not code):
//   cell model            ~ pkg/algorithm/cell.go
//   buddy alloc / binding ~ pkg/algorithm/cell_allocation.go
//   placement engine      ~ pkg/algorithm/topology_aware_scheduler.go
//   algorithm facade      ~ pkg/algorithm/hived_algorithm.go
//   state machines        ~ doc/design/state-machine.md
#pragma once

#include <algorithm>
#include <map>
#include <memory>
#include <optional>
#include <random>
#include <set>
#include <stdexcept>
#include <string>
#include <unordered_map>
#include <unordered_set>
#include <vector>

namespace hived {

// ---------------------------------------------------------------------------
// Constants
// ---------------------------------------------------------------------------
constexpr int kMaxGuaranteedPriority = 1000;
constexpr int kMinGuaranteedPriority = 0;
constexpr int kOpportunisticPriority = -1;
constexpr int kFreePriority = -2;
constexpr int kLowestLevel = 1;
constexpr int kHighestLevel = 100;

enum class CState { Free, Used, Reserving, Reserved };
enum class GState { Allocated, Preempting, BeingPreempted };

const char* to_string(CState s);
const char* to_string(GState s);

// Error with an HTTP status code; translated to a Python exception.
struct HivedError : std::runtime_error {
  int code;
  HivedError(int code_, const std::string& msg) : std::runtime_error(msg), code(code_) {}
  static HivedError BadRequest(const std::string& msg) { return HivedError(400, msg); }
  static HivedError NotFound(const std::string& msg) { return HivedError(404, msg); }
  static HivedError Internal(const std::string& msg) { return HivedError(500, msg); }
};

// ---------------------------------------------------------------------------
// Cell model
// ---------------------------------------------------------------------------
struct Group;
struct PhysicalCell;
struct VirtualCell;

struct Cell {
  std::string chain;        // top-level cell type name identifying the chain
  int level = 0;            // 1 = leaf
  int totalLeaf = 0;        // number of leaf cells contained
  std::string address;      // unique address
  std::string typeName;     // cell type name at this level
  bool atOrAboveNode = false;
  bool isNodeLevel = false;
  Cell* parent = nullptr;
  std::vector<Cell*> children;
  int priority = kFreePriority;
  bool healthy = true;
  // priority -> used leaf-cell count (rolled up the ancestor path)
  std::map<int, int> usedLeafAtPriority;
  // count of leaves under this cell at kFreePriority, maintained by
  // setCellPriority on leaf free<->used transitions. Together with
  // usedLeafAtPriority this gives the placement engine O(#priorities)
  // availability per node instead of an O(subtree) walk — the filter
  // hot path's dominant cost at cluster scale (gprof: availLeaves was
  // 100% of flat samples before this cache).
  int freeLeavesUnder = 0;

  virtual ~Cell() = default;
  virtual bool isPhysical() const = 0;
  int usedAt(int p) const {
    auto it = usedLeafAtPriority.find(p);
    return it == usedLeafAtPriority.end() ? 0 : it->second;
  }
};

struct PhysicalCell : Cell {
  std::vector<std::string> nodes;  // node names covered (size 1 at/below node level)
  std::vector<int> leafIndices;    // leaf (GPU) indices, within-node
  CState state = CState::Free;
  bool split = false;   // children are in the free list instead of this cell
  bool pinned = false;
  std::string pinnedId;
  VirtualCell* virt = nullptr;       // bound virtual cell
  Group* usingGroup = nullptr;       // Allocated / BeingPreempted group (leaf only)
  Group* reservingGroup = nullptr;   // Preempting group (leaf only)
  std::string otVC;                  // VC using this cell opportunistically
  // MI355X hardware attributes (BASELINE north star: 288 GB HBM and
  // xGMI-link health as first-class cell attributes)
  long long hbmBytes = 0;
  // count of degraded xGMI links whose BOTH endpoints are leaves under this
  // cell (leaf cells: 0). Rolled up from setXgmiLinkHealthy: the LCA of the
  // link's endpoints and all its ancestors carry the count. A link-degraded
  // pair/quad stays usable for placements that avoid co-placing the two
  // endpoints (and for 1-GPU work) — unlike leaf badness, which removes the
  // GPU entirely. Extends the reference's healthiness seam (cell.go:302-312)
  // with a per-link dimension.
  int badLinksUnder = 0;
  // leaf cells only: peer leaves connected by a currently-degraded link
  std::vector<PhysicalCell*> badLinkPeers;
  bool isPhysical() const override { return true; }
};

// One xGMI link's measured state (node-local, endpoints are GPU indices).
struct XgmiLink {
  PhysicalCell* a = nullptr;  // leaf cells, a->leafIndices[0] < b->leafIndices[0]
  PhysicalCell* b = nullptr;
  double gbps = 0.0;          // measured bandwidth (0 = not measured)
  bool healthy = true;
};

struct VirtualCell : Cell {
  std::string vc;
  VirtualCell* preassigned = nullptr;  // root ancestor in the VC forest
  PhysicalCell* phys = nullptr;        // bound physical cell
  std::string pinnedId;                // non-empty if in a pinned pool
  bool isPhysical() const override { return false; }
};

// Level-indexed cell lists for one chain (index 1..top).
struct ChainCellList {
  std::vector<std::vector<Cell*>> byLevel;  // byLevel[0] unused
  int top() const { return static_cast<int>(byLevel.size()) - 1; }
  void init(int topLevel) { byLevel.assign(topLevel + 1, {}); }
  std::vector<Cell*>& at(int l) { return byLevel[l]; }
  const std::vector<Cell*>& at(int l) const { return byLevel[l]; }
  bool contains(Cell* c, int l) const {
    if (l < 0 || l >= static_cast<int>(byLevel.size())) return false;
    auto& v = byLevel[l];
    return std::find(v.begin(), v.end(), c) != v.end();
  }
  void remove(Cell* c, int l) {
    auto& v = byLevel[l];
    auto it = std::find(v.begin(), v.end(), c);
    if (it == v.end()) throw HivedError::Internal("cell not found in list when removing: " + c->address);
    *it = v.back();
    v.pop_back();
  }
  void add(Cell* c, int l) { byLevel[l].push_back(c); }
  ChainCellList shallowCopy() const { return *this; }
};

// ---------------------------------------------------------------------------
// Affinity groups
// ---------------------------------------------------------------------------
struct PodPlacementInfo {
  std::string node;
  std::vector<int> leafIndices;
  std::vector<std::string> preassignedTypes;
};
struct BindInfo {
  std::string node;
  std::vector<int> isolation;
  std::string chain;
  // one entry per distinct leafCellNumber; each has podPlacements
  std::vector<std::vector<PodPlacementInfo>> memberBindInfo;
};

struct AllocatedPod {
  bool present = false;
  std::string key;   // "ns/name"
  std::string node;
  BindInfo bindInfo; // full group bind info replicated in every pod
};

struct LazyPreemptionStatus {
  std::string preemptor;
  std::string preemptionTime;
};

struct Group {
  std::string name;
  std::string vc;
  bool lazyPreemptionEnable = false;
  bool ignoreK8sSuggestedNodes = true;
  bool gangReleaseEnable = false;
  int priority = 0;
  GState state = GState::Allocated;
  std::map<int, int> totalPodNums;  // leafCellNum -> pod count
  std::map<int, std::vector<AllocatedPod>> allocatedPods;
  // leafCellNum -> pods -> leaf cells (entries may be null after reconfig)
  std::map<int, std::vector<std::vector<PhysicalCell*>>> physPlacement;
  bool hasVirtualPlacement = true;  // false for opportunistic / lazy-preempted
  std::map<int, std::vector<std::vector<VirtualCell*>>> virtPlacement;
  std::set<std::string> preemptingPods;  // pod keys (Preempting state only)
  std::optional<LazyPreemptionStatus> lazyStatus;
};

// ---------------------------------------------------------------------------
// Normalized cluster spec (filled from Python)
// ---------------------------------------------------------------------------
struct CellTypeSpec {
  std::string child;  // empty = leaf
  int childCount = 0;
  bool isNode = false;
};
struct PhysCellSpec {
  std::string type;
  std::string address;
  std::string pinnedId;
  std::vector<PhysCellSpec> children;
  // measured per-GPU HBM capacity in bytes (0 = unknown); leaf entries only
  long long hbmBytes = 0;
  // measured xGMI links (a, b, gbps, healthy); node-level entries only,
  // emitted by rocm-topo-discover
  std::vector<std::tuple<int, int, double, bool>> xgmiLinks;
};
struct VirtCellSpec {
  std::string typePath;  // "TOP.CHILD....", chain = first component
  int number = 0;
};
struct VCSpec {
  std::vector<VirtCellSpec> virtualCells;
  std::vector<std::string> pinnedIds;
};
struct ClusterSpec {
  std::map<std::string, CellTypeSpec> cellTypes;
  std::vector<PhysCellSpec> physicalCells;
  std::map<std::string, VCSpec> virtualClusters;
};

// ---------------------------------------------------------------------------
// Scheduling request / result
// ---------------------------------------------------------------------------
struct PodSpec {
  std::string vc;
  int priority = 0;
  std::string pinnedCellId;
  std::string leafCellType;
  int leafCellNumber = 0;
  bool gangReleaseEnable = false;
  bool lazyPreemptionEnable = false;
  bool ignoreK8sSuggestedNodes = true;
  std::string groupName;
  std::map<int, int> groupPodNums;  // leafCellNum -> pod count
  // optional: minimum measured HBM per leaf cell (bytes); leaves below are
  // unavailable for this request (MI355X: a GPU reporting < 288 GB is sick)
  long long hbmBytesPerCell = 0;
};

enum class Phase { Filtering, Preempting };

struct ScheduleResult {
  enum class Kind { Bind, Preempt, Wait } kind = Kind::Wait;
  // Bind
  BindInfo bindInfo;
  // Preempt: victims on one node (gang semantics: whole victim groups)
  std::string victimNode;
  std::vector<std::string> victimPodKeys;
  // Wait
  std::string waitReason;
};

// Placement = leafCellNum -> pods -> leaf cells
template <class CellT>
using Placement = std::map<int, std::vector<std::vector<CellT*>>>;

// One globally consistent link-clean view of a chain's free capacity:
// `excluded` = the physical leaves dropped (per-node max independent set of
// the degraded-link graph restricted to free leaves); `caps` = per-level max
// in-world free capacity of any single physical cell.
struct CleanShapeWorld {
  // the scheduling tier this world models: in-world availability counts
  // leaves below this priority (kOpportunisticPriority = free cells only;
  // a request's own priority = free + preemptible)
  int priority = kOpportunisticPriority;
  std::map<int, int> caps;
  std::unordered_set<PhysicalCell*> excluded;
  // unbound physical cells per level, for hint resolution when no bound
  // ancestor scopes the candidates (see topo_sched.cpp pickLeavesWorld)
  std::map<int, std::vector<PhysicalCell*>> physByLevel;
};

struct SchedulingRequest {
  std::string vc;
  std::string pinnedCellId;
  std::string chain;
  std::string groupName;
  std::map<int, int> podLeafCellNums;
  int priority = 0;
  const std::set<std::string>* suggestedNodes = nullptr;
  bool ignoreSuggestedNodes = true;
  long long hbmBytes = 0;  // minimum per-leaf HBM capacity (0 = any)
  // Clean-shape world, set only when the chain carries degraded xGMI links
  // and the gang needs >= 2 leaves: ONE globally consistent choice of
  // link endpoints to avoid. The virtual descent's link-honoring attempts
  // schedule inside this world (excluded leaves unavailable; unbound
  // subtrees capped by the best physical cell's in-world capacity; bound
  // subtrees min-capped by their physical region's in-world capacity), so
  // a request whose clean mapping needs a lower-affinity shape picks that
  // shape up front instead of falling to a dirty placement. Dirty rungs
  // run without the world, so capacity is never sacrificed.
  const CleanShapeWorld* cleanWorld = nullptr;
  // run only the link-honoring placement rungs (the caller iterates over
  // several clean-shape worlds and provides the dirty fallback itself)
  bool honorLinksOnly = false;
};

// ---------------------------------------------------------------------------
// Topology-aware placement engine
// ---------------------------------------------------------------------------
// Places pods onto a cluster view (list of node-level cells, or top-level cells
// below node level) sorted healthy > suggested > packing; inside a node, a
// best-fit recursive descent picks leaf cells with minimal LCA level.
class TopoScheduler {
 public:
  TopoScheduler() = default;
  TopoScheduler(const ChainCellList& ccl, std::map<int, int> levelLeafNum, bool crossPriorityPack);

  // Returns placement or empty with failedReason set. minHbmBytes > 0
  // filters out leaves whose measured HBM capacity falls short.
  // cleanWorld (optional): see SchedulingRequest::cleanWorld.
  bool Schedule(const std::map<int, int>& podLeafCellNums, int priority,
                const std::set<std::string>& suggestedNodes, bool ignoreSuggestedNodes,
                Placement<Cell>* out, std::string* failedReason,
                long long minHbmBytes = 0,
                const CleanShapeWorld* cleanWorld = nullptr, bool honorOnly = false) const;

 private:
  struct NodeView {
    Cell* c = nullptr;
    int freeAtPriority = 0;
    int usedSamePriority = 0;
    int usedHigherPriority = 0;
    bool healthy = true;
    bool suggested = true;
  };
  bool tryScheduleAtPriority(const std::vector<int>& sortedLeafNums, int priority,
                             const std::set<std::string>& suggestedNodes, bool ignoreSuggestedNodes,
                             long long minHbmBytes, bool honorLinks,
                             const CleanShapeWorld* cleanWorld, Placement<Cell>* out,
                             std::string* failedReason) const;

  std::vector<Cell*> viewCells_;
  std::map<int, int> levelLeafNum_;
  bool crossPriorityPack_ = false;
};

// ---------------------------------------------------------------------------
// Intra-VC scheduler
// ---------------------------------------------------------------------------
struct IntraVCScheduler {
  std::map<std::string, ChainCellList> nonPinnedFull;         // chain -> all virtual cells
  std::map<std::string, ChainCellList> nonPinnedPreassigned;  // chain -> preassigned cells
  std::map<std::string, ChainCellList> pinned;                // pinnedId -> subtree cells
  std::map<std::string, TopoScheduler> nonPinnedSchedulers;   // per chain
  std::map<std::string, TopoScheduler> pinnedSchedulers;      // per pinnedId

  bool schedule(const SchedulingRequest& sr, Placement<VirtualCell>* out, std::string* failedReason) const;
};

// A vertex in a cell-binding path: a tree of unbound virtual cells that need
// physical bindings, preserving intra-cell topology.
struct BindingVertex {
  VirtualCell* cell = nullptr;
  std::vector<std::unique_ptr<BindingVertex>> children;
};

// ---------------------------------------------------------------------------
// The algorithm facade
// ---------------------------------------------------------------------------
class HivedCore {
 public:
  explicit HivedCore(const ClusterSpec& spec);
  ~HivedCore();

  // -- node health (informer events) --
  void setNodeHealthy(const std::string& node, bool healthy);
  // -- GPU/xGMI-level health (rocm-smi exporter / probe events): marks one
  // leaf cell; badness rolls up to pair/quad/node cells automatically --
  void setLeafCellHealthy(const std::string& node, int leafIndex, bool healthy);
  // -- xGMI link health: a degraded link between two GPUs of one node marks
  // the LINK (first-class), not the endpoint leaves: multi-GPU placements
  // avoid co-placing the endpoints while 1-GPU work still uses them --
  void setXgmiLinkHealthy(const std::string& node, int a, int b, bool healthy,
                          double gbps = 0.0);
  // per-node link table for inspect: (a, b, gbps, healthy)
  std::vector<std::tuple<int, int, double, bool>> xgmiLinks(const std::string& node) const;
  std::vector<std::string> allNodes() const;
  std::set<std::string> badNodes() const { return badNodes_; }

  // -- scheduling --
  ScheduleResult schedule(const PodSpec& s, const std::string& podKey,
                          const std::set<std::string>& suggestedNodes, Phase phase);
  void deleteUnallocatedPod(const PodSpec& s, const std::string& podKey);
  void addAllocatedPod(const PodSpec& s, const BindInfo& info, const std::string& podKey);
  void deleteAllocatedPod(const PodSpec& s, const BindInfo& info, const std::string& podKey);

  // -- inspect --
  const std::map<std::string, std::unique_ptr<Group>>& groups() const { return groups_; }
  const std::map<std::string, ChainCellList>& fullCellList() const { return fullCellList_; }
  const std::map<std::string, ChainCellList>& freeCellList() const { return freeCellList_; }
  const std::map<std::string, IntraVCScheduler>& vcSchedulers() const { return vcSchedulers_; }
  std::vector<std::string> chains() const;
  std::map<int, std::string> chainLevelTypes(const std::string& chain) const;

 private:
  friend struct BuildContext;

  // --- construction ---
  void buildFromSpec(const ClusterSpec& spec);
  void initCellNums();
  void initPinnedCells();
  void initBadNodes();

  // --- health ---
  void setBadCell(PhysicalCell* c);
  void setHealthyCell(PhysicalCell* c);
  void addBadFreeCell(PhysicalCell* c);
  void removeBadFreeCell(PhysicalCell* c);
  bool doomAllocationIsSafe(PhysicalCell* pc);
  // Doomed-bad checks are DEFERRED to the end of the enclosing top-level
  // operation (schedule commit, pod delete, health event): executing them
  // mid-operation re-enters free-list surgery against a half-updated list
  // and can consume cells an in-flight placement depends on (fuzz-found
  // corruption). tryBind/tryUnbind queue when an operation is active; the
  // bodies run from drainDoomChecks() once state is consistent.
  void tryBindDoomedBadCell(const std::string& chain, int level);
  void tryUnbindDoomedBadCell(const std::string& chain, int level);
  void doBindDoomedBadCell(const std::string& chain, int level);
  void doUnbindDoomedBadCell(const std::string& chain, int level);
  void drainDoomChecks();

  class OpGuard {
   public:
    explicit OpGuard(HivedCore* c)
        : c_(c), outer_(!c->inOperation_), exceptionsAtEntry_(std::uncaught_exceptions()) {
      c_->inOperation_ = true;
    }
    // On normal exit (including early returns) the outermost guard drains
    // the deferred doom checks; on exception unwind it only resets the flag
    // (queued checks stay pending and run at the next operation's end).
    ~OpGuard() noexcept(false) {
      if (!outer_) return;
      c_->inOperation_ = false;
      if (std::uncaught_exceptions() == exceptionsAtEntry_) c_->drainDoomChecks();
    }

   private:
    HivedCore* c_;
    bool outer_;
    int exceptionsAtEntry_;
  };

  // --- scheduling internals ---
  ScheduleResult generateResult(const Placement<PhysicalCell>& phys, bool hasVirtual,
                                const Placement<VirtualCell>& virt,
                                const std::map<std::string, std::set<std::string>>& victims,
                                const std::string& waitReason, int currentLeafNum, int podIndex,
                                Group* group, const std::string& groupName);
  bool schedulePodFromExistingGroup(Group* g, const PodSpec& s,
                                    const std::set<std::string>& suggestedNodes, Phase phase,
                                    const std::string& podKey, Placement<PhysicalCell>* phys,
                                    bool* hasVirtual, Placement<VirtualCell>* virt,
                                    std::map<std::string, std::set<std::string>>* victims,
                                    int* podIndex);
  void schedulePodFromNewGroup(const PodSpec& s, const std::set<std::string>& suggestedNodes,
                               Phase phase, const std::string& podKey,
                               Placement<PhysicalCell>* phys, bool* hasVirtual,
                               Placement<VirtualCell>* virt,
                               std::map<std::string, std::set<std::string>>* victims,
                               std::string* waitReason);
  bool scheduleNewAffinityGroup(const PodSpec& s, const std::set<std::string>& suggestedNodes,
                                const std::string& podKey, Placement<PhysicalCell>* phys,
                                bool* hasVirtual, Placement<VirtualCell>* virt,
                                std::string* failedReason);
  bool scheduleForLeafCellType(SchedulingRequest& sr, const std::string& leafCellType,
                               const std::string& podKey, bool typeSpecified,
                               Placement<PhysicalCell>* phys, bool* hasVirtual,
                               Placement<VirtualCell>* virt, std::string* failedReason);
  bool handleSchedulingRequest(const SchedulingRequest& sr, Placement<PhysicalCell>* phys,
                               bool* hasVirtual, Placement<VirtualCell>* virt,
                               std::string* failedReason);
  bool scheduleGuaranteedGroup(const SchedulingRequest& sr, Placement<PhysicalCell>* phys,
                               Placement<VirtualCell>* virt, std::string* failedReason);
  bool chainHasBadLinks(const std::string& chain);
  bool scheduleOpportunisticGroup(const SchedulingRequest& sr, Placement<PhysicalCell>* phys,
                                  std::string* failedReason);
  void validateSchedulingRequest(const SchedulingRequest& sr, const std::string& podKey);

  std::map<std::string, Placement<VirtualCell>> tryLazyPreempt(const Placement<VirtualCell>& p,
                                                               const std::string& groupName);

  // --- group lifecycle ---
  void createAllocatedGroup(const PodSpec& s, const BindInfo& info, const std::string& podKey);
  void deleteAllocatedGroup(Group* g, const std::string& podKey);
  void createPreemptingGroup(const PodSpec& s, const Placement<PhysicalCell>& phys,
                             const Placement<VirtualCell>& virt, const std::string& podKey);
  void deletePreemptingGroup(Group* g, const std::string& podKey);
  void allocatePreemptingGroup(Group* g, const std::string& podKey);
  Placement<VirtualCell> lazyPreemptGroup(Group* victim, const std::string& preemptor);
  void lazyPreemptCell(VirtualCell* c, const std::string& preemptor);
  void revertLazyPreempt(Group* g, const Placement<VirtualCell>& virt);

  // --- leaf-cell allocate/release + safety accounting ---
  std::pair<PhysicalCell*, VirtualCell*> findAllocatedLeafCell(
      int index, const PodPlacementInfo& placement, const std::string& chain, const PodSpec& s,
      Group* group, const std::string& podKey, bool* lazyPreempt, bool* isOpportunistic);
  std::pair<bool, std::string> allocateLeafCell(PhysicalCell* p, VirtualCell* v, int priority,
                                                const std::string& vc);
  void releaseLeafCell(PhysicalCell* p, const std::string& vc);
  std::pair<bool, std::string> allocatePreassignedCell(PhysicalCell* c, const std::string& vc,
                                                       bool doomedBad);
  void releasePreassignedCell(PhysicalCell* c, const std::string& vc, bool doomedBad);
  void allocateBadCell(PhysicalCell* c);
  void releaseBadCell(PhysicalCell* c);
  int removeCellFromFreeList(PhysicalCell* c);
  int addCellToFreeList(PhysicalCell* c);

  // --- mapping virtual -> physical (buddy allocation) ---
  bool mapVirtualPlacementToPhysical(std::vector<BindingVertex*>& preassigned,
                                     std::vector<std::vector<BindingVertex*>>& nonPreassigned,
                                     ChainCellList freeList, std::map<int, int> freeCellNum,
                                     const std::set<std::string>& suggestedNodes,
                                     bool ignoreSuggestedNodes,
                                     std::unordered_map<VirtualCell*, PhysicalCell*>& bindings,
                                     long long minHbmBytes = 0, bool honorLinks = false,
                                     const CleanShapeWorld* world = nullptr);

 public:
  // state (public for inspect/bindings simplicity; external mutation forbidden)
  std::map<std::string, ChainCellList> fullCellList_;
  std::map<std::string, ChainCellList> freeCellList_;
  bool inOperation_ = false;
  std::vector<std::pair<std::string, int>> pendingDoomChecks_;
  std::map<std::string, IntraVCScheduler> vcSchedulers_;
  std::map<std::string, TopoScheduler> opportunisticSchedulers_;
  std::map<std::string, std::unique_ptr<Group>> groups_;

  // vc -> chain -> level -> free preassigned cell count
  std::map<std::string, std::map<std::string, std::map<int, int>>> vcFreeCellNum_;
  std::map<std::string, std::map<int, int>> allVCFreeCellNum_;
  std::map<std::string, std::map<int, int>> totalLeftCellNum_;

  std::map<std::string, ChainCellList> badFreeCells_;
  std::map<std::string, std::map<std::string, ChainCellList>> vcDoomedBadCells_;
  std::map<std::string, std::map<int, int>> allVCDoomedBadCellNum_;

  std::set<std::string> badNodes_;
  // leaves individually marked bad (GPU/xGMI level), independent of node health
  std::set<PhysicalCell*> badLeafMarks_;
  // node -> (minIdx, maxIdx) -> link record (first-class xGMI link state)
  std::map<std::string, std::map<std::pair<int, int>, XgmiLink>> xgmiLinks_;
  std::map<std::string, std::vector<std::string>> cellChains_;        // leaf type -> chains
  std::map<std::string, std::map<int, std::string>> cellTypes_;      // chain -> level -> type
  std::map<std::string, std::map<int, int>> leafCellNums_;           // chain -> level -> leaf num
  // pinned: vc -> pinnedId -> physical cell
  std::map<std::string, std::map<std::string, PhysicalCell*>> pinnedPhysical_;

  // seeded PRNG for victim-node selection (deterministic for tests/fuzz,
  // spreads victim churn across nodes like the reference's rand)
  std::minstd_rand victimRng_{12345};
  // cell ownership
  std::vector<std::unique_ptr<Cell>> cellStore_;
  // node name -> leaf cells on that node (for health propagation)
  std::map<std::string, std::vector<PhysicalCell*>> nodeLeafCellsStorage_;
  long long scheduleCount_ = 0;
  // Clean-shape-world cache, validated against gWorldEpochCounter (bumped
  // by every mutation that can change a world: leaf priorities, bindings,
  // health, links). Wait-storms — the common case under contention: many
  // failed filters with no allocation in between — reuse cached worlds
  // instead of recomputing them per Schedule.
  struct WorldCacheEntry {
    unsigned long long epoch = 0;
    std::vector<CleanShapeWorld> worlds;
  };
  std::map<std::pair<std::string, int>, WorldCacheEntry> worldCache_;
};

// helpers shared across translation units
extern unsigned long long gWorldEpochCounter;
void setCellPriority(Cell* c, int p);
void updateUsedLeafCellNumAtPriority(Cell* c, int p, bool increase);
void bindCell(PhysicalCell* pc, VirtualCell* vc);
void unbindCell(PhysicalCell* c);
VirtualCell* getUnboundVirtualCell(const std::vector<Cell*>& cl);
VirtualCell* getLowestPriorityVirtualCell(const std::vector<Cell*>& cl, int p);
VirtualCell* mapPhysicalCellToVirtual(PhysicalCell* c, const ChainCellList& vccl,
                                      int preassignedLevel, int p, std::string* message);
bool inFreeCellList(PhysicalCell* c);
void setCellState(PhysicalCell* c, CState s);
PhysicalCell* findPhysicalLeafCell(std::map<std::string, ChainCellList>& fullCellList,
                                   const std::string& chain, const std::string& node,
                                   int leafIndex);
Cell* ancestorNoHigherThanNode(Cell* c);
// the chain's clean-shape worlds: one per enumerated consistent endpoint
// choice (different max independent sets admit different clean shapes)
std::vector<CleanShapeWorld> computeCleanShapeWorlds(
    const ChainCellList& ccl, const std::set<std::string>* suggestedNodes,
    size_t maxWorlds = 4, int priority = kOpportunisticPriority);
void checkInvariants(const HivedCore& core);

}  // namespace hived
