// Topology-aware placement engine.
//
// Semantics parity: pkg/algorithm/topology_aware_scheduler.go (node sorting,
// two-pass preemption, gang fit). The intra-node search is redesigned: instead
// of backtracking over leaf-cell combinations with LCA pruning (reference
// l.309-387), a best-fit recursive descent over the cell tree directly yields
// an LCA-minimal, buddy-packed placement in O(depth * fanout) per pod — on the
// fixed MI355X chain (leaf->pair->quad->node) this is a handful of comparisons,
// which is what makes microsecond-scale Schedule() latency possible.
//
// MI355X-native additions over the reference:
//  - per-leaf health (a sick GPU, not a whole node, is unavailable);
//  - first-class xGMI link health: a degraded link between two GPUs excludes
//    CO-PLACING its endpoints in one gang (the gang's all-reduce would ride
//    the sick link) while both GPUs stay available for 1-GPU work — the
//    reference's node/leaf health model cannot express this;
//  - per-leaf HBM capacity: a request may demand hbmBytesPerCell; leaves
//    whose measured capacity falls short are unavailable for it.
#include <climits>
#include <functional>

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

// Resolve a view cell (virtual or physical) to its physical identity, if any.
// Virtual cells not (yet) bound have no physical identity: no link/HBM facts.
PhysicalCell* physOf(Cell* c) {
  if (c->isPhysical()) return static_cast<PhysicalCell*>(c);
  return static_cast<VirtualCell*>(c)->phys;
}

int badLinksUnderOf(Cell* c) {
  PhysicalCell* p = physOf(c);
  return p == nullptr ? 0 : p->badLinksUnder;
}

struct PickSession {
  std::unordered_set<Cell*> taken;
  // leaves excluded for THIS request because their bad-link peer may be
  // co-placed (gang-wide: one session spans all pods of the gang)
  std::unordered_set<Cell*> excluded;
  // minimum per-leaf HBM capacity demanded by the request (0 = any)
  long long minHbm = 0;
  // restrict leaf usability to these nodes (caps computation honors the
  // K8s suggested-node set the mapping stage will enforce)
  const std::set<std::string>* suggestedFilter = nullptr;
  // physical leaves consumed by earlier world-mode picks of THIS session
  // (gang pods share the session): the hint math must not offer the same
  // physical capacity to two pods — static in-world counts made pod 3's
  // matching hint a pair whose leaves pod 2 had already spoken for, and
  // the mapping then had no clean assignment (found by the gang oracle)
  std::unordered_set<PhysicalCell*> physTaken;
  // virtual leaf -> the hint leaf it consumed (for exact rollback)
  std::unordered_map<Cell*, PhysicalCell*> hintTaken;
  // link-honoring attempts on VIRTUAL views only: the chain's clean-shape
  // world. Excluded physical leaves are unavailable (applied to bound
  // leaves directly and to bound regions via a physical min-cap); unbound
  // virtual cells — which must eventually map onto ONE physical cell of
  // their level — are capped by the best physical cell's in-world
  // capacity. This makes the (otherwise link-blind) virtual descent choose
  // shapes that admit a clean physical mapping (e.g. straddle quads when
  // no single quad has 4 clean leaves).
  const CleanShapeWorld* world = nullptr;
};

// in-world free capacity of a PHYSICAL region (free + healthy + not
// excluded + not consumed by an earlier world-mode pick of this session)
int physFreeInWorld(PhysicalCell* pc, const PickSession& s) {
  if (pc->level == kLowestLevel) {
    return (pc->priority < s.world->priority && pc->healthy && !s.world->excluded.count(pc) &&
            (s.physTaken.empty() || !s.physTaken.count(pc)))
               ? 1
               : 0;
  }
  int n = 0;
  for (Cell* child : pc->children) n += physFreeInWorld(static_cast<PhysicalCell*>(child), s);
  return n;
}

// best in-world capacity of any single UNBOUND physical cell at `level`
// under region `pc` (an unbound virtual cell whose nearest bound ancestor
// is bound to pc will map onto one of these)
int maxInWorldAtLevelUnder(PhysicalCell* pc, int level, const PickSession& s) {
  if (pc->level == level) return pc->virt == nullptr ? physFreeInWorld(pc, s) : 0;
  int best = 0;
  for (Cell* child : pc->children) {
    best = std::max(best, maxInWorldAtLevelUnder(static_cast<PhysicalCell*>(child), level, s));
  }
  return best;
}

// Leaf-granular availability: a bad physical leaf (sick GPU) is never
// available; a virtual leaf bound to a bad physical leaf (doomed-bad binding)
// is equally unavailable; a leaf with measured HBM below the request's demand
// is unavailable for this request. This is finer than the reference, whose
// health model stops at node granularity.
bool leafUsable(Cell* c, const PickSession& s) {
  PhysicalCell* p = physOf(c);
  if (p == nullptr) return true;  // unbound virtual: no physical facts yet
  if (!p->healthy) return false;
  if (s.minHbm > 0 && p->hbmBytes > 0 && p->hbmBytes < s.minHbm) return false;
  if (s.suggestedFilter != nullptr &&
      (p->nodes.empty() || !s.suggestedFilter->count(p->nodes[0]))) {
    return false;
  }
  if (s.world != nullptr && s.world->excluded.count(p)) return false;
  return true;
}

// (availableTotal, availableFree): usable leaves that are free or
// lower-priority (preemptible), excluding ones already taken or excluded in
// this session.
//
// O(#priorities) fast path from the freeLeavesUnder / usedLeafAtPriority
// caches, valid when the session carries no per-leaf state and the subtree
// has no bad leaves: a PHYSICAL cell's `healthy` roll-up proves that; a
// bound virtual cell inherits its physical partner's proof; an UNBOUND
// virtual cell has no bound descendants at all (bindings are upward-
// contiguous: bindCell binds leaf->root, exposure bindings require a bound
// parent), so every leaf under it is usable by construction.
std::pair<int, int> availLeaves(Cell* c, int p, const PickSession& s) {
  if (c->level == kLowestLevel) {
    if ((!s.taken.empty() && s.taken.count(c)) || (!s.excluded.empty() && s.excluded.count(c)) ||
        !leafUsable(c, s)) {
      return {0, 0};
    }
    if (c->priority == kFreePriority) return {1, 1};
    if (c->priority < p) return {1, 0};
    return {0, 0};
  }
  // the clean-shape world constrains virtual subtrees RECURSIVELY (parent
  // sums constrained children, then caps itself), so whenever it is active
  // the counter fast path is skipped EVERYWHERE — a parent's unconstrained
  // counters would overstate what its constrained children can deliver and
  // underflow the drain in pickLeaves
  bool inWorld = s.world != nullptr && !c->isPhysical();
  if (s.world == nullptr && s.minHbm == 0 && s.suggestedFilter == nullptr && s.taken.empty() &&
      s.excluded.empty()) {
    bool proven;
    if (c->isPhysical()) {
      proven = static_cast<PhysicalCell*>(c)->healthy;
    } else {
      PhysicalCell* ph = static_cast<VirtualCell*>(c)->phys;
      proven = ph == nullptr || ph->healthy;
    }
    if (proven) {
      int preemptible = 0;
      for (auto& [pp, n] : c->usedLeafAtPriority) {
        if (pp >= p) break;
        preemptible += n;
      }
      return {c->freeLeavesUnder + preemptible, c->freeLeavesUnder};
    }
  }
  int at = 0, af = 0;
  for (Cell* child : c->children) {
    auto [t, f] = availLeaves(child, p, s);
    at += t;
    af += f;
  }
  if (inWorld) {
    PhysicalCell* ph = static_cast<VirtualCell*>(c)->phys;
    if (ph == nullptr) {
      // unbound: will map onto SOME physical cell of this level — within
      // the nearest bound ancestor's region if one exists (its physical
      // identity pins the candidates), else anywhere in the chain
      PhysicalCell* region = nullptr;
      for (Cell* a = c->parent; a != nullptr; a = a->parent) {
        PhysicalCell* ap = static_cast<VirtualCell*>(a)->phys;
        if (ap != nullptr) {
          region = ap;
          break;
        }
      }
      int cap = INT_MAX;
      if (region != nullptr) {
        cap = maxInWorldAtLevelUnder(region, c->level, s);
      } else {
        auto it = s.world->caps.find(c->level);
        if (it != s.world->caps.end()) cap = it->second;
      }
      if (cap != INT_MAX) {
        at = std::min(at, cap);
        af = std::min(af, cap);
      }
    } else {
      // bound: the physical region is known; its in-world free capacity
      // bounds what the subtree (incl. positionally-ambiguous unbound
      // leaves) can deliver cleanly
      int cap = physFreeInWorld(ph, s);
      // preemptible leaves are outside the free-world analysis; only the
      // free component is capped, the preemptible surplus rides on top
      int preemptible = at - af;
      af = std::min(af, cap);
      at = af + preemptible;
    }
  }
  return {at, af};
}

void collectAvailableLeaves(Cell* c, int p, const PickSession& s, std::vector<Cell*>& out) {
  if (c->level == kLowestLevel) {
    if (!s.taken.count(c) && !s.excluded.count(c) && leafUsable(c, s) && c->priority < p) {
      out.push_back(c);
    }
    return;
  }
  for (Cell* child : c->children) collectAvailableLeaves(child, p, s, out);
}

// Max independent set in the bad-link graph over `verts` (adjacency masks).
// Bad links are rare (usually 0-2 per node), so the branch recursion is tiny;
// callers cap the vertex count.
unsigned long long maxIndepSet(const std::vector<unsigned long long>& adj,
                               unsigned long long cand) {
  if (cand == 0) return 0;
  int v = __builtin_ctzll(cand);
  unsigned long long bit = 1ull << v;
  // exclude v
  unsigned long long best = maxIndepSet(adj, cand & ~bit);
  // include v (drop its bad-link neighbors)
  unsigned long long with = bit | maxIndepSet(adj, cand & ~bit & ~adj[v]);
  if (__builtin_popcountll(with) > __builtin_popcountll(best)) best = with;
  return best;
}

// Link-clean analysis of `viewCell` for a request needing `need` leaves that
// will all communicate (one gang on one node): find the largest set of
// available leaves with no degraded xGMI link INSIDE the set. Returns the
// max clean count; when `excludeOut` is non-null and the clean capacity
// covers `need`, fills it with the leaves to exclude (the bad-link-incident
// leaves NOT in the chosen independent set).
// The bad-link graph over a view's available leaves: vertex v = incident
// available leaf (index into `leaves` via verts[v]); adjacency as bitmasks.
struct BadLinkGraph {
  std::vector<Cell*> leaves;
  std::vector<int> verts;
  std::vector<unsigned long long> adj;
};

BadLinkGraph buildBadLinkGraph(Cell* viewCell, int p, const PickSession& s) {
  BadLinkGraph g;
  collectAvailableLeaves(viewCell, p, s, g.leaves);
  // map physical leaf -> view leaf to resolve link endpoints in this view
  std::unordered_map<PhysicalCell*, int> physIdx;
  for (size_t i = 0; i < g.leaves.size(); i++) {
    PhysicalCell* ph = physOf(g.leaves[i]);
    if (ph != nullptr) physIdx[ph] = static_cast<int>(i);
  }
  std::unordered_map<int, int> leafToVert;
  auto vertOf = [&](int leafIdx) {
    auto it = leafToVert.find(leafIdx);
    if (it != leafToVert.end()) return it->second;
    int v = static_cast<int>(g.verts.size());
    leafToVert[leafIdx] = v;
    g.verts.push_back(leafIdx);
    g.adj.push_back(0);
    return v;
  };
  for (size_t i = 0; i < g.leaves.size(); i++) {
    PhysicalCell* ph = physOf(g.leaves[i]);
    if (ph == nullptr || ph->badLinkPeers.empty()) continue;
    for (PhysicalCell* peer : ph->badLinkPeers) {
      auto it = physIdx.find(peer);
      if (it == physIdx.end()) continue;
      int a = vertOf(static_cast<int>(i));
      int b = vertOf(it->second);
      if (a < 64 && b < 64) {
        g.adj[a] |= 1ull << b;
        g.adj[b] |= 1ull << a;
      }
    }
  }
  return g;
}

int cleanAvailAnalysis(Cell* viewCell, int p, const PickSession& s, int need,
                       std::vector<Cell*>* excludeOut) {
  auto [availTotal, availFree] = availLeaves(viewCell, p, s);
  (void)availFree;
  if (badLinksUnderOf(viewCell) == 0) return availTotal;
  BadLinkGraph g = buildBadLinkGraph(viewCell, p, s);
  int nv = static_cast<int>(g.verts.size());
  if (nv == 0) return availTotal;
  if (nv > 60) return availTotal - nv;  // degenerate; be conservative
  unsigned long long all = nv == 64 ? ~0ull : ((1ull << nv) - 1);
  unsigned long long chosen = maxIndepSet(g.adj, all);
  int clean = availTotal - nv + __builtin_popcountll(chosen);
  if (excludeOut != nullptr && clean >= need) {
    for (int v = 0; v < nv; v++) {
      if (!(chosen & (1ull << v))) excludeOut->push_back(g.leaves[g.verts[v]]);
    }
  }
  return clean;
}

// All maximum independent sets of the graph (masks), deduped by the
// caller, branch-and-bound pruned: a branch that cannot reach the best
// size seen so far is cut, so the maximum size is always found and only
// maximum-size sets accumulate (include-first order reaches a maximal
// set immediately, seeding the bound).
void enumerateMaxIndepMasks(const std::vector<unsigned long long>& adj,
                            unsigned long long cand, unsigned long long acc,
                            std::vector<unsigned long long>& out, size_t cap) {
  int best = 0;
  for (auto m : out) best = std::max(best, __builtin_popcountll(m));
  if (__builtin_popcountll(acc) + __builtin_popcountll(cand) < best) return;
  if (cand == 0) {
    out.push_back(acc);
    return;
  }
  if (out.size() >= cap * 32) return;  // hard stop on pathological graphs
  int v = __builtin_ctzll(cand);
  unsigned long long bit = 1ull << v;
  enumerateMaxIndepMasks(adj, cand & ~bit & ~adj[v], acc | bit, out, cap);
  enumerateMaxIndepMasks(adj, cand & ~bit, acc, out, cap);
}

// Exclusion-set VARIANTS for a top cell: every maximum independent set of
// its bad-link graph yields one valid "which endpoints to avoid" choice;
// different choices admit different clean shapes, so the ladder tries a few.
std::vector<std::vector<Cell*>> enumerateCleanExclusionVariants(Cell* top,
                                                                const PickSession& s,
                                                                size_t cap, int priority) {
  std::vector<std::vector<Cell*>> out;
  if (badLinksUnderOf(top) == 0) return out;
  BadLinkGraph g = buildBadLinkGraph(top, priority, s);
  int nv = static_cast<int>(g.verts.size());
  if (nv == 0 || nv > 24) {
    if (nv > 24) {  // degenerate: one conservative variant via the analysis
      std::vector<Cell*> excl;
      cleanAvailAnalysis(top, priority, s, 0, &excl);
      if (!excl.empty()) out.push_back(std::move(excl));
    }
    return out;
  }
  unsigned long long all = nv == 64 ? ~0ull : ((1ull << nv) - 1);
  std::vector<unsigned long long> masks;
  enumerateMaxIndepMasks(g.adj, all, 0, masks, cap);
  size_t best = 0;
  for (auto m : masks) best = std::max(best, (size_t)__builtin_popcountll(m));
  std::set<unsigned long long> seen;
  for (auto m : masks) {
    if ((size_t)__builtin_popcountll(m) != best || !seen.insert(m).second) continue;
    std::vector<Cell*> excl;
    for (int v = 0; v < nv; v++) {
      if (!(m & (1ull << v))) excl.push_back(g.leaves[g.verts[v]]);
    }
    out.push_back(std::move(excl));
    if (out.size() >= cap) break;
  }
  return out;
}

// Exclude, for the rest of this session (= this gang), the bad-link-incident
// leaves that a clean `need`-leaf placement on `viewCell` must avoid. No-op
// when clean capacity cannot cover the need (a dirty placement is then
// allowed: capacity guarantees outrank link quality).
void applyLinkExclusions(Cell* viewCell, int need, int p, PickSession& s) {
  if (need < 2 || badLinksUnderOf(viewCell) == 0) return;
  std::vector<Cell*> toExclude;
  cleanAvailAnalysis(viewCell, p, s, need, &toExclude);
  for (Cell* c : toExclude) s.excluded.insert(c);
}

// Lowest LCA level achievable for q leaves anywhere inside `cell`
// (INT_MAX if the subtree cannot hold q). This is the lookahead that makes
// the descent LCA-minimal: a child with MORE availability may reach a LOWER
// LCA (e.g. free = {0,1,2} in quad A vs {5,6} straddling quad B's pairs —
// A serves a pair request at pair level, B only at quad level). Verified
// against brute force in tests/test_placement_optimality.py.
int bestLCALevel(Cell* c, int q, int p, const PickSession& s) {
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
  // reaches the lowest LCA, then needs the fewest preemptions, then — for
  // 1-leaf requests — prefers link-degraded subtrees (parking 1-GPU work on
  // degraded pairs keeps clean pairs free for multi-GPU gangs; multi-leaf
  // requests avoid degraded endpoints via session exclusions instead), then
  // is the tightest fit (packing).
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
      int linkKey = badLinksUnderOf(cell->children[i]);
      int bestLinkKey = badLinksUnderOf(cell->children[best]);
      if (q == 1) {
        linkKey = -linkKey;
        bestLinkKey = -bestLinkKey;
      }
      if (lca < bestLca ||
          (lca == bestLca &&
           (needPreempt < bestPreempt ||
            (needPreempt == bestPreempt &&
             (linkKey < bestLinkKey ||
              (linkKey == bestLinkKey && av[i].first < av[best].first)))))) {
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

// World-mode descent with PHYSICAL MIRRORING: each virtual cell is picked
// against a tentative physical "hint" cell whose in-world structure the
// drain mirrors exactly. This is what makes multi-link shapes exact: level-
// max caps cannot express "this quad's two in-world leaves sit in DIFFERENT
// pairs", but the hint's own children can. Bound virtual children identify
// their hint child directly; unbound ones (all-unbound subtrees, since
// bindings are upward-contiguous) are rank-matched to the hint's unbound
// children by capacity. Returns false (instead of throwing) when the shape
// cannot be realized — the caller falls to the next world / dirty rung.
// Children of a world-mode (cell, hint) pair, matched: bound virtual
// children identify their hint child directly; unbound ones (all-unbound
// subtrees, since bindings are upward-contiguous) are rank-matched to the
// hint's unbound children by capacity. Used by BOTH the availability
// recursion and the descent so their answers agree.
struct WorldPair {
  Cell* child;
  PhysicalCell* hint;
  int avail;
};

bool matchWorldChildren(Cell* cell, PhysicalCell* hint, int p, const PickSession& s,
                        std::vector<WorldPair>& pairs) {
  std::vector<Cell*> unboundChildren;
  std::vector<PhysicalCell*> freeHints;
  for (Cell* childC : cell->children) {
    auto* vc = static_cast<VirtualCell*>(childC);
    if (vc->phys != nullptr) {
      pairs.push_back({vc, vc->phys, 0});
    } else {
      unboundChildren.push_back(vc);
    }
  }
  for (Cell* hc : hint->children) {
    auto* ph = static_cast<PhysicalCell*>(hc);
    if (ph->virt == nullptr) freeHints.push_back(ph);
  }
  if (freeHints.size() < unboundChildren.size()) return false;
  std::stable_sort(freeHints.begin(), freeHints.end(), [&](PhysicalCell* a, PhysicalCell* b) {
    return physFreeInWorld(a, s) > physFreeInWorld(b, s);
  });
  std::stable_sort(unboundChildren.begin(), unboundChildren.end(), [&](Cell* a, Cell* b) {
    return availLeaves(a, p, s).first > availLeaves(b, p, s).first;
  });
  for (size_t i = 0; i < unboundChildren.size(); i++) {
    pairs.push_back({unboundChildren[i], freeHints[i], 0});
  }
  return true;
}

// RECURSIVE world availability of a matched (cell, hint) pair: the virtual
// and physical structures are intersected level by level. A parent-level
// min(virtualAvail, physInWorld) is NOT enough — with session-taken leaves
// from earlier gang pods, the virtual side's remaining leaves and the
// hint's in-world leaves can sit in different sub-cells, and only the
// recursive intersection sees it (found by the gang-union oracle: a
// 3x2-pod gang claimed a quad could serve pod 2 when its pairs could
// deliver only 1 leaf each).
int availWorld(Cell* cell, PhysicalCell* hint, int p, const PickSession& s) {
  if (cell->level == kLowestLevel) {
    auto [t, f] = availLeaves(cell, p, s);
    (void)f;
    return std::min(t, physFreeInWorld(hint, s));
  }
  std::vector<WorldPair> pairs;
  if (!matchWorldChildren(cell, hint, p, s, pairs)) return 0;
  int total = 0;
  for (auto& pr : pairs) total += availWorld(pr.child, pr.hint, p, s);
  return total;
}

bool pickLeavesWorld(Cell* cell, PhysicalCell* hint, int q, int p, PickSession& s,
                     std::vector<Cell*>& out) {
  if (cell->level == kLowestLevel) {
    // re-check eligibility (the caller's counts came from caps, not hints)
    if (availWorld(cell, hint, p, s) < 1) return false;
    s.taken.insert(cell);
    s.physTaken.insert(hint);
    s.hintTaken[cell] = hint;
    out.push_back(cell);
    return true;
  }
  std::vector<WorldPair> pairs;
  if (!matchWorldChildren(cell, hint, p, s, pairs)) return false;
  int total = 0;
  for (auto& pr : pairs) {
    pr.avail = availWorld(pr.child, pr.hint, p, s);
    total += pr.avail;
  }
  if (total < q) return false;
  // single child fits: tightest such pair (packing); LCA-minimal since the
  // recursion descends whenever possible
  int best = -1;
  for (size_t i = 0; i < pairs.size(); i++) {
    if (pairs[i].avail >= q && (best < 0 || pairs[i].avail < pairs[best].avail)) {
      best = static_cast<int>(i);
    }
  }
  if (best >= 0) {
    return pickLeavesWorld(pairs[best].child, pairs[best].hint, q, p, s, out);
  }
  // drain, largest first
  std::stable_sort(pairs.begin(), pairs.end(), [](const WorldPair& a, const WorldPair& b) {
    return a.avail > b.avail;
  });
  int remaining = q;
  size_t outStart = out.size();
  for (auto& pr : pairs) {
    if (remaining == 0) break;
    int t = std::min(remaining, pr.avail);
    if (t > 0) {
      if (!pickLeavesWorld(pr.child, pr.hint, t, p, s, out)) {
        // roll back this request's takes (virtual AND hint-leaf claims)
        for (size_t i = outStart; i < out.size(); i++) {
          s.taken.erase(out[i]);
          auto it = s.hintTaken.find(out[i]);
          if (it != s.hintTaken.end()) {
            s.physTaken.erase(it->second);
            s.hintTaken.erase(it);
          }
        }
        out.resize(outStart);
        return false;
      }
      remaining -= t;
    }
  }
  if (remaining != 0) {
    for (size_t i = outStart; i < out.size(); i++) {
      s.taken.erase(out[i]);
      auto it = s.hintTaken.find(out[i]);
      if (it != s.hintTaken.end()) {
        s.physTaken.erase(it->second);
        s.hintTaken.erase(it);
      }
    }
    out.resize(outStart);
    return false;
  }
  return true;
}

// Candidate physical targets for a world-mode pick at `cell`: its own
// binding if bound; otherwise the unbound physical cells with static
// in-world capacity >= q, tightest first — scoped under the nearest bound
// ancestor's region, or chain-wide via the world's per-level cell list.
// The caller tries them in order: the static capacity is an upper bound
// and the recursive availWorld intersection can reject a candidate.
std::vector<PhysicalCell*> resolveWorldHints(Cell* cell, int q, const PickSession& s) {
  auto* vc = static_cast<VirtualCell*>(cell);
  if (vc->phys != nullptr) return {vc->phys};
  PhysicalCell* region = nullptr;
  for (Cell* a = cell->parent; a != nullptr; a = a->parent) {
    PhysicalCell* ap = static_cast<VirtualCell*>(a)->phys;
    if (ap != nullptr) {
      region = ap;
      break;
    }
  }
  std::vector<PhysicalCell*> candidates;
  auto consider = [&](PhysicalCell* ph) {
    if (ph->virt != nullptr) return;
    if (physFreeInWorld(ph, s) >= q) candidates.push_back(ph);
  };
  if (region != nullptr) {
    std::function<void(PhysicalCell*)> walk = [&](PhysicalCell* ph) {
      if (ph->level == cell->level) {
        consider(ph);
        return;
      }
      for (Cell* c : ph->children) walk(static_cast<PhysicalCell*>(c));
    };
    walk(region);
  } else {
    auto it = s.world->physByLevel.find(cell->level);
    if (it != s.world->physByLevel.end()) {
      for (PhysicalCell* ph : it->second) consider(ph);
    }
  }
  std::stable_sort(candidates.begin(), candidates.end(),
                   [&](PhysicalCell* a, PhysicalCell* b) {
                     return physFreeInWorld(a, s) < physFreeInWorld(b, s);
                   });
  return candidates;
}

std::tuple<bool, bool> healthyAndSuggested(Cell* c, const std::set<std::string>& suggestedNodes,
                                           bool ignoreSuggestedNodes) {
  PhysicalCell* pc = physOf(c);
  if (pc == nullptr) return {true, true};
  bool suggested =
      ignoreSuggestedNodes || (!pc->nodes.empty() && suggestedNodes.count(pc->nodes[0]) > 0);
  return {pc->healthy, suggested};
}

}  // namespace

std::vector<CleanShapeWorld> computeCleanShapeWorlds(
    const ChainCellList& ccl, const std::set<std::string>* suggestedNodes, size_t maxWorlds,
    int priority) {
  PickSession base;
  base.suggestedFilter = suggestedNodes;
  int top = ccl.top();
  // per top cell: the exclusion-set variants (every max independent set of
  // its bad-link graph). Links are node-local, so top cells are independent.
  std::vector<std::vector<std::vector<Cell*>>> perCell;
  for (Cell* c : ccl.at(top)) {
    auto variants = enumerateCleanExclusionVariants(c, base, maxWorlds, priority);
    if (!variants.empty()) perCell.push_back(std::move(variants));
  }
  std::vector<CleanShapeWorld> worlds;
  // shared across worlds: the unbound-cell lists, and each clean cell's
  // capacity via the O(1) counter fast path (exclusions only ever sit
  // under bad-link cells, so cells with badLinksUnder == 0 are unaffected
  // by any world's exclusion set — this keeps the per-schedule world
  // computation at O(bad-link subtrees), not O(chain), per world)
  std::map<int, std::vector<PhysicalCell*>> physByLevel;
  std::map<int, int> cleanCellBest;  // per level, max capacity among clean cells
  PickSession noExcl = base;
  for (int l = kLowestLevel; l <= top; l++) {
    int best = 0;
    for (Cell* c : ccl.at(l)) {
      auto* ph = static_cast<PhysicalCell*>(c);
      if (ph->virt == nullptr) physByLevel[l].push_back(ph);
      if (ph->badLinksUnder == 0) {
        best = std::max(best, availLeaves(c, priority, noExcl).first);
      }
    }
    cleanCellBest[l] = best;
  }
  // odometer over the per-cell variants, capped
  std::vector<size_t> idx(perCell.size(), 0);
  for (;;) {
    CleanShapeWorld w;
    PickSession session = base;
    for (size_t i = 0; i < perCell.size(); i++) {
      for (Cell* e : perCell[i][idx[i]]) {
        session.excluded.insert(e);
        w.excluded.insert(static_cast<PhysicalCell*>(e));
      }
    }
    w.priority = priority;
    for (int l = kLowestLevel; l <= top; l++) {
      int best = cleanCellBest[l];
      for (Cell* c : ccl.at(l)) {
        auto* ph = static_cast<PhysicalCell*>(c);
        if (ph->badLinksUnder > 0) {
          best = std::max(best, availLeaves(c, priority, session).first);
        }
      }
      w.caps[l] = best;
    }
    w.physByLevel = physByLevel;
    worlds.push_back(std::move(w));
    if (worlds.size() >= maxWorlds || perCell.empty()) break;
    // advance the odometer
    size_t d = 0;
    while (d < perCell.size()) {
      if (++idx[d] < perCell[d].size()) break;
      idx[d] = 0;
      d++;
    }
    if (d == perCell.size()) break;
  }
  if (worlds.empty()) worlds.push_back(CleanShapeWorld{});
  return worlds;
}


bool TopoScheduler::tryScheduleAtPriority(const std::vector<int>& sortedLeafNums, int priority,
                                          const std::set<std::string>& suggestedNodes,
                                          bool ignoreSuggestedNodes, long long minHbmBytes,
                                          bool honorLinks, const CleanShapeWorld* cleanWorld,
                                          Placement<Cell>* out,
                                          std::string* failedReason) const {
  // Build and sort the cluster view: healthy > suggested > same-priority used
  // (desc, packing) > higher-priority used (asc, stay away).
  std::vector<NodeView> cv;
  cv.reserve(viewCells_.size());
  PickSession probe;  // empty: availability before any placement
  probe.minHbm = minHbmBytes;
  if (honorLinks) probe.world = cleanWorld;
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
    // nodes stay usable for their healthy pairs/quads. In link-honoring mode
    // a node's capacity is its CLEAN capacity (largest co-placeable set with
    // no degraded xGMI link inside); the dirty retry lifts that.
    if (honorLinks && probe.world == nullptr && badLinksUnderOf(c) > 0) {
      // physical/bound view without a precomputed world: per-node analysis
      n.freeAtPriority = cleanAvailAnalysis(c, priority, probe, INT_MAX, nullptr);
    } else {
      // with a world, availLeaves is already the in-world (clean) capacity
      n.freeAtPriority = availLeaves(c, priority, probe).first;
    }
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
  session.minHbm = minHbmBytes;
  if (honorLinks) {
    session.world = cleanWorld;
    // Gang-wide link cleanliness per node: every pod of the gang on one node
    // communicates with every other (the gang's collective), so exclusions
    // are computed from the node's TOTAL gang demand before any pick.
    // With a precomputed clean-shape world the exclusions are already
    // global (world.excluded), so the per-node analysis is skipped.
    if (cleanWorld == nullptr) {
      std::map<int, int> nodeDemand;  // nodeIndex -> total gang leaves
      for (size_t i = 0; i < sortedLeafNums.size(); i++) {
        nodeDemand[pickedNodeIndices[i]] += sortedLeafNums[i];
      }
      for (auto& [nodeIndex, demand] : nodeDemand) {
        if (demand >= 2) applyLinkExclusions(cv[nodeIndex].c, demand, priority, session);
      }
    }
  }
  out->clear();
  for (size_t i = 0; i < sortedLeafNums.size(); i++) {
    int q = sortedLeafNums[i];
    Cell* node = cv[pickedNodeIndices[i]].c;
    std::vector<Cell*> leaves;
    leaves.reserve(q);
    if (session.world != nullptr && !node->isPhysical()) {
      // world mode on a virtual view: physical-mirroring descent over
      // tightest-first candidate hints; a shape no candidate can realize
      // fails the attempt (next world / dirty rung)
      bool placed = false;
      for (PhysicalCell* hint : resolveWorldHints(node, q, session)) {
        if (pickLeavesWorld(node, hint, q, priority, session, leaves)) {
          placed = true;
          break;
        }
      }
      if (!placed) {
        *failedReason = "no link-clean shape in this world";
        return false;
      }
    } else {
      pickLeaves(node, q, priority, session, leaves);
    }
    (*out)[q].push_back(std::move(leaves));
  }
  return true;
}

bool TopoScheduler::Schedule(const std::map<int, int>& podLeafCellNums, int priority,
                             const std::set<std::string>& suggestedNodes,
                             bool ignoreSuggestedNodes, Placement<Cell>* out,
                             std::string* failedReason, long long minHbmBytes,
                             const CleanShapeWorld* cleanWorld, bool honorOnly) const {
  std::vector<int> sortedLeafNums;
  for (auto& [leafNum, podNum] : podLeafCellNums) {
    for (int i = 0; i < podNum; i++) sortedLeafNums.push_back(leafNum);
  }
  // a clean-shape world is only passed when the chain carries degraded
  // links, so its presence alone demands the dirty-retry rungs of the
  // ladder (virtual views are mostly unbound and show badLinksUnderOf == 0)
  bool anyBadLinks = cleanWorld != nullptr;
  for (Cell* c : viewCells_) {
    if (anyBadLinks || badLinksUnderOf(c) > 0) {
      anyBadLinks = true;
      break;
    }
  }
  // Attempt ladder: free cells with clean links, free cells ignoring link
  // state (capacity outranks link quality), then the same two with
  // preemption. Avoiding a preemption (killing pods) outranks avoiding a
  // degraded link (slow xGMI), hence free+dirty before preempt+clean.
  // honorOnly (set when the caller iterates over several clean-shape
  // worlds) runs only the link-honoring rungs; the caller provides the
  // dirty fallback itself after every world failed.
  auto attempt = [&](int p, bool honorLinks) {
    return tryScheduleAtPriority(sortedLeafNums, p, suggestedNodes, ignoreSuggestedNodes,
                                 minHbmBytes, honorLinks, cleanWorld, out, failedReason);
  };
  // world-mode (honorOnly) attempts run ONLY at the world's own tier: the
  // caller iterates worlds built at kOpportunistic (free cells) and, if
  // those fail, worlds built at the request's priority (free +
  // preemptible); each world's availability math matches its tier.
  if (honorOnly) return attempt(cleanWorld != nullptr ? cleanWorld->priority : priority, true);
  if (attempt(kOpportunisticPriority, true)) return true;
  if (anyBadLinks && attempt(kOpportunisticPriority, false)) return true;
  if (priority > kOpportunisticPriority) {
    if (attempt(priority, true)) return true;
    if (anyBadLinks && attempt(priority, false)) return true;
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
                           sr.ignoreSuggestedNodes, &generic, failedReason, sr.hbmBytes,
                           sr.cleanWorld, sr.honorLinksOnly)) {
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
