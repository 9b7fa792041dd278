// pybind11 bindings for the scheduler core. Dict-in / dict-out at the API
// boundary (wire formats match hivedscheduler_amd.api.types); all hot-path
// work happens in C++.
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <functional>

#include "core.hpp"

namespace py = pybind11;
using namespace hived;

namespace {

std::string dstr(const py::dict& d, const char* key, const std::string& def = "") {
  if (!d.contains(key)) return def;
  py::object v = d[key];
  if (v.is_none()) return def;
  return py::cast<std::string>(py::str(v));
}
long long dint(const py::dict& d, const char* key, long long def = 0) {
  if (!d.contains(key)) return def;
  py::object v = d[key];
  if (v.is_none()) return def;
  return py::cast<long long>(v);
}
bool dbool(const py::dict& d, const char* key, bool def = false) {
  if (!d.contains(key)) return def;
  py::object v = d[key];
  if (v.is_none()) return def;
  return py::cast<bool>(v);
}

ClusterSpec parseClusterSpec(const py::dict& d) {
  ClusterSpec spec;
  if (d.contains("cellTypes")) {
    for (auto item : py::cast<py::dict>(d["cellTypes"])) {
      py::dict td = py::cast<py::dict>(item.second);
      CellTypeSpec ct;
      ct.child = dstr(td, "childCellType");
      ct.childCount = static_cast<int>(dint(td, "childCellNumber"));
      ct.isNode = dbool(td, "isNodeLevel");
      spec.cellTypes[py::cast<std::string>(item.first)] = ct;
    }
  }
  std::function<PhysCellSpec(const py::dict&)> parseCell = [&](const py::dict& cd) {
    PhysCellSpec c;
    c.type = dstr(cd, "cellType");
    c.address = dstr(cd, "cellAddress");
    c.pinnedId = dstr(cd, "pinnedCellId");
    if (cd.contains("hbmBytes") && !cd["hbmBytes"].is_none()) {
      c.hbmBytes = py::cast<long long>(cd["hbmBytes"]);
    }
    // discovery-measured xGMI link table (node-level entries):
    // [{a, b, gbps, healthy}, ...]
    if (cd.contains("xgmiLinks") && !cd["xgmiLinks"].is_none()) {
      for (auto lo : py::cast<py::list>(cd["xgmiLinks"])) {
        py::dict ld = py::cast<py::dict>(lo);
        double gbps = ld.contains("gbps") && !ld["gbps"].is_none() ? py::cast<double>(ld["gbps"]) : 0.0;
        c.xgmiLinks.emplace_back(static_cast<int>(dint(ld, "a")), static_cast<int>(dint(ld, "b")),
                                 gbps, dbool(ld, "healthy", true));
      }
    }
    if (cd.contains("cellChildren") && !cd["cellChildren"].is_none()) {
      for (auto child : py::cast<py::list>(cd["cellChildren"])) {
        c.children.push_back(parseCell(py::cast<py::dict>(child)));
      }
    }
    return c;
  };
  if (d.contains("physicalCells")) {
    for (auto cd : py::cast<py::list>(d["physicalCells"])) {
      spec.physicalCells.push_back(parseCell(py::cast<py::dict>(cd)));
    }
  }
  if (d.contains("virtualClusters")) {
    for (auto item : py::cast<py::dict>(d["virtualClusters"])) {
      py::dict vd = py::cast<py::dict>(item.second);
      VCSpec vc;
      if (vd.contains("virtualCells") && !vd["virtualCells"].is_none()) {
        for (auto vcell : py::cast<py::list>(vd["virtualCells"])) {
          py::dict vcd = py::cast<py::dict>(vcell);
          VirtCellSpec vs;
          vs.typePath = dstr(vcd, "cellType");
          vs.number = static_cast<int>(dint(vcd, "cellNumber"));
          vc.virtualCells.push_back(vs);
        }
      }
      if (vd.contains("pinnedCells") && !vd["pinnedCells"].is_none()) {
        for (auto pcell : py::cast<py::list>(vd["pinnedCells"])) {
          py::dict pcd = py::cast<py::dict>(pcell);
          vc.pinnedIds.push_back(dstr(pcd, "pinnedCellId"));
        }
      }
      spec.virtualClusters[py::cast<std::string>(item.first)] = vc;
    }
  }
  return spec;
}

PodSpec parsePodSpec(const py::dict& d) {
  PodSpec s;
  s.vc = dstr(d, "virtualCluster");
  s.priority = static_cast<int>(dint(d, "priority"));
  s.pinnedCellId = dstr(d, "pinnedCellId");
  s.leafCellType = dstr(d, "leafCellType");
  s.leafCellNumber = static_cast<int>(dint(d, "leafCellNumber"));
  s.gangReleaseEnable = dbool(d, "gangReleaseEnable");
  s.lazyPreemptionEnable = dbool(d, "lazyPreemptionEnable");
  s.ignoreK8sSuggestedNodes = dbool(d, "ignoreK8sSuggestedNodes", true);
  if (d.contains("hbmBytesPerCell") && !d["hbmBytesPerCell"].is_none()) {
    s.hbmBytesPerCell = py::cast<long long>(d["hbmBytesPerCell"]);
    if (s.hbmBytesPerCell < 0) {
      throw HivedError::BadRequest("hbmBytesPerCell must be non-negative");
    }
  }
  if (d.contains("affinityGroup") && !d["affinityGroup"].is_none()) {
    py::dict ag = py::cast<py::dict>(d["affinityGroup"]);
    s.groupName = dstr(ag, "name");
    if (ag.contains("members") && !ag["members"].is_none()) {
      for (auto m : py::cast<py::list>(ag["members"])) {
        py::dict md = py::cast<py::dict>(m);
        s.groupPodNums[static_cast<int>(dint(md, "leafCellNumber"))] +=
            static_cast<int>(dint(md, "podNumber"));
      }
    }
  }
  if (s.groupPodNums.empty() && s.leafCellNumber > 0) {
    s.groupPodNums[s.leafCellNumber] = 1;  // singleton group default
  }
  return s;
}

BindInfo parseBindInfo(const py::dict& d) {
  BindInfo info;
  info.node = dstr(d, "node");
  info.chain = dstr(d, "cellChain");
  if (d.contains("leafCellIsolation") && !d["leafCellIsolation"].is_none()) {
    for (auto i : py::cast<py::list>(d["leafCellIsolation"])) {
      info.isolation.push_back(py::cast<int>(i));
    }
  }
  if (d.contains("affinityGroupBindInfo") && !d["affinityGroupBindInfo"].is_none()) {
    for (auto mbiObj : py::cast<py::list>(d["affinityGroupBindInfo"])) {
      py::dict mbiD = py::cast<py::dict>(mbiObj);
      std::vector<PodPlacementInfo> placements;
      if (mbiD.contains("podPlacements") && !mbiD["podPlacements"].is_none()) {
        for (auto plObj : py::cast<py::list>(mbiD["podPlacements"])) {
          py::dict plD = py::cast<py::dict>(plObj);
          PodPlacementInfo pl;
          pl.node = dstr(plD, "physicalNode");
          if (plD.contains("physicalLeafCellIndices") && !plD["physicalLeafCellIndices"].is_none()) {
            for (auto i : py::cast<py::list>(plD["physicalLeafCellIndices"])) {
              pl.leafIndices.push_back(py::cast<int>(i));
            }
          }
          if (plD.contains("preassignedCellTypes") && !plD["preassignedCellTypes"].is_none()) {
            for (auto t : py::cast<py::list>(plD["preassignedCellTypes"])) {
              pl.preassignedTypes.push_back(t.is_none() ? "" : py::cast<std::string>(py::str(t)));
            }
          }
          placements.push_back(std::move(pl));
        }
      }
      info.memberBindInfo.push_back(std::move(placements));
    }
  }
  return info;
}

py::dict bindInfoToDict(const BindInfo& info) {
  py::dict d;
  d["node"] = info.node;
  d["leafCellIsolation"] = info.isolation;
  d["cellChain"] = info.chain;
  py::list mbis;
  for (auto& mbi : info.memberBindInfo) {
    py::list placements;
    for (auto& pl : mbi) {
      py::dict pd;
      pd["physicalNode"] = pl.node;
      pd["physicalLeafCellIndices"] = pl.leafIndices;
      pd["preassignedCellTypes"] = pl.preassignedTypes;
      placements.append(pd);
    }
    py::dict md;
    md["podPlacements"] = placements;
    mbis.append(md);
  }
  d["affinityGroupBindInfo"] = mbis;
  return d;
}

py::dict physicalCellStatus(PhysicalCell* c, bool withChildren = true) {
  py::dict d;
  d["cellType"] = c->typeName;
  d["cellAddress"] = c->address;
  d["isNodeLevel"] = c->isNodeLevel;
  d["cellState"] = to_string(c->state);
  d["cellHealthiness"] = c->healthy ? "Healthy" : "Bad";
  d["cellPriority"] = c->priority;
  if (!c->otVC.empty()) d["vc"] = c->otVC;
  if (c->virt != nullptr) {
    d["vc"] = c->virt->vc;
    d["virtualCell"] = c->virt->address;
  }
  if (c->level == kLowestLevel) {
    d["physicalNode"] = c->nodes.empty() ? "" : c->nodes[0];
    d["leafCellIndex"] = c->leafIndices.empty() ? -1 : c->leafIndices[0];
    d["hbmBytes"] = c->hbmBytes;
  } else if (c->badLinksUnder > 0) {
    // first-class xGMI link state: degraded links under this pair/quad/node
    // (the cell stays Healthy — its GPUs work — but multi-GPU placements
    // avoid co-placing the degraded link's endpoints)
    d["badXgmiLinksUnder"] = c->badLinksUnder;
  }
  if (withChildren && !c->children.empty()) {
    py::list children;
    for (Cell* child : c->children) {
      children.append(physicalCellStatus(static_cast<PhysicalCell*>(child)));
    }
    d["cellChildren"] = children;
  }
  return d;
}

py::dict virtualCellStatus(VirtualCell* c, bool withChildren = true) {
  py::dict d;
  d["cellType"] = c->typeName;
  d["cellAddress"] = c->address;
  CState state = c->phys != nullptr ? c->phys->state
                                    : (c->priority > kFreePriority ? CState::Used : CState::Free);
  d["cellState"] = to_string(state);
  d["cellHealthiness"] = (c->phys == nullptr || c->phys->healthy) ? "Healthy" : "Bad";
  d["cellPriority"] = c->priority;
  if (c->phys != nullptr) d["physicalCell"] = c->phys->address;
  if (withChildren && !c->children.empty()) {
    py::list children;
    for (Cell* child : c->children) {
      children.append(virtualCellStatus(static_cast<VirtualCell*>(child)));
    }
    d["cellChildren"] = children;
  }
  return d;
}

py::dict groupToDict(const Group* g) {
  py::dict d;
  d["name"] = g->name;
  d["vc"] = g->vc;
  d["priority"] = g->priority;
  d["state"] = to_string(g->state);
  if (g->lazyStatus.has_value()) {
    py::dict lp;
    lp["preemptor"] = g->lazyStatus->preemptor;
    lp["preemptionTime"] = g->lazyStatus->preemptionTime;
    d["lazyPreemptionStatus"] = lp;
  } else {
    d["lazyPreemptionStatus"] = py::none();
  }
  // physical placement: node -> leaf cell indices
  py::dict physD;
  for (auto& [leafNum, pods] : g->physPlacement) {
    (void)leafNum;
    for (auto& pod : pods) {
      for (PhysicalCell* c : pod) {
        if (c == nullptr) continue;
        std::string node = c->nodes.empty() ? "" : c->nodes[0];
        if (!physD.contains(py::str(node))) physD[py::str(node)] = py::list();
        py::cast<py::list>(physD[py::str(node)]).append(c->leafIndices[0]);
      }
    }
  }
  d["physicalPlacement"] = physD;
  // virtual placement: preassigned cell address -> leaf cell addresses
  py::dict virtD;
  if (g->hasVirtualPlacement) {
    for (auto& [leafNum, pods] : g->virtPlacement) {
      (void)leafNum;
      for (auto& pod : pods) {
        for (VirtualCell* v : pod) {
          if (v == nullptr) continue;
          std::string pre = v->preassigned->address;
          if (!virtD.contains(py::str(pre))) virtD[py::str(pre)] = py::list();
          py::cast<py::list>(virtD[py::str(pre)]).append(v->address);
        }
      }
    }
  }
  d["virtualPlacement"] = virtD;
  py::list allocated;
  for (auto& [ln, pods] : g->allocatedPods) {
    (void)ln;
    for (auto& p : pods) {
      if (p.present) allocated.append(p.key);
    }
  }
  d["allocatedPods"] = allocated;
  py::list preempting;
  for (auto& k : g->preemptingPods) preempting.append(k);
  d["preemptingPods"] = preempting;
  return d;
}

class PyHivedCore {
 public:
  explicit PyHivedCore(const py::dict& spec) : core_(parseClusterSpec(spec)) {}

  void setNodeHealthy(const std::string& node, bool healthy) { core_.setNodeHealthy(node, healthy); }
  void setLeafCellHealthy(const std::string& node, int leafIndex, bool healthy) {
    core_.setLeafCellHealthy(node, leafIndex, healthy);
  }
  void setXgmiLinkHealthy(const std::string& node, int a, int b, bool healthy, double gbps) {
    core_.setXgmiLinkHealthy(node, a, b, healthy, gbps);
  }
  py::list xgmiLinks(const std::string& node) const {
    py::list out;
    for (auto& [a, b, gbps, healthy] : core_.xgmiLinks(node)) {
      py::dict d;
      d["a"] = a;
      d["b"] = b;
      d["gbps"] = gbps;
      d["healthy"] = healthy;
      out.append(d);
    }
    return out;
  }
  std::vector<std::string> allNodes() const { return core_.allNodes(); }
  std::vector<std::string> badNodes() const {
    auto s = core_.badNodes();
    return {s.begin(), s.end()};
  }

  py::dict schedule(const py::dict& spec, const std::string& podKey,
                    const std::vector<std::string>& suggestedNodes, const std::string& phase) {
    PodSpec s = parsePodSpec(spec);
    std::set<std::string> suggested(suggestedNodes.begin(), suggestedNodes.end());
    Phase ph = (phase == "Preempting") ? Phase::Preempting : Phase::Filtering;
    ScheduleResult r = core_.schedule(s, podKey, suggested, ph);
    py::dict d;
    switch (r.kind) {
      case ScheduleResult::Kind::Bind:
        d["kind"] = "bind";
        d["bindInfo"] = bindInfoToDict(r.bindInfo);
        break;
      case ScheduleResult::Kind::Preempt:
        d["kind"] = "preempt";
        d["victimNode"] = r.victimNode;
        d["victimPodKeys"] = r.victimPodKeys;
        break;
      case ScheduleResult::Kind::Wait:
        d["kind"] = "wait";
        d["reason"] = r.waitReason;
        break;
    }
    return d;
  }

  void deleteUnallocatedPod(const py::dict& spec, const std::string& podKey) {
    core_.deleteUnallocatedPod(parsePodSpec(spec), podKey);
  }
  void addAllocatedPod(const py::dict& spec, const py::dict& bindInfo, const std::string& podKey) {
    core_.addAllocatedPod(parsePodSpec(spec), parseBindInfo(bindInfo), podKey);
  }
  void deleteAllocatedPod(const py::dict& spec, const py::dict& bindInfo,
                          const std::string& podKey) {
    core_.deleteAllocatedPod(parsePodSpec(spec), parseBindInfo(bindInfo), podKey);
  }

  py::list getAllAffinityGroups() const {
    py::list out;
    for (auto& [name, g] : core_.groups()) {
      (void)name;
      out.append(groupToDict(g.get()));
    }
    return out;
  }

  py::dict getAffinityGroup(const std::string& name) const {
    auto it = core_.groups().find(name);
    if (it == core_.groups().end()) {
      throw HivedError::BadRequest("Affinity group " + name +
                                   " does not exist since it is not allocated or preempting");
    }
    return groupToDict(it->second.get());
  }

  py::dict getClusterStatus() {
    py::dict d;
    d["physicalCluster"] = getPhysicalClusterStatus();
    d["virtualClusters"] = getAllVirtualClustersStatus();
    return d;
  }

  py::list getPhysicalClusterStatus() {
    py::list out;
    for (auto& [chain, ccl] : core_.fullCellList_) {
      (void)chain;
      int top = ccl.top();
      for (Cell* c : ccl.at(top)) {
        out.append(physicalCellStatus(static_cast<PhysicalCell*>(c)));
      }
    }
    return out;
  }

  py::dict getAllVirtualClustersStatus() {
    py::dict out;
    for (auto& [vcName, vcs] : core_.vcSchedulers_) {
      out[py::str(vcName)] = getVirtualClusterStatus(vcName);
    }
    return out;
  }

  py::list getVirtualClusterStatus(const std::string& vcName) {
    auto it = core_.vcSchedulers_.find(vcName);
    if (it == core_.vcSchedulers_.end()) {
      throw HivedError::NotFound("VC " + vcName + " not found");
    }
    py::list out;
    for (auto& [chain, ccl] : it->second.nonPinnedPreassigned) {
      (void)chain;
      for (int l = 1; l <= ccl.top(); l++) {
        for (Cell* c : ccl.at(l)) {
          out.append(virtualCellStatus(static_cast<VirtualCell*>(c)));
        }
      }
    }
    for (auto& [pid, ccl] : it->second.pinned) {
      (void)pid;
      int top = ccl.top();
      for (Cell* c : ccl.at(top)) {
        out.append(virtualCellStatus(static_cast<VirtualCell*>(c)));
      }
    }
    // opportunistic cells used by this VC, exposed as fake virtual cells
    for (auto& [chain, ccl] : core_.fullCellList_) {
      (void)chain;
      for (Cell* c : ccl.at(kLowestLevel)) {
        auto* pc = static_cast<PhysicalCell*>(c);
        if (pc->otVC == vcName) {
          py::dict d;
          d["cellType"] = pc->typeName;
          d["cellAddress"] = pc->address + "-opp";
          d["cellState"] = to_string(CState::Used);
          d["cellHealthiness"] = pc->healthy ? "Healthy" : "Bad";
          d["cellPriority"] = kOpportunisticPriority;
          d["physicalCell"] = pc->address;
          out.append(d);
        }
      }
    }
    return out;
  }

  long long scheduleCount() const { return core_.scheduleCount_; }
  void checkInvariants() const { hived::checkInvariants(core_); }

  py::dict debugCounters() const {
    // safety-accounting introspection for tests/debugging: per chain+level
    // totalLeft / allVCFree / per-VC free / bad-free count
    py::dict d;
    for (auto& [chain, perLevel] : core_.totalLeftCellNum_) {
      py::dict cd;
      for (auto& [level, left] : perLevel) {
        py::dict ld;
        ld["totalLeft"] = left;
        int avf = 0;
        auto ai = core_.allVCFreeCellNum_.find(chain);
        if (ai != core_.allVCFreeCellNum_.end()) {
          auto li = ai->second.find(level);
          if (li != ai->second.end()) avf = li->second;
        }
        ld["allVCFree"] = avf;
        py::dict vd;
        for (auto& [vcn, perChain] : core_.vcFreeCellNum_) {
          auto ci = perChain.find(chain);
          if (ci == perChain.end()) continue;
          auto li = ci->second.find(level);
          if (li != ci->second.end() && li->second != 0) vd[py::str(vcn)] = li->second;
        }
        ld["vcFree"] = vd;
        auto fi = core_.freeCellList_.find(chain);
        if (fi != core_.freeCellList_.end() && level <= fi->second.top()) {
          py::list fl;
          for (auto* fc : fi->second.at(level)) fl.append(fc->address);
          ld["freeList"] = fl;
        }
        auto di = core_.allVCDoomedBadCellNum_.find(chain);
        if (di != core_.allVCDoomedBadCellNum_.end()) {
          auto li2 = di->second.find(level);
          if (li2 != di->second.end() && li2->second != 0) ld["doomed"] = li2->second;
        }
        auto bi = core_.badFreeCells_.find(chain);
        if (bi != core_.badFreeCells_.end() && level <= bi->second.top()) {
          ld["badFree"] = static_cast<int>(bi->second.at(level).size());
        }
        cd[py::int_(level)] = ld;
      }
      d[py::str(chain)] = cd;
    }
    return d;
  }

 private:
  HivedCore core_;
};

}  // namespace

PYBIND11_MODULE(hivedcore, m) {
  m.doc() = "MI355X-native gang scheduler core (C++)";

  static py::exception<HivedError> exc(m, "CoreError");
  py::register_exception_translator([](std::exception_ptr p) {
    try {
      if (p) std::rethrow_exception(p);
    } catch (const HivedError& e) {
      py::object pyExc = exc;
      py::object inst = pyExc(e.what());
      inst.attr("code") = e.code;
      PyErr_SetObject(pyExc.ptr(), inst.ptr());
    }
  });

  py::class_<PyHivedCore>(m, "HivedCore")
      .def(py::init<const py::dict&>(), py::arg("spec"))
      .def("set_node_healthy", &PyHivedCore::setNodeHealthy, py::arg("node"), py::arg("healthy"))
      .def("set_xgmi_link_healthy", &PyHivedCore::setXgmiLinkHealthy, py::arg("node"),
           py::arg("a"), py::arg("b"), py::arg("healthy"), py::arg("gbps") = 0.0)
      .def("xgmi_links", &PyHivedCore::xgmiLinks, py::arg("node"))
      .def("set_leaf_cell_healthy", &PyHivedCore::setLeafCellHealthy, py::arg("node"),
           py::arg("leaf_index"), py::arg("healthy"))
      .def("all_nodes", &PyHivedCore::allNodes)
      .def("bad_nodes", &PyHivedCore::badNodes)
      .def("schedule", &PyHivedCore::schedule, py::arg("pod_spec"), py::arg("pod_key"),
           py::arg("suggested_nodes"), py::arg("phase"))
      .def("delete_unallocated_pod", &PyHivedCore::deleteUnallocatedPod, py::arg("pod_spec"),
           py::arg("pod_key"))
      .def("add_allocated_pod", &PyHivedCore::addAllocatedPod, py::arg("pod_spec"),
           py::arg("bind_info"), py::arg("pod_key"))
      .def("delete_allocated_pod", &PyHivedCore::deleteAllocatedPod, py::arg("pod_spec"),
           py::arg("bind_info"), py::arg("pod_key"))
      .def("get_all_affinity_groups", &PyHivedCore::getAllAffinityGroups)
      .def("get_affinity_group", &PyHivedCore::getAffinityGroup, py::arg("name"))
      .def("get_cluster_status", &PyHivedCore::getClusterStatus)
      .def("get_physical_cluster_status", &PyHivedCore::getPhysicalClusterStatus)
      .def("get_all_virtual_clusters_status", &PyHivedCore::getAllVirtualClustersStatus)
      .def("get_virtual_cluster_status", &PyHivedCore::getVirtualClusterStatus, py::arg("vc"))
      .def("schedule_count", &PyHivedCore::scheduleCount)
      .def("check_invariants", &PyHivedCore::checkInvariants)
      .def("debug_counters", &PyHivedCore::debugCounters);
}
