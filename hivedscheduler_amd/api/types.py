"""API types: cluster spec, pod scheduling spec, bind info, inspect DTOs.

Parity with reference pkg/api/types.go:42-273. Implemented as plain dataclasses
with YAML-dict (de)serialization helpers; field names keep the reference's
camelCase wire format so existing HiveD clients/configs parse unchanged.
"""
from __future__ import annotations

import dataclasses
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from . import constants


class WebServerError(Exception):
    """Error carrying an HTTP status code (reference pkg/common's panic protocol)."""

    def __init__(self, code: int, message: str):
        super().__init__(message)
        self.code = code
        self.message = message

    @staticmethod
    def bad_request(message: str) -> "WebServerError":
        return WebServerError(400, message)

    @staticmethod
    def not_found(message: str) -> "WebServerError":
        return WebServerError(404, message)


# ---------------------------------------------------------------------------
# Cluster spec (reference pkg/api/types.go:42-76)
# ---------------------------------------------------------------------------

CellType = str  # e.g. "MI355X-NODE" or hierarchical path "POOL.MI355X-NODE.MI355X-QUAD"


@dataclass
class CellTypeSpec:
    """One element of the cellTypes forest (reference pkg/api/types.go:44-49)."""

    childCellType: Optional[str] = None
    childCellNumber: int = 0
    isNodeLevel: bool = False

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "CellTypeSpec":
        return CellTypeSpec(
            childCellType=d.get("childCellType"),
            childCellNumber=int(d.get("childCellNumber") or 0),
            isNodeLevel=bool(d.get("isNodeLevel", False)),
        )


@dataclass
class PhysicalCellSpec:
    """A physical cell instance (reference pkg/api/types.go:51-58).

    cellAddress: node name at node level; device index below node level.
    """

    cellType: CellType = ""
    cellAddress: str = ""
    pinnedCellId: str = ""
    # MI355X extensions (rocm-topo-discover output): measured per-GPU HBM
    # bytes (leaf entries) and the measured xGMI link table
    # [{a, b, gbps, healthy}] (node-level entries)
    hbmBytes: int = 0
    xgmiLinks: List[Dict[str, Any]] = field(default_factory=list)
    cellChildren: List["PhysicalCellSpec"] = field(default_factory=list)

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "PhysicalCellSpec":
        return PhysicalCellSpec(
            cellType=str(d.get("cellType", "") or ""),
            cellAddress=str(d.get("cellAddress", "") if d.get("cellAddress") is not None else ""),
            pinnedCellId=str(d.get("pinnedCellId", "") or ""),
            hbmBytes=int(d.get("hbmBytes") or 0),
            xgmiLinks=list(d.get("xgmiLinks") or []),
            cellChildren=[PhysicalCellSpec.from_dict(c) for c in (d.get("cellChildren") or [])],
        )

    def to_dict(self) -> Dict[str, Any]:
        out: Dict[str, Any] = {"cellType": self.cellType, "cellAddress": self.cellAddress}
        if self.pinnedCellId:
            out["pinnedCellId"] = self.pinnedCellId
        if self.hbmBytes:
            out["hbmBytes"] = self.hbmBytes
        if self.xgmiLinks:
            out["xgmiLinks"] = [dict(l) for l in self.xgmiLinks]
        if self.cellChildren:
            out["cellChildren"] = [c.to_dict() for c in self.cellChildren]
        return out


@dataclass
class PhysicalClusterSpec:
    cellTypes: Dict[str, CellTypeSpec] = field(default_factory=dict)
    physicalCells: List[PhysicalCellSpec] = field(default_factory=list)

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "PhysicalClusterSpec":
        return PhysicalClusterSpec(
            cellTypes={k: CellTypeSpec.from_dict(v or {}) for k, v in (d.get("cellTypes") or {}).items()},
            physicalCells=[PhysicalCellSpec.from_dict(c) for c in (d.get("physicalCells") or [])],
        )


@dataclass
class VirtualCellSpec:
    """VC quota entry: cellNumber cells of hierarchical cellType path."""

    cellType: CellType = ""
    cellNumber: int = 0

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "VirtualCellSpec":
        return VirtualCellSpec(cellType=str(d.get("cellType", "")), cellNumber=int(d.get("cellNumber") or 0))


@dataclass
class PinnedCellSpec:
    pinnedCellId: str = ""

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "PinnedCellSpec":
        return PinnedCellSpec(pinnedCellId=str(d.get("pinnedCellId", "")))


@dataclass
class VirtualClusterSpec:
    virtualCells: List[VirtualCellSpec] = field(default_factory=list)
    pinnedCells: List[PinnedCellSpec] = field(default_factory=list)

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "VirtualClusterSpec":
        return VirtualClusterSpec(
            virtualCells=[VirtualCellSpec.from_dict(c) for c in (d.get("virtualCells") or [])],
            pinnedCells=[PinnedCellSpec.from_dict(c) for c in (d.get("pinnedCells") or [])],
        )


# ---------------------------------------------------------------------------
# Pod scheduling request (reference pkg/api/types.go:78-98)
# ---------------------------------------------------------------------------


@dataclass
class AffinityGroupMemberSpec:
    podNumber: int = 0
    leafCellNumber: int = 0

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "AffinityGroupMemberSpec":
        # Accept legacy gpuNumber alias.
        leaf = d.get("leafCellNumber", d.get("gpuNumber", 0))
        return AffinityGroupMemberSpec(podNumber=int(d.get("podNumber") or 0), leafCellNumber=int(leaf or 0))

    def to_dict(self) -> Dict[str, Any]:
        return {"podNumber": self.podNumber, "leafCellNumber": self.leafCellNumber}


@dataclass
class AffinityGroupSpec:
    name: str = ""
    members: List[AffinityGroupMemberSpec] = field(default_factory=list)

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "AffinityGroupSpec":
        return AffinityGroupSpec(
            name=str(d.get("name", "")),
            members=[AffinityGroupMemberSpec.from_dict(m) for m in (d.get("members") or [])],
        )

    def to_dict(self) -> Dict[str, Any]:
        return {"name": self.name, "members": [m.to_dict() for m in self.members]}


@dataclass
class PodSchedulingSpec:
    virtualCluster: str = ""
    priority: int = 0
    pinnedCellId: str = ""
    leafCellType: str = ""
    leafCellNumber: int = 0
    gangReleaseEnable: bool = False
    lazyPreemptionEnable: bool = False
    ignoreK8sSuggestedNodes: bool = True
    # MI355X extension: minimum measured HBM per leaf cell in bytes (0 = any).
    # Leaves whose discovered/agent-measured capacity falls short (a GPU
    # reporting < 288 GB is sick) are avoided for this request.
    hbmBytesPerCell: int = 0
    affinityGroup: Optional[AffinityGroupSpec] = None

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "PodSchedulingSpec":
        # Legacy gpu* key conversion (reference pkg/internal/utils.go:189-197).
        leaf_type = d.get("leafCellType", d.get("gpuType", "")) or ""
        leaf_num = d.get("leafCellNumber", d.get("gpuNumber", 0)) or 0
        ag = d.get("affinityGroup")
        return PodSchedulingSpec(
            virtualCluster=str(d.get("virtualCluster", "") or ""),
            priority=int(d.get("priority") or 0),
            pinnedCellId=str(d.get("pinnedCellId", "") or ""),
            leafCellType=str(leaf_type),
            leafCellNumber=int(leaf_num),
            gangReleaseEnable=bool(d.get("gangReleaseEnable", False)),
            lazyPreemptionEnable=bool(d.get("lazyPreemptionEnable", False)),
            ignoreK8sSuggestedNodes=bool(d.get("ignoreK8sSuggestedNodes", True)),
            hbmBytesPerCell=int(d.get("hbmBytesPerCell") or 0),
            affinityGroup=AffinityGroupSpec.from_dict(ag) if ag else None,
        )

    def to_dict(self) -> Dict[str, Any]:
        out: Dict[str, Any] = {
            "virtualCluster": self.virtualCluster,
            "priority": self.priority,
            "leafCellNumber": self.leafCellNumber,
            "gangReleaseEnable": self.gangReleaseEnable,
            "lazyPreemptionEnable": self.lazyPreemptionEnable,
            "ignoreK8sSuggestedNodes": self.ignoreK8sSuggestedNodes,
        }
        if self.hbmBytesPerCell:
            out["hbmBytesPerCell"] = self.hbmBytesPerCell
        if self.pinnedCellId:
            out["pinnedCellId"] = self.pinnedCellId
        if self.leafCellType:
            out["leafCellType"] = self.leafCellType
        if self.affinityGroup is not None:
            out["affinityGroup"] = self.affinityGroup.to_dict()
        return out


# ---------------------------------------------------------------------------
# Bind decision / recovery (reference pkg/api/types.go:101-118)
# ---------------------------------------------------------------------------


@dataclass
class PodPlacementInfo:
    physicalNode: str = ""
    physicalLeafCellIndices: List[int] = field(default_factory=list)
    # Preassigned (top-level virtual) cell type per leaf cell; empty for
    # opportunistic pods. Needed for recovery mapping physical -> virtual.
    preassignedCellTypes: List[str] = field(default_factory=list)

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "PodPlacementInfo":
        return PodPlacementInfo(
            physicalNode=str(d.get("physicalNode", "")),
            physicalLeafCellIndices=[int(i) for i in (d.get("physicalLeafCellIndices") or d.get("physicalGpuIndices") or [])],
            preassignedCellTypes=[str(t) for t in (d.get("preassignedCellTypes") or [])],
        )

    def to_dict(self) -> Dict[str, Any]:
        return {
            "physicalNode": self.physicalNode,
            "physicalLeafCellIndices": list(self.physicalLeafCellIndices),
            "preassignedCellTypes": list(self.preassignedCellTypes),
        }


@dataclass
class AffinityGroupMemberBindInfo:
    podPlacements: List[PodPlacementInfo] = field(default_factory=list)

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "AffinityGroupMemberBindInfo":
        return AffinityGroupMemberBindInfo(
            podPlacements=[PodPlacementInfo.from_dict(p) for p in (d.get("podPlacements") or [])]
        )

    def to_dict(self) -> Dict[str, Any]:
        return {"podPlacements": [p.to_dict() for p in self.podPlacements]}


@dataclass
class PodBindInfo:
    """Written into every bound pod's annotation; the full group placement is
    replicated into EVERY member pod so any single surviving pod can restore
    the whole group (reference pkg/api/types.go:101-118)."""

    node: str = ""
    leafCellIsolation: List[int] = field(default_factory=list)
    cellChain: str = ""
    affinityGroupBindInfo: List[AffinityGroupMemberBindInfo] = field(default_factory=list)

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "PodBindInfo":
        return PodBindInfo(
            node=str(d.get("node", "")),
            leafCellIsolation=[int(i) for i in (d.get("leafCellIsolation") or d.get("gpuIsolation") or [])],
            cellChain=str(d.get("cellChain", "")),
            affinityGroupBindInfo=[
                AffinityGroupMemberBindInfo.from_dict(m) for m in (d.get("affinityGroupBindInfo") or [])
            ],
        )

    def to_dict(self) -> Dict[str, Any]:
        return {
            "node": self.node,
            "leafCellIsolation": list(self.leafCellIsolation),
            "cellChain": self.cellChain,
            "affinityGroupBindInfo": [m.to_dict() for m in self.affinityGroupBindInfo],
        }


# ---------------------------------------------------------------------------
# Config (reference pkg/api/config.go:39-118)
# ---------------------------------------------------------------------------


@dataclass
class Config:
    kubeApiServerAddress: Optional[str] = None
    kubeConfigFilePath: Optional[str] = None
    webServerAddress: str = constants.DefaultWebServerAddress
    forcePodBindThreshold: int = 3
    waitingPodSchedulingBlockMilliSec: int = 0
    physicalCluster: PhysicalClusterSpec = field(default_factory=PhysicalClusterSpec)
    virtualClusters: Dict[str, VirtualClusterSpec] = field(default_factory=dict)


def dataclass_to_plain(obj: Any) -> Any:
    """Recursively convert dataclasses to plain YAML/JSON-serializable values."""
    if dataclasses.is_dataclass(obj) and not isinstance(obj, type):
        return {k: dataclass_to_plain(v) for k, v in dataclasses.asdict(obj).items()}
    if isinstance(obj, dict):
        return {k: dataclass_to_plain(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return [dataclass_to_plain(v) for v in obj]
    return obj
