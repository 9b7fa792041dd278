"""API constants: annotation keys, priority ranges, HTTP paths.

Parity with reference pkg/api/constants.go:34-94 (microsoft/hivedscheduler),
re-namespaced for the AMD-native scheduler. Legacy (microsoft + gpu*) keys are
accepted on read for drop-in compatibility (reference pkg/internal/utils.go:189-197).
"""

ComponentName = "hivedscheduler"
GroupName = "hivedscheduler.amd.com"

# Legacy group accepted on read so existing HiveD clients keep working.
LegacyGroupName = "hivedscheduler.microsoft.com"

# Pod opt-in: resource limit {AnnotationKeyEnable}: 1 on any container.
ResourceNamePodSchedulingEnable = GroupName + "/pod-scheduling-enable"
LegacyResourceNamePodSchedulingEnable = LegacyGroupName + "/pod-scheduling-enable"

# Request annotation (YAML PodSchedulingSpec).
AnnotationKeyPodSchedulingSpec = GroupName + "/pod-scheduling-spec"
# Decision/recovery annotation (YAML PodBindInfo).
AnnotationKeyPodBindInfo = GroupName + "/pod-bind-info"
# Runtime isolation annotation: comma-joined leaf (GPU) indices, consumed as
# ROCR_VISIBLE_DEVICES / AMD_VISIBLE_DEVICES via downward-API fieldRef.
AnnotationKeyPodLeafCellIsolation = GroupName + "/pod-leaf-cell-isolation"
# Deprecated alias (reference pkg/api/constants.go:51).
AnnotationKeyPodGpuIsolation = GroupName + "/pod-gpu-isolation"

LegacyAnnotationKeyPodSchedulingSpec = LegacyGroupName + "/pod-scheduling-spec"
LegacyAnnotationKeyPodBindInfo = LegacyGroupName + "/pod-bind-info"
LegacyAnnotationKeyPodLeafCellIsolation = LegacyGroupName + "/pod-leaf-cell-isolation"
LegacyAnnotationKeyPodGpuIsolation = LegacyGroupName + "/pod-gpu-isolation"

# Priorities (reference pkg/api/constants.go:56-63).
MaxGuaranteedPriority = 1000
MinGuaranteedPriority = 0
OpportunisticPriority = -1

# Internal-only priorities (reference pkg/algorithm/constants.go:32-35).
FreePriority = -2

# Environment variable that locates the YAML config file.
EnvNameConfigFilePath = "CONFIG"
DefaultConfigFilePath = "/hivedscheduler-config/hivedscheduler.yaml"

# HTTP paths (reference pkg/api/constants.go:72-94).
RootPath = "/"
VersionPath = RootPath + "v1"
ExtenderPath = VersionPath + "/extender"
FilterPath = ExtenderPath + "/filter"
BindPath = ExtenderPath + "/bind"
PreemptPath = ExtenderPath + "/preempt"
InspectPath = VersionPath + "/inspect"
AffinityGroupsPath = InspectPath + "/affinitygroups/"
ClusterStatusPath = InspectPath + "/clusterstatus"
PhysicalClusterPath = ClusterStatusPath + "/physicalcluster"
VirtualClustersPath = ClusterStatusPath + "/virtualclusters/"
MetricsPath = "/metrics"

DefaultWebServerAddress = ":9096"

# Built-in MI355X (CDNA4) cell chain type names; see topo/mi355x.py.
MI355XLeafCellType = "MI355X"
MI355XPairCellType = "MI355X-PAIR"
MI355XQuadCellType = "MI355X-QUAD"
MI355XNodeCellType = "MI355X-NODE"

# MI355X hardware facts attached to cells (first-class cell attributes).
MI355XHBMBytes = 288 * 1024**3  # 288 GB HBM3E per GPU
MI355XXGMILinksPerGPU = 7       # fully-connected 8-GPU node, 1 link per peer
MI355XXGMILinkGBps = 153.0      # per-link peak, one direction
