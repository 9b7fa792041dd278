"""Built-in CDNA4 cell-type chain for MI355X nodes.

The reference makes admins hand-transcribe `nvidia-smi topo --matrix` into
cellTypes YAML (reference doc/user-manual.md:44-72). Here the MI355X chain is a
first-class built-in, because CDNA4 node topology is fixed:

    MI355X (leaf, 288 GB HBM3E)
      -> MI355X-PAIR  (xGMI-adjacent pair)
      -> MI355X-QUAD  (4-GPU half-mesh)
      -> MI355X-NODE  (8-GPU fully-connected node: 7 xGMI links/GPU x ~153 GB/s)
      -> k-MI355X-NODE (rack/pool groupings, k configurable)

On an 8-GPU MI355X node every GPU pair is directly connected, so intra-node
"affinity" collapses to pair/quad alignment; the pair/quad levels still matter
because (a) they are the units VCs buy, (b) buddy allocation on them prevents
fragmentation, and (c) an unhealthy xGMI link degrades a *pair/quad* cell, not
just a leaf.
"""
from __future__ import annotations

from typing import Dict, Iterable, List

from ..api import constants
from ..api.types import CellTypeSpec, PhysicalCellSpec


def mi355x_cell_types(pool_sizes: Iterable[int] = (2, 4)) -> Dict[str, CellTypeSpec]:
    """Return the built-in cellTypes forest for MI355X."""
    types: Dict[str, CellTypeSpec] = {
        constants.MI355XPairCellType: CellTypeSpec(
            childCellType=constants.MI355XLeafCellType, childCellNumber=2
        ),
        constants.MI355XQuadCellType: CellTypeSpec(
            childCellType=constants.MI355XPairCellType, childCellNumber=2
        ),
        constants.MI355XNodeCellType: CellTypeSpec(
            childCellType=constants.MI355XQuadCellType, childCellNumber=2, isNodeLevel=True
        ),
    }
    for k in pool_sizes:
        types[f"{k}-{constants.MI355XNodeCellType}"] = CellTypeSpec(
            childCellType=constants.MI355XNodeCellType, childCellNumber=int(k)
        )
    return types


def mi355x_node_cell(node_name: str, gpu_indices: List[int] = None) -> PhysicalCellSpec:
    """Build an MI355X-NODE physical cell spec for one node.

    gpu_indices: physical device indices in xGMI-topology order (pairs first:
    indices [0,1] form a pair, [0..3] a quad). Defaults to 0..7.
    """
    if gpu_indices is None:
        gpu_indices = list(range(8))
    if len(gpu_indices) != 8:
        raise ValueError(f"MI355X node {node_name!r} needs exactly 8 GPUs, got {len(gpu_indices)}")
    quads = []
    for q in range(2):
        pairs = []
        for p in range(2):
            leaves = [
                PhysicalCellSpec(
                    cellType=constants.MI355XLeafCellType,
                    cellAddress=str(gpu_indices[q * 4 + p * 2 + g]),
                )
                for g in range(2)
            ]
            pairs.append(PhysicalCellSpec(cellType=constants.MI355XPairCellType, cellChildren=leaves))
        quads.append(PhysicalCellSpec(cellType=constants.MI355XQuadCellType, cellChildren=pairs))
    return PhysicalCellSpec(
        cellType=constants.MI355XNodeCellType, cellAddress=node_name, cellChildren=quads
    )
