import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a real MI355X GPU (run via gpurun)")


@pytest.fixture(scope="session")
def design_config():
    """Heterogeneous simulated cluster exercising all features: two MI355X
    rack chains + single-node chains + a 2-GPU CT1 chain, pinned cell, two VCs.

    Modeled after the reference's design config
    (example/config/design/hivedscheduler.yaml), MI355X-flavored.
    """
    from hivedscheduler_amd.api import config as apicfg
    from hivedscheduler_amd.api.types import (
        CellTypeSpec,
        Config,
        PhysicalCellSpec,
        PhysicalClusterSpec,
        PinnedCellSpec,
        VirtualCellSpec,
        VirtualClusterSpec,
    )
    from hivedscheduler_amd.topo.mi355x import mi355x_cell_types, mi355x_node_cell

    cell_types = mi355x_cell_types(pool_sizes=(2, 3))
    cell_types["CT1-NODE"] = CellTypeSpec(childCellType="CT1", childCellNumber=2, isNodeLevel=True)

    rack3 = PhysicalCellSpec(
        cellType="3-MI355X-NODE",
        cellChildren=[mi355x_node_cell(n) for n in ("n1", "n2", "n3")],
    )
    rack2 = PhysicalCellSpec(
        cellType="2-MI355X-NODE",
        cellChildren=[mi355x_node_cell(n) for n in ("n4", "n5")],
    )
    rack2.cellChildren[0].pinnedCellId = "VC1-PIN"

    cfg = Config(
        physicalCluster=PhysicalClusterSpec(
            cellTypes=cell_types,
            physicalCells=[
                rack3,
                rack2,
                mi355x_node_cell("n6"),
                mi355x_node_cell("n7"),
                mi355x_node_cell("n8"),
                PhysicalCellSpec(cellType="CT1-NODE", cellAddress="c1"),
                PhysicalCellSpec(cellType="CT1-NODE", cellAddress="c2"),
            ],
        ),
        virtualClusters={
            "VC1": VirtualClusterSpec(
                virtualCells=[
                    VirtualCellSpec(cellType="3-MI355X-NODE.MI355X-NODE", cellNumber=2),
                    VirtualCellSpec(cellType="MI355X-NODE", cellNumber=1),
                    VirtualCellSpec(cellType="MI355X-NODE.MI355X-QUAD", cellNumber=1),
                ],
                pinnedCells=[PinnedCellSpec(pinnedCellId="VC1-PIN")],
            ),
            "VC2": VirtualClusterSpec(
                virtualCells=[
                    VirtualCellSpec(cellType="MI355X-NODE", cellNumber=1),
                    VirtualCellSpec(cellType="2-MI355X-NODE.MI355X-NODE", cellNumber=1),
                    VirtualCellSpec(cellType="CT1-NODE", cellNumber=1),
                ],
            ),
        },
    )
    apicfg.infer_physical_cluster(cfg.physicalCluster)
    return cfg


@pytest.fixture()
def design_sim(design_config):
    from hivedscheduler_amd.sim import SimScheduler

    return SimScheduler(design_config)
