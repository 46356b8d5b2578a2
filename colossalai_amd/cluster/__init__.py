from .dist_coordinator import DistCoordinator
from .process_group_mesh import ProcessGroupMesh

__all__ = ["DistCoordinator", "ProcessGroupMesh"]
