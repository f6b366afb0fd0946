from .hive import GPUTopology, load_topology, preferred_sets, score_set  # noqa: F401
from .kfd import KFDNode, read_kfd_topology  # noqa: F401
