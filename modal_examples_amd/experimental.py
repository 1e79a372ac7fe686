"""modal.experimental analog: clustered multi-rank functions + cluster info."""
from .parallel.cluster import ClusterInfo, clustered, get_cluster_info  # noqa: F401
