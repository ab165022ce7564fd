"""grove_amd — MI355X-native Kubernetes inference-orchestration framework.

Brand-new implementation with the capabilities of ai-dynamo/grove: PodCliqueSet /
PodClique / PodCliqueScalingGroup / PodGang orchestration with hierarchical gang
scheduling, startup ordering, multi-level autoscaling, rolling updates, gang termination
and xGMI-topology-aware placement for 8×MI355X nodes.
"""
__version__ = "0.1.0"

from .cluster import Cluster  # noqa: F401
