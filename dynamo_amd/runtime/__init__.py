from .discovery import (DiscoveryBackend, FileDiscovery, Instance,
                        MemoryDiscovery, make_discovery, new_instance_id)
from .endpoint import (Component, DistributedRuntime, Endpoint, Namespace,
                       NoInstancesError, PushClient)
from .request_plane import (EndpointError, RequestContext, RequestPlaneClient,
                            RequestPlaneServer)

__all__ = [
    "DiscoveryBackend", "FileDiscovery", "Instance", "MemoryDiscovery",
    "make_discovery", "new_instance_id", "Component", "DistributedRuntime",
    "Endpoint", "Namespace", "NoInstancesError", "PushClient",
    "EndpointError", "RequestContext", "RequestPlaneClient",
    "RequestPlaneServer",
]
