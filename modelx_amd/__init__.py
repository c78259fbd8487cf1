"""modelx_amd — MI355X-native model-artifact registry.

A from-scratch rebuild of kubegems/modelx's capabilities with an AMD-native
data plane: the control plane is a C++ `modelxd` HTTP server speaking the
modelx OCI-style manifest/blob JSON API; the client data plane streams
presigned S3 range-GETs through a pinned-host ring into MI355X HBM3E via
hipMemcpyAsync, verifies content with hand-written CDNA4 HIP SHA-256 kernels,
and fans checkpoints out to all 8 GPUs of a node with RCCL over xGMI.
"""
from ._version import __version__  # noqa: F401
from .wire import digest, errors, paths, types  # noqa: F401


def _client():
    from .client import Client

    return Client


def __getattr__(name):
    if name == "Client":
        from .client import Client

        return Client
    raise AttributeError(name)
