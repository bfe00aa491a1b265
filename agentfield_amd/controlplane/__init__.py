"""MI355X-native agent control plane.

Reimplements the reference's Go control plane surface
as an async Python service with native C++ components where it counts
(Ed25519 DID/VC signing, the engine-side scheduler): REST execute API with
sync + durable-async paths, node registry/heartbeats/presence, HMAC-signed
webhooks with DB-backed retry, workflow DAG tracing, hierarchical KV+vector
memory with change events, W3C DID/VC audit, Prometheus metrics.
"""
from .server import create_app, ControlPlane

__all__ = ["create_app", "ControlPlane"]
