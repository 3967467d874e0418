"""HTTP facade for remote elastic rollout instances.

Speaks the engine HTTP contract the reference scheduler relies on
(/generate with SSE streaming, /health, /health_generate,
/get_server_info, /flush_cache, /abort_request, /update_weights_from_agent,
/release_memory_occupation, /resume_memory_occupation, /shutdown), so a
spot/preemptible node can join an existing run at runtime (§3.4 lifecycle).
"""
from .engine_server import create_app
from .http_instance import HttpInstance

__all__ = ["create_app", "HttpInstance"]
