"""Admin gRPC service (reference parity: C2 — AdminReasonerService on port
HTTP+100, env AGENTFIELD_ADMIN_GRPC_PORT).

grpcio is available offline but protoc is not, so messages are JSON-encoded
over real gRPC/HTTP-2 via generic handlers; proto/admin.proto documents the
intended schema for protoc users.
"""
from __future__ import annotations

import json
from concurrent import futures

import grpc

SERVICE = "agentfield.admin.AdminService"


def _ser(d: dict) -> bytes:
    return json.dumps(d).encode()


def _de(b: bytes) -> dict:
    return json.loads(b or b"{}")


def start_admin_grpc(cp, host: str = "127.0.0.1", port: int = 8620,
                     workers: int = 4) -> grpc.Server:
    storage = cp.storage

    def list_reasoners(req, ctx):
        out = []
        for n in storage.list_nodes():
            if req.get("node_id") and n["id"] != req["node_id"]:
                continue
            for r in n.get("reasoners", []):
                out.append({"node_id": n["id"],
                            **(r if isinstance(r, dict) else {"id": r})})
        return {"reasoners": out}

    def list_nodes(req, ctx):
        return {"nodes": storage.list_nodes()}

    def get_execution(req, ctx):
        rec = storage.get_execution(req.get("execution_id", ""))
        if rec is None:
            ctx.set_code(grpc.StatusCode.NOT_FOUND)
            return {}
        return {"execution": cp.envelope(rec)}

    def server_status(req, ctx):
        return {"status": "healthy",
                "nodes": len(storage.list_nodes()),
                "queue_depth": cp._async_q.qsize() if cp._async_q else 0}

    methods = {
        "ListReasoners": list_reasoners,
        "ListNodes": list_nodes,
        "GetExecution": get_execution,
        "ServerStatus": server_status,
    }
    handlers = {
        name: grpc.unary_unary_rpc_method_handler(
            fn, request_deserializer=_de, response_serializer=_ser)
        for name, fn in methods.items()
    }
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=workers))
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(SERVICE, handlers),))
    server.add_insecure_port(f"{host}:{port}")
    server.start()
    return server


class AdminClient:
    def __init__(self, target: str):
        self.channel = grpc.insecure_channel(target)

    def _call(self, method: str, req: dict) -> dict:
        fn = self.channel.unary_unary(f"/{SERVICE}/{method}",
                                      request_serializer=_ser,
                                      response_deserializer=_de)
        return fn(req)

    def list_reasoners(self, node_id: str | None = None) -> dict:
        return self._call("ListReasoners", {"node_id": node_id})

    def list_nodes(self) -> dict:
        return self._call("ListNodes", {})

    def get_execution(self, execution_id: str) -> dict:
        return self._call("GetExecution", {"execution_id": execution_id})

    def server_status(self) -> dict:
        return self._call("ServerStatus", {})
