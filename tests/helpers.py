"""Test helpers: run ASGI apps on real localhost ports in threads (the
reference's integration tier does the same with the real server binary)."""
from __future__ import annotations

import socket
import threading
import time

import httpx
import uvicorn


class AppServer:
    def __init__(self, app, host: str = "127.0.0.1"):
        self.app = app
        self.host = host
        sock = socket.socket()
        sock.bind((host, 0))
        self.port = sock.getsockname()[1]
        sock.close()
        self.base_url = f"http://{host}:{self.port}"
        config = uvicorn.Config(app, host=host, port=self.port,
                                log_level="error", lifespan="on")
        self.server = uvicorn.Server(config)
        self.thread = threading.Thread(target=self.server.run, daemon=True)

    def start(self, wait: float = 10.0) -> "AppServer":
        self.thread.start()
        deadline = time.time() + wait
        while time.time() < deadline:
            if self.server.started:
                return self
            time.sleep(0.01)
        raise TimeoutError("server did not start")

    def stop(self):
        self.server.should_exit = True
        self.thread.join(timeout=5)

    def wait_healthy(self, path: str = "/api/v1/health", timeout: float = 5.0):
        deadline = time.time() + timeout
        while time.time() < deadline:
            try:
                if httpx.get(self.base_url + path, timeout=1.0).status_code == 200:
                    return self
            except httpx.HTTPError:
                pass
            time.sleep(0.02)
        raise TimeoutError("server not healthy")


def wait_until(cond, timeout: float = 10.0, interval: float = 0.02):
    deadline = time.time() + timeout
    while time.time() < deadline:
        v = cond()
        if v:
            return v
        time.sleep(interval)
    raise TimeoutError("condition not met")
