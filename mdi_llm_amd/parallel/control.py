"""HTTP control plane: node bootstrap (init / stop / info) + topology JSON.

Capability parity with the reference's control plane
(/root/reference/src/sub/model_dist.py:402-573 ``configure_nodes`` /
``stop_nodes`` / ``_request_to_node`` and the CherryPy REST resource in
gptserver.py:1114-1226), rebuilt on the stdlib HTTP server (CherryPy-free)
and with RCCL bootstrap info riding the init message: the starter tells
every secondary the torch.distributed rendezvous (master addr/port, rank,
world) so the RCCL/gloo ring comes up off the hot path, exactly where the
reference exchanged socket ports.

Topology JSON schema is the reference's
(settings_distr/configuration*.json): ``nodes.starter{addr,
communication.port, inference.{port_in,port_out}, [device]}`` +
``nodes.secondary[i]{...}``; ``inference.port_in`` of the starter doubles
as the torch.distributed master port.
"""

from __future__ import annotations

import io
import json
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from pathlib import Path
from typing import Any, Dict, List, Optional, Union

import torch
import urllib.request
import urllib.error

__all__ = ["NodeTopology", "ControlServer", "ControlClient"]


class NodeTopology:
    """Parsed node-topology JSON (reference schema)."""

    def __init__(self, raw: dict):
        self.raw = raw
        nodes = raw["nodes"]
        self.starter = nodes["starter"]
        self.secondary: List[dict] = nodes.get("secondary", [])

    @classmethod
    def from_file(cls, path: Union[str, Path]) -> "NodeTopology":
        with open(path, encoding="utf-8") as fp:
            return cls(json.load(fp))

    @property
    def n_nodes(self) -> int:
        return 1 + len(self.secondary)

    def node(self, rank: int) -> dict:
        return self.starter if rank == 0 else self.secondary[rank - 1]

    def http_endpoint(self, rank: int):
        n = self.node(rank)
        return n["addr"], int(n["communication"]["port"])

    @property
    def master_addr(self) -> str:
        return self.starter["addr"]

    @property
    def master_port(self) -> int:
        # starter's inference.port_in doubles as the dist rendezvous port
        return int(self.starter["inference"]["port_in"])

    def device_for(self, rank: int, override: Optional[str] = None) -> str:
        # priority CLI > JSON > default (reference gptserver.py:601-617)
        if override:
            return override
        n = self.node(rank)
        if "device" in n:
            return n["device"]
        return "cuda" if torch.cuda.is_available() else "cpu"


def _serialize(msg: Dict[str, Any]) -> bytes:
    buf = io.BytesIO()
    torch.save(msg, buf)
    return buf.getvalue()


def _deserialize(data: bytes) -> Dict[str, Any]:
    return torch.load(io.BytesIO(data), map_location="cpu",
                      weights_only=True)


class ControlServer:
    """Secondary-side HTTP server: waits for /init, serves /info, obeys
    /stop."""

    def __init__(self, host: str, port: int):
        self.init_msg: Optional[dict] = None
        self.init_event = threading.Event()
        self.stop_event = threading.Event()
        outer = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, *a):  # quiet
                pass

            def do_GET(self):
                body = json.dumps(
                    {"role": "secondary", "ready": outer.init_msg is not None}
                ).encode()
                self.send_response(200)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def do_POST(self):
                if self.path.rstrip("/") != "/init":
                    self.send_response(404)
                    self.end_headers()
                    return
                n = int(self.headers.get("Content-Length", 0))
                data = self.rfile.read(n)
                try:
                    outer.init_msg = _deserialize(data)
                except Exception as e:  # noqa: BLE001
                    self.send_response(400)
                    self.end_headers()
                    self.wfile.write(str(e).encode())
                    return
                outer.init_event.set()
                self.send_response(200)
                self.send_header("Content-Length", "2")
                self.end_headers()
                self.wfile.write(b"ok")

            def do_PUT(self):
                if self.path.rstrip("/") != "/stop":
                    self.send_response(404)
                    self.end_headers()
                    return
                outer.stop_event.set()
                self.send_response(200)
                self.send_header("Content-Length", "2")
                self.end_headers()
                self.wfile.write(b"ok")

        self.httpd = ThreadingHTTPServer((host, port), Handler)
        self.thread = threading.Thread(target=self.httpd.serve_forever,
                                       daemon=True)
        self.thread.start()

    def wait_for_init(self, timeout: Optional[float] = None) -> dict:
        if not self.init_event.wait(timeout):
            raise TimeoutError("no /init received")
        return self.init_msg

    def shutdown(self):
        self.httpd.shutdown()
        self.thread.join(timeout=5)


class ControlClient:
    """Starter-side client with the reference's retry discipline
    (model_dist.py:499-573: up to ``max_tries`` attempts, 2 s apart)."""

    def __init__(self, max_tries: int = 100, retry_delay: float = 2.0):
        self.max_tries = max_tries
        self.retry_delay = retry_delay

    def _request(self, method: str, url: str, data: Optional[bytes] = None):
        last = None
        for _ in range(self.max_tries):
            try:
                req = urllib.request.Request(url, data=data, method=method)
                with urllib.request.urlopen(req, timeout=300) as resp:
                    return resp.read()
            except (urllib.error.URLError, ConnectionError, OSError) as e:
                last = e
                time.sleep(self.retry_delay)
        raise ConnectionError(f"{method} {url} failed after "
                              f"{self.max_tries} tries: {last}")

    def init_node(self, addr: str, port: int, msg: Dict[str, Any]) -> None:
        self._request("POST", f"http://{addr}:{port}/init", _serialize(msg))

    def stop_node(self, addr: str, port: int) -> None:
        self._request("PUT", f"http://{addr}:{port}/stop")

    def node_info(self, addr: str, port: int) -> dict:
        return json.loads(self._request("GET", f"http://{addr}:{port}/"))
