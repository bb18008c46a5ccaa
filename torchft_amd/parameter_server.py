"""Prototype fault-tolerant parameter server.

An HTTP endpoint hands out fresh sessions: each ``/new_session`` creates a
unique TCPStore prefix and a 2-rank reconfigurable process group (server is
rank 0, client rank 1) over which the client exchanges parameters with
``ParameterServer.forward``-style handlers.

Reference parity: torchft/parameter_server.py:30-194.
"""

from __future__ import annotations

import json
import logging
import socket
import threading
import urllib.request
import uuid
from abc import ABC, abstractmethod
from datetime import timedelta
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

import torch
from torch.distributed import TCPStore

from torchft_amd.process_group import ProcessGroup, ProcessGroupGloo

logger = logging.getLogger(__name__)


def _hostname() -> str:
    import os

    env = os.environ.get("TORCHFT_AMD_HOSTNAME")
    if env:
        return env
    host = socket.gethostname()
    try:
        socket.getaddrinfo(host, None)
        return host
    except socket.gaierror:
        return "127.0.0.1"


class ParameterServer(ABC):
    """Serve model state to transient clients over per-session PGs."""

    def __init__(self, port: int = 0) -> None:
        self._store = TCPStore("0.0.0.0", 0, is_master=True, wait_for_workers=False)
        ps = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, fmt: str, *args: object) -> None:
                logger.debug("parameter_server: " + fmt % args)

            def do_POST(self) -> None:  # noqa: N802
                if self.path != "/new_session":
                    self.send_error(404)
                    return
                session_id = str(uuid.uuid4())
                store_addr = (
                    f"{_hostname()}:{ps._store.port}/session/{session_id}"
                )
                threading.Thread(
                    target=ps._handle_session,
                    args=(session_id, store_addr),
                    daemon=True,
                ).start()
                body = json.dumps({"session_id": session_id, "store_addr": store_addr}).encode()
                self.send_response(200)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

        self._server = ThreadingHTTPServer(("0.0.0.0", port), Handler)
        self._http_port = self._server.server_address[1]
        self._thread = threading.Thread(target=self._server.serve_forever, daemon=True)
        self._thread.start()

    def address(self) -> str:
        return f"http://{_hostname()}:{self._http_port}"

    @classmethod
    def new_process_group(cls) -> ProcessGroup:
        """Override to choose the session PG backend (RCCL on MI355X)."""
        return ProcessGroupGloo(timeout=timedelta(seconds=60))

    def _handle_session(self, session_id: str, store_addr: str) -> None:
        try:
            pg = self.new_process_group()
            pg.configure(store_addr, f"ps_{session_id}", rank=0, world_size=2)
            self.forward(session_id, pg)
        except Exception:  # noqa: BLE001
            logger.exception(f"parameter server session {session_id} failed")

    @abstractmethod
    def forward(self, session_id: str, pg: ProcessGroup) -> None:
        """Serve one client session over ``pg`` (server is rank 0)."""
        ...

    @classmethod
    def connect(cls, address: str, timeout: timedelta = timedelta(seconds=60)) -> ProcessGroup:
        """Client side: open a new session and return the configured PG
        (client is rank 1)."""
        with urllib.request.urlopen(
            f"{address}/new_session", data=b"", timeout=timeout.total_seconds()
        ) as resp:
            info = json.loads(resp.read())
        pg = cls.new_process_group()
        pg.configure(info["store_addr"], f"psc_{info['session_id']}", rank=1, world_size=2)
        return pg

    def shutdown(self) -> None:
        self._server.shutdown()
