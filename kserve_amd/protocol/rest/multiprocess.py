"""Multi-process REST serving: N workers sharing one listening socket.

Reference parity: python/kserve protocol/rest/multiprocess/server.py —
``RESTServerMultiProcess`` (:124, socket-sharing worker processes with a
health-monitor restart loop), ``RESTServerProcess`` (:33).

Intended for CPU-bound predictive runtimes (sklearn/xgb/...); GPU engines
run single-process (the engine owns the device).
"""

from __future__ import annotations

import multiprocessing as mp
import os
import socket
import time
from typing import Callable, List, Optional

from kserve_amd.logging import logger


def _worker_entry(app_factory, sock, log_level):
    import asyncio

    import uvicorn

    app = app_factory()
    config = uvicorn.Config(app, log_config=None, access_log=False)
    server = uvicorn.Server(config)
    asyncio.run(server.serve(sockets=[sock]))


class RESTServerMultiProcess:
    """Spawn ``workers`` uvicorn processes sharing one bound socket; restart
    workers that die (reference health-monitor loop)."""

    def __init__(
        self,
        app_factory: Callable,
        http_port: int = 8080,
        host: str = "0.0.0.0",
        workers: int = 2,
        monitor_interval_s: float = 2.0,
    ):
        self.app_factory = app_factory
        self.workers = workers
        self.monitor_interval_s = monitor_interval_s
        self.sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self.sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self.sock.bind((host, http_port))
        self.sock.listen(2048)
        self.port = self.sock.getsockname()[1]
        self._procs: List[mp.Process] = []
        self._stopping = False

    def _spawn_one(self) -> mp.Process:
        ctx = mp.get_context("fork")  # share the bound socket fd
        p = ctx.Process(
            target=_worker_entry, args=(self.app_factory, self.sock, "info")
        )
        p.daemon = True
        p.start()
        return p

    def start(self):
        for _ in range(self.workers):
            self._procs.append(self._spawn_one())
        logger.info(
            "REST multiprocess: %d workers on port %d", self.workers, self.port
        )

    def monitor(self):
        """Blocking restart loop (run in a thread or as the main loop)."""
        while not self._stopping:
            for i, p in enumerate(self._procs):
                if not p.is_alive() and not self._stopping:
                    logger.warning("REST worker %d died (exit %s); restarting", i, p.exitcode)
                    self._procs[i] = self._spawn_one()
            time.sleep(self.monitor_interval_s)

    def stop(self):
        self._stopping = True
        for p in self._procs:
            p.terminate()
        for p in self._procs:
            p.join(timeout=5)
        self.sock.close()
