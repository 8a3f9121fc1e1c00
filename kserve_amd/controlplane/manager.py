"""Controller manager: one process running every controller.

Reference parity: cmd/manager/main.go — the manager wires all reconcilers
(InferenceService, InferenceGraph, TrainedModel, LLMInferenceService,
LocalModelCache) onto one API-server connection with health endpoints;
cmd/localmodelnode/main.go's per-node daemon maps to --node-name mode.

Real clusters use the kubectl adapter (``--kubectl``); ``--fake`` runs the
in-memory server (demos/tests). Each controller runs its own watch/worker
threads (Controller.start); /healthz and /readyz are served for the
Deployment probes the manager's own manifest declares.
"""

from __future__ import annotations

import argparse
import threading
import time
from typing import List

from kserve_amd.logging import logger


def build_controllers(server, node_name: str = "") -> List:
    from kserve_amd.controlplane.crd_controllers import (
        InferenceGraphController,
        TrainedModelController,
    )
    from kserve_amd.controlplane.isvc_controller import (
        InferenceServiceController,
    )
    from kserve_amd.controlplane.llmisvc_controller import (
        LLMInferenceServiceController,
    )
    from kserve_amd.controlplane.localmodel_controllers import (
        LocalModelCacheController,
        LocalModelNodeController,
    )

    controllers = [
        InferenceServiceController(server).build(),
        InferenceGraphController(server).build(),
        TrainedModelController(server).build(),
        LLMInferenceServiceController(server).build(),
        LocalModelCacheController(server).build(),
    ]
    if node_name:
        controllers.append(
            LocalModelNodeController(server, node_name).build()
        )
    return controllers


def serve_health(port: int, ready: threading.Event):
    from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

    class H(BaseHTTPRequestHandler):
        def log_message(self, *a):  # quiet
            pass

        def do_GET(self):
            if self.path == "/readyz" and not ready.is_set():
                self.send_response(503)
            else:
                self.send_response(200)
            self.end_headers()
            self.wfile.write(b"ok")

    srv = ThreadingHTTPServer(("0.0.0.0", port), H)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    return srv


def main(argv=None):
    ap = argparse.ArgumentParser(description="kserve-amd controller manager")
    ap.add_argument("--kubectl", default="kubectl",
                    help="kubectl binary for the real-cluster adapter")
    ap.add_argument("--fake", action="store_true",
                    help="in-memory API server (demo/test mode)")
    ap.add_argument("--node-name", default="",
                    help="also run the LocalModelNode daemon for this node")
    ap.add_argument("--health-port", type=int, default=8081)
    args = ap.parse_args(argv)

    if args.fake:
        from kserve_amd.controlplane.apiserver import FakeAPIServer

        server = FakeAPIServer()
    else:
        from kserve_amd.controlplane.apiserver import KubectlAPIServer

        server = KubectlAPIServer(kubectl=args.kubectl)

    ready = threading.Event()
    health = serve_health(args.health_port, ready)
    controllers = build_controllers(server, node_name=args.node_name)
    for c in controllers:
        c.start()
    ready.set()
    logger.info(
        "controller manager up: %d controllers, health on :%d",
        len(controllers), args.health_port,
    )
    try:
        while True:
            time.sleep(3600)
    except KeyboardInterrupt:
        pass
    finally:
        for c in controllers:
            c.stop()
        health.shutdown()


if __name__ == "__main__":
    main()
