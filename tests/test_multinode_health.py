"""Multi-node health probes (reference: huggingfaceserver multinode
health_check.py Ray probes; ours checks the torchrun rendezvous —
kserve_amd/parallel/health.py) and their wiring into the LWS workload."""

import datetime
import socket
import threading

import torch.distributed as dist

from kserve_amd.parallel.health import main, probe_master, probe_store


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def test_probe_master_up_and_down():
    srv = socket.socket()
    srv.bind(("127.0.0.1", 0))
    srv.listen(1)
    port = srv.getsockname()[1]
    try:
        assert probe_master("127.0.0.1", port, timeout_s=1.0)
    finally:
        srv.close()
    assert not probe_master("127.0.0.1", port, timeout_s=0.5)


def test_probe_store_roundtrip():
    port = free_port()
    master = dist.TCPStore(
        "127.0.0.1", port, 2, is_master=True,
        timeout=datetime.timedelta(seconds=10),
        wait_for_workers=False,
    )
    ok = {}
    t = threading.Thread(
        target=lambda: ok.setdefault(
            "v", probe_store("127.0.0.1", port, rank=1, world_size=2,
                             timeout_s=5.0)
        )
    )
    t.start()
    t.join(timeout=10)
    assert ok.get("v") is True
    del master


def test_probe_store_no_master_fails_fast():
    assert not probe_store("127.0.0.1", free_port(), rank=0, world_size=1,
                           timeout_s=0.5)


def test_cli_exit_codes():
    srv = socket.socket()
    srv.bind(("127.0.0.1", 0))
    srv.listen(1)
    port = srv.getsockname()[1]
    try:
        rc = main(["master", "--master-addr", "127.0.0.1",
                   "--master-port", str(port), "--timeout", "1"])
        assert rc == 0
    finally:
        srv.close()
    rc = main(["master", "--master-addr", "127.0.0.1",
               "--master-port", str(port), "--timeout", "0.5"])
    assert rc == 1


def test_lws_workload_carries_probes():
    from kserve_amd.controlplane.llmisvc import (
        LLMInferenceService,
        LLMInferenceServiceSpec,
        LLMModelSpec,
        ParallelismSpec,
        WorkloadSpec,
        render_workload,
    )

    llm = LLMInferenceService(
        name="big",
        namespace="ns",
        spec=LLMInferenceServiceSpec(
            model=LLMModelSpec(uri="hf://meta/llama-70b", name="llama"),
            workload=WorkloadSpec(
                parallelism=ParallelismSpec(tensor=8, pipeline=2)
            ),
        ),
    )
    lws = render_workload(llm)
    assert lws["kind"] == "LeaderWorkerSet"
    tmpl = lws["spec"]["leaderWorkerTemplate"]
    worker = tmpl["workerTemplate"]["spec"]["containers"][0]
    leader = tmpl["leaderTemplate"]["spec"]["containers"][0]
    assert worker["startupProbe"]["exec"]["command"][-1] == "master"
    assert worker["readinessProbe"]["exec"]["command"][-1] == "store"
    assert leader["livenessProbe"]["exec"]["command"][-1] == "gpu"
    assert "startupProbe" not in leader
