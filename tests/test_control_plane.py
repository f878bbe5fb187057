import threading

import torch

from mdi_llm_amd.parallel.control import (
    ControlClient,
    ControlServer,
    NodeTopology,
)


def test_topology_reference_schema():
    topo = NodeTopology.from_file("settings_distr/configuration_2.json")
    assert topo.n_nodes == 2
    addr, port = topo.http_endpoint(1)
    assert addr == "127.0.0.1"
    assert topo.master_addr == "127.0.0.1"
    assert topo.master_port > 0


def test_topology_device_override():
    topo = NodeTopology.from_file("settings_distr/config_2gpus.json")
    assert topo.device_for(0) == "cuda:0"
    assert topo.device_for(1) == "cuda:1"
    assert topo.device_for(1, "cpu") == "cpu"


def test_init_stop_roundtrip():
    srv = ControlServer("127.0.0.1", 18731)
    try:
        client = ControlClient(max_tries=5, retry_delay=0.2)
        info = client.node_info("127.0.0.1", 18731)
        assert info["ready"] is False

        msg = {
            "role": "secondary:0",
            "rank": 1,
            "model_config": {"name": "nano-test"},
            "params": {"w": torch.arange(6, dtype=torch.float32).view(2, 3)},
        }
        client.init_node("127.0.0.1", 18731, msg)
        got = srv.wait_for_init(timeout=5)
        assert got["rank"] == 1
        assert torch.equal(got["params"]["w"], msg["params"]["w"])
        assert client.node_info("127.0.0.1", 18731)["ready"] is True

        client.stop_node("127.0.0.1", 18731)
        assert srv.stop_event.wait(timeout=5)
    finally:
        srv.shutdown()


def test_client_retries_until_server_appears():
    client = ControlClient(max_tries=20, retry_delay=0.1)
    srv_holder = {}

    def delayed_start():
        import time

        time.sleep(0.5)
        srv_holder["srv"] = ControlServer("127.0.0.1", 18732)

    t = threading.Thread(target=delayed_start)
    t.start()
    info = client.node_info("127.0.0.1", 18732)  # retries until up
    assert "ready" in info
    t.join()
    srv_holder["srv"].shutdown()
