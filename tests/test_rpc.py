import torch

from elasticdl_amd.common.rpc import RpcClient, start_server


def test_rpc_roundtrip_with_tensors():
    received = {}

    def echo(req):
        received.update(req)
        return {"sum": req["t"].sum(), "tag": req["tag"]}

    server = start_server("127.0.0.1:0", {"Test": {"echo": echo}}, max_workers=2)
    try:
        client = RpcClient(f"127.0.0.1:{server.port}")
        t = torch.ones(10)
        resp = client.call("Test", "echo", {"t": t, "tag": "hi"})
        assert float(resp["sum"]) == 10.0
        assert resp["tag"] == "hi"
        assert torch.equal(received["t"], t)
        client.close()
    finally:
        server.stop(0)


def test_rpc_futures_fanout():
    def double(req):
        return {"v": req["v"] * 2}

    server = start_server("127.0.0.1:0", {"S": {"double": double}})
    try:
        client = RpcClient(f"127.0.0.1:{server.port}")
        futs = [client.call_future("S", "double", {"v": i}) for i in range(8)]
        results = [RpcClient.resolve(f)["v"] for f in futs]
        assert results == [i * 2 for i in range(8)]
        client.close()
    finally:
        server.stop(0)
