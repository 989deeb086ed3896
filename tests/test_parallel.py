"""Multi-process DP shard tests (gloo, world_size 2 — runs on CPU).

Covers the MI355X control plane that replaces the reference's in-process
shared state (discovery.go:122-127 atomic swap; manager.go sessions):
descriptor broadcast, checksum verification, stats all-reduce, session
shard affinity.
"""

import json
import os

import pytest
import torch.multiprocessing as mp

from ggrmcp_amd.parallel.dist import ShardGroup, fnv1a64, shard_for_session


def test_shard_for_session_stable():
    assert shard_for_session("abc", 1) == 0
    a = shard_for_session("session-1", 8)
    assert 0 <= a < 8
    assert shard_for_session("session-1", 8) == a  # stable


def test_shard_distribution_roughly_uniform():
    counts = [0] * 8
    for i in range(4000):
        counts[shard_for_session(f"sess-{i:08x}", 8)] += 1
    assert min(counts) > 300  # no empty/starved shard


def test_fnv_matches_kernel_constant():
    # same FNV-1a64 the HIP kernels use for name hashing (common.h:177-184)
    assert fnv1a64(b"") == 0xCBF29CE484222325
    assert fnv1a64(b"a") == 0xAF63DC4C8601EC8C


def test_degenerate_group_world1():
    g = ShardGroup(rank=0, world=1)
    assert g.broadcast_blob(b"xyz") == b"xyz"
    assert g.verify_consistent(b"xyz")
    assert g.allreduce_stats({"a": 1.5})["a"] == 1.5
    g.barrier()  # no-op


# ---- world_size 2 over gloo -------------------------------------------------

def _worker(rank: int, world: int, port: int, results_dir: str):
    import torch.distributed as dist

    dist.init_process_group(
        backend="gloo",
        init_method=f"tcp://127.0.0.1:{port}",
        rank=rank,
        world_size=world,
    )
    g = ShardGroup.attach(dist)
    assert g.rank == rank and g.world == world

    # broadcast: only rank 0 has the payload
    payload = b"descriptor-snapshot-\x00\xff" * 100 if rank == 0 else None
    got = g.broadcast_blob(payload, src=0)
    assert got == b"descriptor-snapshot-\x00\xff" * 100
    assert g.verify_consistent(got)

    # sync_discovery against a stub discoverer
    class StubDisc:
        def __init__(self):
            self.loaded = None
            self.tools_version = 1

        def descriptor_blob(self):
            return b"BLOB" * 64

        def load_descriptor_blob(self, blob):
            self.loaded = blob
            self.tools_version += 1

    from ggrmcp_amd.parallel.dist import sync_discovery

    d = StubDisc()
    sync_discovery(d, g, src=0)
    if rank != 0:
        assert d.loaded == b"BLOB" * 64

    stats = g.allreduce_stats({"requests": 10.0 * (rank + 1), "errors": 1.0})
    assert stats["requests"] == 30.0
    assert stats["errors"] == 2.0

    # rediscovery: a CHANGED tool set must swap atomically on every shard
    # (the cross-shard analogue of discovery.go:122-127's atomic publish)
    d2 = StubDisc()
    d2.descriptor_blob = lambda: b"NEWSET" * 32
    sync_discovery(d2, g, src=0)
    if rank != 0:
        assert d2.loaded == b"NEWSET" * 32
    # sync_discovery's checksum all-gather + barrier already proved every
    # shard holds identical bytes before anyone proceeds

    with open(os.path.join(results_dir, f"rank{rank}.json"), "w") as f:
        json.dump({"ok": True}, f)
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_world2_gloo(tmp_path):
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    ctx = mp.spawn(
        _worker, args=(2, port, str(tmp_path)), nprocs=2, join=True,
        start_method="spawn",
    )
    del ctx
    for rank in range(2):
        with open(tmp_path / f"rank{rank}.json") as f:
            assert json.load(f)["ok"]


@pytest.mark.timeout(180)
@pytest.mark.parametrize("use_descriptor", [True, False],
                         ids=["descriptor", "reflection"])
def test_serve_dp_world2_shared_port(tmp_path, use_descriptor):
    """tools/serve_dp.py end to end: two gateway ranks (gloo, CPU pipeline)
    share ONE port via SO_REUSEPORT, rank 0 discovers (descriptor file OR
    live reflection) and broadcasts the snapshot, and real HTTP tool calls
    round-trip through whichever rank the kernel picks."""
    import http.client
    import socket
    import subprocess
    import sys
    import time as _t

    from examples.hello_service import serve as serve_hello
    from examples.protos import ALL_FDPS
    from google.protobuf import descriptor_pb2

    sock_path = str(tmp_path / "hello.sock")
    server, _ = serve_hello(target=f"unix:{sock_path}")
    try:
        fdset = descriptor_pb2.FileDescriptorSet()
        fdset.file.extend(ALL_FDPS)
        desc_path = tmp_path / "svc.binpb"
        desc_path.write_bytes(fdset.SerializeToString())

        def free_port():
            s = socket.socket()
            s.bind(("127.0.0.1", 0))
            p = s.getsockname()[1]
            s.close()
            return p

        http_port, master_port = free_port(), free_port()
        cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
               "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
               "--master-port", str(master_port), "tools/serve_dp.py",
               "--port", str(http_port), "--uds", sock_path,
               "--no-gpu", "--run-seconds", "60"]
        if use_descriptor:
            cmd += ["--descriptor", str(desc_path)]
        # else: rank 0 discovers via live gRPC REFLECTION, then broadcasts
        proc = subprocess.Popen(
            cmd, stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
        )
        try:
            body = json.dumps({
                "jsonrpc": "2.0", "id": 5, "method": "tools/call",
                "params": {"name": "hello_helloservice_sayhello",
                           "arguments": {"name": "dp"}},
            })
            deadline = _t.time() + 90
            ok = 0
            while _t.time() < deadline and ok < 6:
                try:
                    conn = http.client.HTTPConnection("127.0.0.1", http_port,
                                                      timeout=5)
                    conn.request("POST", "/", body=body,
                                 headers={"Content-Type": "application/json"})
                    r = conn.getresponse()
                    data = json.loads(r.read())
                    conn.close()
                    if r.status == 200 and not data["result"]["isError"]:
                        inner = json.loads(
                            data["result"]["content"][0]["text"])
                        assert inner == {"message": "Hello, dp!"}
                        ok += 1
                        continue
                except (ConnectionError, OSError, TimeoutError):
                    pass
                _t.sleep(0.5)
            assert ok >= 6, f"only {ok} round-trips; output:\n" + (
                proc.stdout.read() if proc.poll() is not None else "")

            if use_descriptor:
                # cross-rank session affinity: the same
                # Mcp-Session-Id over FRESH connections (kernel may land each
                # on either rank) must hit ONE shared session — the
                # /dev/shm-backed C++ table both ranks map.
                def post(sid=None):
                    hdrs = {"Content-Type": "application/json"}
                    if sid:
                        hdrs["Mcp-Session-Id"] = sid
                    conn = http.client.HTTPConnection("127.0.0.1", http_port,
                                                      timeout=5)
                    conn.request("POST", "/", body=body, headers=hdrs)
                    r = conn.getresponse()
                    r.read()
                    out_sid = r.getheader("Mcp-Session-Id")
                    conn.close()
                    return out_sid

                def metrics():
                    conn = http.client.HTTPConnection("127.0.0.1", http_port,
                                                      timeout=5)
                    conn.request("GET", "/metrics")
                    r = conn.getresponse()
                    data = json.loads(r.read())
                    conn.close()
                    return data["sessions"]

                m0 = metrics()
                sid = post()
                assert sid
                for _ in range(5):
                    assert post(sid) == sid
                m1 = metrics()
                # 6 calls booked on ONE new session, regardless of which
                # rank each fresh connection landed on
                assert m1["totalCalls"] - m0["totalCalls"] == 6
                assert m1["activeSessions"] - m0["activeSessions"] == 1
        finally:
            proc.terminate()
            try:
                proc.wait(timeout=15)
            except Exception:
                proc.kill()
    finally:
        server.stop(0)
