"""SCALE-path rehearsal on ONE GPU: keep the exact
command shape the driver uses for multi-GPU scaling well-rehearsed so the
unattended 1/2/4/8 run works first try.

* ``test_bench_world2_gloo``: bench.py under torch.distributed.run with
  2 ranks on the single visible device (GGRMCP_DIST_BACKEND=gloo, since
  RCCL refuses two ranks on one GPU) — full serving stack per rank, rank 0
  emits the whole-job JSON.
* ``test_rccl_world1_smoke``: RCCL (backend "nccl") process-group init +
  the sync_discovery broadcast/checksum path on device — the collective
  code the 8-GPU run uses, in its degenerate world-1 form.
"""

import json
import os
import socket
import subprocess
import sys
from pathlib import Path

import pytest

pytestmark = pytest.mark.gpu

REPO = Path(__file__).resolve().parent.parent


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


@pytest.mark.timeout(600)
def test_bench_world2_gloo():
    env = dict(os.environ)
    env["GGRMCP_DIST_BACKEND"] = "gloo"  # 2 ranks, 1 physical device
    env["MASTER_ADDR"] = "127.0.0.1"
    cmd = [
        sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
        "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
        "--master-port", str(_free_port()), "bench.py", "--gpus", "2",
        "--steps", "20", "--warmup", "5", "--batch", "128",
        "--streams", "2", "--reactors", "2", "--client-threads", "2",
    ]
    p = subprocess.run(cmd, capture_output=True, text=True, timeout=540,
                       cwd=str(REPO), env=env)
    assert p.returncode == 0, p.stdout[-2000:] + p.stderr[-2000:]
    line = next(
        (ln for ln in reversed(p.stdout.strip().splitlines())
         if ln.startswith("{")),
        None,
    )
    assert line, "rank 0 must print exactly one JSON line:\n" + p.stdout[-2000:]
    d = json.loads(line)
    assert d["n_gpus"] == 2
    assert d["value"] > 0
    assert d["config"]["parallelism"] == "dp2"
    assert d["config"]["mode"] == "serving-gpu"


@pytest.mark.timeout(300)
def test_rccl_world1_smoke():
    """RCCL init + broadcast/verify on device, world 1 (the collective
    path of sync_discovery with the real nccl/RCCL backend)."""
    import torch
    import torch.distributed as dist

    from ggrmcp_amd.parallel.dist import ShardGroup

    store = dist.TCPStore("127.0.0.1", _free_port(), 1, True)
    dist.init_process_group("nccl", store=store, rank=0, world_size=1)
    try:
        torch.cuda.set_device(0)
        # a REAL RCCL collective (world-1): exercises communicator init +
        # kernel launch on the device — the exact backend the 8-GPU
        # scaling run uses
        t = torch.ones(1 << 20, dtype=torch.float32, device="cuda:0")
        dist.all_reduce(t)
        torch.cuda.synchronize()
        assert float(t.sum().item()) == float(1 << 20)
        # ShardGroup wrappers (degenerate world short-circuits documented)
        g = ShardGroup.attach(dist, device=0)
        blob = b"descriptor-snapshot-payload" * 100
        out = g.broadcast_blob(blob, src=0)
        assert out == blob
        assert g.verify_consistent(out)
        agg = g.allreduce_stats({"requests": 7.0, "errors": 1.0})
        assert agg["requests"] == 7.0
        g.barrier()
    finally:
        dist.destroy_process_group()
