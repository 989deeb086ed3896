"""Isolate: {tcp,uds} x {same,sep} process, and batch-size scaling."""
import os, subprocess, sys, tempfile, time
from pathlib import Path
ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(ROOT))
from ggrmcp_amd.backend.native_invoker import NativeWireClient, load_module

def bench(cli, label, batch=1024, iters=5):
    paths = ["/bench.EchoService/Echo"] * batch
    payloads = [b"x" * 1024] * batch
    cli.invoke_batch(paths, payloads, 15.0, [[]] * batch)
    t0 = time.perf_counter()
    for _ in range(iters):
        res = cli.invoke_batch(paths, payloads, 15.0, [[]] * batch)
    dt = time.perf_counter() - t0
    bad = sum(1 for r in res if isinstance(r, Exception))
    print(f"{label:34s} batch={batch:5d} {batch*iters/dt:9.0f} req/s  {dt/iters*1e3:8.2f} ms/batch err={bad}")

mod = load_module()

# same-process UDS
sock1 = os.path.join(tempfile.gettempdir(), f"tb3a_{os.getpid()}.sock")
srv = mod.Server(f"unix:{sock1}")
srv.add_route("/bench.EchoService/Echo", "echo")
srv.start()
cli = NativeWireClient(f"unix:{sock1}", connections=8)
bench(cli, "uds same-process")
for b in (1, 16, 128):
    bench(cli, "uds same-process", batch=b, iters=20)
cli.close(); srv.stop()

# separate-process TCP
proc = subprocess.Popen([sys.executable, "-m", "examples.bench_backend",
                         "--port", "0", "--native"],
                        stdout=subprocess.PIPE, cwd=str(ROOT), text=True)
line = proc.stdout.readline(); assert line.startswith("READY")
target = line.split()[1]
cli = NativeWireClient(target, connections=8)
bench(cli, "tcp separate-process")
for b in (1, 16, 128):
    bench(cli, "tcp separate-process", batch=b, iters=20)
cli.close(); proc.terminate()

# separate-process UDS
sock2 = os.path.join(tempfile.gettempdir(), f"tb3b_{os.getpid()}.sock")
proc = subprocess.Popen([sys.executable, "-m", "examples.bench_backend",
                         "--uds", sock2, "--native"],
                        stdout=subprocess.PIPE, cwd=str(ROOT), text=True)
line = proc.stdout.readline(); assert line.startswith("READY")
cli = NativeWireClient(f"unix:{sock2}", connections=8)
bench(cli, "uds separate-process")
for b in (1, 16, 128):
    bench(cli, "uds separate-process", batch=b, iters=20)
cli.close(); proc.terminate()
