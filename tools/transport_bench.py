"""Micro-bench of the native h2c transport (client+server, no GPU)."""
import sys, time
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from ggrmcp_amd.backend.native_invoker import NativeWireClient, load_module

mod = load_module()
srv = mod.Server("127.0.0.1:0")
srv.add_route("/bench.EchoService/Echo", "echo")
bound = srv.start()

payload = b"x" * 1024
for conns in (1, 2, 4, 8, 16):
    cli = NativeWireClient(bound, connections=conns)
    paths = ["/bench.EchoService/Echo"] * 1024
    payloads = [payload] * 1024
    # warmup
    cli.invoke_batch(paths, payloads, 15.0, [[]] * 1024)
    t0 = time.perf_counter()
    iters = 5
    for _ in range(iters):
        res = cli.invoke_batch(paths, payloads, 15.0, [[]] * 1024)
    dt = time.perf_counter() - t0
    bad = sum(1 for r in res if isinstance(r, Exception))
    print(f"conns={conns:3d}  {1024*iters/dt:9.0f} req/s  {dt/iters*1e3:7.1f} ms/batch  errors={bad}")
    cli.close()
srv.stop()
