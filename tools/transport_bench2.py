"""Same micro-bench but over UDS + separate backend process (bench.py shape)."""
import os, subprocess, sys, tempfile, time
from pathlib import Path
ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(ROOT))
from ggrmcp_amd.backend.native_invoker import NativeWireClient

sock = os.path.join(tempfile.gettempdir(), f"tb2_{os.getpid()}.sock")
proc = subprocess.Popen([sys.executable, "-m", "examples.bench_backend",
                         "--uds", sock, "--native"],
                        stdout=subprocess.PIPE, cwd=str(ROOT), text=True)
line = proc.stdout.readline()
assert line.startswith("READY"), line
try:
    payload = b"x" * 1024
    for conns in (8, 16):
        cli = NativeWireClient(f"unix:{sock}", connections=conns)
        paths = ["/bench.EchoService/Echo"] * 1024
        payloads = [payload] * 1024
        cli.invoke_batch(paths, payloads, 15.0, [[]] * 1024)
        t0 = time.perf_counter()
        iters = 5
        for _ in range(iters):
            res = cli.invoke_batch(paths, payloads, 15.0, [[]] * 1024)
        dt = time.perf_counter() - t0
        bad = sum(1 for r in res if isinstance(r, Exception))
        print(f"uds conns={conns:3d}  {1024*iters/dt:9.0f} req/s  {dt/iters*1e3:7.1f} ms/batch errors={bad}")
        cli.close()
    # and the real SayHello path with ~1KB hello wire
    import random
    from ggrmcp_amd.utils.synthetic import hello_payload
    cli = NativeWireClient(f"unix:{sock}", connections=8)
    rng = random.Random(1)
    wire = b"\x0a" + bytes([0xec, 0x07]) + b"y" * 1004  # name field ~1KB
    paths = ["/hello.HelloService/SayHello"] * 1024
    payloads = [wire] * 1024
    cli.invoke_batch(paths, payloads, 15.0, [[]] * 1024)
    t0 = time.perf_counter()
    for _ in range(5):
        res = cli.invoke_batch(paths, payloads, 15.0, [[]] * 1024)
    dt = time.perf_counter() - t0
    bad = sum(1 for r in res if isinstance(r, Exception))
    print(f"hello    conns=8  {1024*5/dt:9.0f} req/s  {dt/5*1e3:7.1f} ms/batch errors={bad}")
    cli.close()
finally:
    proc.terminate()
