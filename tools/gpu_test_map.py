"""Run every GPU test nodeid in its own subprocess so a GPU memory fault
aborts only that test, producing a full pass/fail/crash map."""
import subprocess, sys, json

def nodeids():
    out = subprocess.run(
        [sys.executable, "-m", "pytest", "tests/test_gpu_transcode.py",
         "tests/test_gpu_pipeline.py", "-m", "gpu", "--collect-only", "-q"],
        capture_output=True, text=True)
    ids = [l.strip() for l in out.stdout.splitlines() if "::" in l]
    return ids

def main():
    results = {}
    for nid in nodeids():
        try:
            p = subprocess.run(
                [sys.executable, "-m", "pytest", nid, "-q", "-x"],
                capture_output=True, text=True, timeout=180)
            if p.returncode == 0:
                results[nid] = "PASS"
            elif p.returncode in (134, -6, -11, 139):
                results[nid] = f"CRASH rc={p.returncode}"
            else:
                tail = "\n".join((p.stdout + p.stderr).splitlines()[-12:])
                results[nid] = f"FAIL rc={p.returncode}\n{tail}"
        except subprocess.TimeoutExpired:
            results[nid] = "TIMEOUT"
        print(f"{results[nid].splitlines()[0]:24s} {nid}", flush=True)
    crash = [k for k, v in results.items() if not v.startswith("PASS")]
    print(json.dumps({"bad": crash}, indent=1))
    with open("gpurun_out/test_map.json", "w") as f:
        json.dump(results, f, indent=1)

if __name__ == "__main__":
    main()
