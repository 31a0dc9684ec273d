import os, subprocess, sys
for tag, env in [
    ("graphs+streams", {}),
    ("graphs,nostreams", {"ADANET_NO_STREAMS": "1"}),
    ("nographs+streams", {"ADANET_NO_GRAPHS": "1"}),
]:
    e = dict(os.environ); e.update(env)
    r = subprocess.run([sys.executable, "bench.py", "--gpus", "1",
                        "--steps", "1", "--warmup", "0", "--dropout", "0.1"],
                       capture_output=True, text=True, timeout=240, env=e)
    nan = "NanLoss" in (r.stdout + r.stderr)
    print(tag, "rc", r.returncode, "NaN" if nan else "ok")
