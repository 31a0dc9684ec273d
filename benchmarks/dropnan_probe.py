# A/B: dropout bench divergence — graphs on vs off, with NaN warnings logged
import logging, os, sys, json, subprocess
for graphs in ("on", "off"):
    env = dict(os.environ)
    env["ADANET_LOG"] = "WARNING"
    env["ADANET_NO_GRAPHS"] = "1" if graphs == "off" else ""
    r = subprocess.run(
        [sys.executable, "bench.py", "--gpus", "1", "--steps", "2",
         "--warmup", "1", "--dropout", "0.1"],
        capture_output=True, text=True, timeout=300, env=env)
    tail = (r.stdout + r.stderr)[-600:]
    print("graphs", graphs, "rc", r.returncode)
    print(tail)
