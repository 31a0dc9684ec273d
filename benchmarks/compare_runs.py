"""Compare two bench runs for determinism: winner sequence, ensemble
sizes, and final ensemble accuracy must match exactly.

python benchmarks/compare_runs.py runA.log runA_iters.json \
                                  runB.log runB_iters.json
"""

import json
import sys


def last_json_line(path):
    out = None
    for line in open(path):
        line = line.strip()
        if line.startswith("{"):
            try:
                d = json.loads(line)
                if "metric" in d:
                    out = d
            except ValueError:
                pass
    return out


def main():
    loga, itera, logb, iterb = sys.argv[1:5]
    a, b = last_json_line(loga), last_json_line(logb)
    ca, cb = json.load(open(itera))["curve"], json.load(open(iterb))["curve"]
    wa = [(e.get("winner"), e.get("ensemble_size")) for e in ca]
    wb = [(e.get("winner"), e.get("ensemble_size")) for e in cb]
    ok = True
    if wa != wb:
        ok = False
        print("MISMATCH winners:")
        for i, (x, y) in enumerate(zip(wa, wb)):
            flag = "  <-- differs" if x != y else ""
            print(" iter %d: %s vs %s%s" % (i, x, y, flag))
    acc_a = a["config"]["final_ensemble_accuracy"]
    acc_b = b["config"]["final_ensemble_accuracy"]
    if acc_a != acc_b:
        ok = False
        print("MISMATCH accuracy: %r vs %r" % (acc_a, acc_b))
    print(json.dumps({
        "deterministic": ok,
        "winners_match": wa == wb,
        "accuracy": [acc_a, acc_b],
        "value": [round(a["value"], 1), round(b["value"], 1)],
        "n_iters_compared": min(len(wa), len(wb)),
    }))
    sys.exit(0 if ok else 1)


if __name__ == "__main__":
    main()
