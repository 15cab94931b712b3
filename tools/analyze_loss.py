"""Offline training-log analysis.

Parity: the reference greps '***Test:' lines from captured stdout and
plots the loss curve (/root/reference/analyze_test_loss.py:12-24), and
check_loss.py re-computes the warp loss in numpy.  Here training writes
structured JSONL (logs/<run>/metrics.jsonl); this tool summarizes it
and plots curves when matplotlib is available.

    python tools/analyze_loss.py logs/<run>/metrics.jsonl
"""

import json
import sys


def main(path):
    steps, totals, aees, imgs = [], [], [], []
    with open(path) as f:
        for line in f:
            rec = json.loads(line)
            if "total" in rec:
                steps.append(rec.get("step", len(steps)))
                totals.append(rec["total"])
                imgs.append(rec.get("imgs_per_sec"))
            if "aee" in rec:
                aees.append((rec.get("epoch"), rec["aee"]))

    if totals:
        n = len(totals)
        print(f"{n} loss records: first={totals[0]:.4f} "
              f"last={totals[-1]:.4f} min={min(totals):.4f}")
        valid_imgs = [x for x in imgs if x]
        if valid_imgs:
            print(f"imgs/sec: median="
                  f"{sorted(valid_imgs)[len(valid_imgs)//2]:.1f}")
    for ep, aee in aees:
        print(f"***Test: epoch {ep} AEE {aee:.4f}")

    try:
        import matplotlib

        matplotlib.use("Agg")
        import matplotlib.pyplot as plt

        fig, ax = plt.subplots()
        ax.plot(steps, totals)
        ax.set_xlabel("step")
        ax.set_ylabel("total loss")
        out = path.replace(".jsonl", "_loss.png")
        fig.savefig(out)
        print(f"wrote {out}")
    except ImportError:
        out = path.replace(".jsonl", "_loss.csv")
        with open(out, "w") as f:
            f.write("step,total\n")
            for s, t in zip(steps, totals):
                f.write(f"{s},{t}\n")
        print(f"matplotlib unavailable; wrote {out}")


if __name__ == "__main__":
    main(sys.argv[1])
