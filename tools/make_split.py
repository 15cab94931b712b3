"""Generate a train/val split file in the reference's format.

The reference ships canonical split files (FlyingChairs_train_val.txt:
22,872 lines of 1=train/2=val; Sintel_train_val.txt: 1,041 lines —
SURVEY §2.1 #29) that arrive with the datasets.  This tool produces a
compatible file for new or synthetic data:

    python tools/make_split.py 22872 --val 640 --seed 0 > split.txt
"""

import argparse

import numpy as np


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("count", type=int, help="total samples")
    ap.add_argument("--val", type=int, default=640, help="val samples")
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()
    rng = np.random.default_rng(args.seed)
    labels = np.ones(args.count, dtype=np.int64)
    labels[rng.choice(args.count, size=args.val, replace=False)] = 2
    print("\n".join(map(str, labels)))


if __name__ == "__main__":
    main()
