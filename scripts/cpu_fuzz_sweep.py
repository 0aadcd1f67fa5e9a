"""Long CPU fuzz sweep: the distributed random-expression fuzz
(tests/test_distributed_fuzz.py) across many seeds and world sizes on
gloo. Free extended validation of the engine's distributed paths.

Usage: python scripts/cpu_fuzz_sweep.py [--rounds 20]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.join(os.path.dirname(
    os.path.dirname(os.path.abspath(__file__))), "tests"))

import torch.multiprocessing as mp  # noqa

import test_distributed_fuzz as tdf  # noqa


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rounds", type=int, default=20)
    args = ap.parse_args()
    port = 29890
    t0 = time.time()
    for r in range(args.rounds):
        seed = 1000 + 13 * r
        os.environ["MATREL_FUZZ_SEED"] = str(seed)
        for world in (2, 4):
            port += 1
            mp.spawn(tdf._worker, args=(world, port, 25), nprocs=world,
                     join=True)
            print(f"round {r} seed {seed} world {world}: 25 trees ok "
                  f"({time.time()-t0:.0f}s)", flush=True)
    print(f"SWEEP PASS: {args.rounds * 2 * 25} trees")


if __name__ == "__main__":
    main()
