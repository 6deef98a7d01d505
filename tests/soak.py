"""Long-running randomized parity soak (not a pytest): drives the Dual
oracle/GPU differ with random mixed workloads for a wall-clock budget,
covering many fold boundaries, compactions, TTL sweeps, watches and streams.

Usage: python tests/soak.py [minutes]
"""
import os
import random
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import parity

parity.small_env()

NS = [b"/registry/pods/ns-%02d" % i for i in range(10)] + \
     [b"/registry/configmaps/ns-%02d" % i for i in range(5)] + \
     [b"/registry/events/ns-%02d" % i for i in range(3)]


def run_one(seed, steps, budget_deadline):
    rng = random.Random(seed)
    d = parity.Dual(events_ttl=5)
    ops = 0
    try:
        live = {}
        w = d.watch(b"/registry/", 0)
        for step in range(steps):
            if time.time() > budget_deadline:
                break
            op = rng.random()
            ns = rng.choice(NS)
            key = ns + b"/obj-%05d" % rng.randrange(120)
            if op < 0.34:
                r = d.create(key, b"c%d" % step)
                if r.succeeded:
                    live[key] = r.header_revision
            elif op < 0.58:
                prev = live.get(key, 0) if rng.random() < 0.7 else rng.randrange(1, 5000)
                r = d.update(key, b"u%d" % step * rng.randrange(1, 40), prev)
                if r.succeeded:
                    live[key] = r.header_revision
            elif op < 0.68:
                r = d.delete(key, live.get(key, 0) if rng.random() < 0.5 else 0)
                if r.succeeded:
                    live.pop(key, None)
            elif op < 0.82:
                rev = 0 if rng.random() < 0.5 else max(1, d.p.current_rev() - rng.randrange(200))
                d.list(ns + b"/", ns + b"0", rev, rng.choice([0, 1, 13, 120, 500]))
            elif op < 0.90:
                d.get(key, 0 if rng.random() < 0.5 else max(1, d.p.current_rev() - rng.randrange(100)))
            elif op < 0.94:
                d.count(ns + b"/", ns + b"0")
            elif op < 0.97:
                d.stream(ns + b"/", ns + b"0", 0)
            else:
                d.clock_advance(rng.randrange(3))
                d.compact(max(1, d.p.current_rev() - rng.randrange(200)))
            ops += 1
            if step % 500 == 499:
                d.poll(w)
                d.diff_dump()
        d.poll(w)
        d.diff_dump()
        d.diff_event_log()
    finally:
        d.close()
    return ops


def main():
    minutes = float(sys.argv[1]) if len(sys.argv) > 1 else 5.0
    deadline = time.time() + minutes * 60
    total, rounds = 0, 0
    seed = 1000
    while time.time() < deadline:
        total += run_one(seed, 100000, deadline)
        rounds += 1
        seed += 1
        print(f"[soak] round {rounds} done, total ops {total}", flush=True)
    print(f"SOAK PASS: {total} ops across {rounds} stores, all bit-exact",
          flush=True)


if __name__ == "__main__":
    main()
