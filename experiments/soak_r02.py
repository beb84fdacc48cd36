#!/usr/bin/env python3
"""Round-2 soak: continuous mixed-mode rolling upgrades on the CACHED
substrate (informers -> REST -> HTTP apiserver) with interleaved native GPU
health checks.  Round 1 soaked the in-process store; this soaks the
production-shaped stack, including watch streams, RV barriers and the
keep-alive transport, for --seconds of wall time.

Usage: python experiments/soak_r02.py --seconds 300 [--deep-every 10]
Writes a one-line JSON summary to stdout and progress to stderr.
"""

import argparse
import json
import statistics
import sys
import time


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=float, default=300.0)
    ap.add_argument("--nodes", type=int, default=8)
    ap.add_argument("--deep-every", type=int, default=10)
    ap.add_argument("--no-gpu", action="store_true")
    args = ap.parse_args()

    sys.path.insert(0, ".")
    import bench
    from k8s_operator_libs_amd.validation import gpu_health_check

    t_end = time.monotonic() + args.seconds
    walls = {"inplace": [], "requestor": [], "anic": []}
    upgrades = 0
    checks = 0
    healthy = 0
    failures = []
    modes = ["inplace", "requestor", "inplace", "anic"]
    i = 0
    while time.monotonic() < t_end:
        mode = modes[i % len(modes)]
        i += 1
        try:
            r = bench.run_rolling_upgrade_benchmark(
                n_nodes=args.nodes, steps=3, warmup=0,
                gpu_validate=not args.no_gpu, mode=mode, substrate="cached",
            )
            walls[mode].extend(r["wall_times"])
            upgrades += r["upgrades_completed"]
        except Exception as exc:  # noqa - soak must record, not die
            failures.append(f"{mode}: {exc}")
            if len(failures) > 5:
                break
        if not args.no_gpu:
            try:
                deep = (i % args.deep_every) == 0
                rep = gpu_health_check(require_gpu=True, deep=deep)
                checks += 1
                healthy += 1 if rep["healthy"] else 0
                if not rep["healthy"]:
                    failures.append(f"health: {rep}")
            except Exception as exc:
                failures.append(f"health: {exc}")
        if i % 10 == 0:
            print(f"[soak] {upgrades} upgrades, {checks} checks, "
                  f"{round(t_end - time.monotonic())}s left", file=sys.stderr)

    def stats(xs):
        if not xs:
            return None
        xs = sorted(xs)
        return {"n": len(xs), "mean_ms": round(statistics.mean(xs) * 1000, 2),
                "p99_ms": round(xs[int(len(xs) * 0.99)] * 1000, 2),
                "max_ms": round(xs[-1] * 1000, 2)}

    print(json.dumps({
        "soak_seconds": args.seconds,
        "upgrades_completed": upgrades,
        "gpu_checks": checks, "gpu_checks_healthy": healthy,
        "failures": failures,
        "substrate": "cached",
        "walls": {k: stats(v) for k, v in walls.items()},
    }))
    return 1 if failures else 0


if __name__ == "__main__":
    sys.exit(main())
