#!/usr/bin/env python3
"""Top-N report over local-store pprof files — offline analog of
`pprof -top` for nodes without a Parca server.

    python tools/report.py /tmp/profiles [--type samples] [-n 20]
                                         [--cum] [--by-stack]

Aggregates every `*.<type>.pb.gz` in the directory: flat (leaf) or
cumulative value per function, or whole-stack rollups with --by-stack.
"""

import argparse
import glob
import os
import sys
from collections import Counter

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

from parca_agent_amd.pprof.profile import decode_profile  # noqa: E402


def _s(p, v):
    """pprof fields may be pre-resolved strings or string-table ids."""
    return p.strings[v] if isinstance(v, int) and 0 <= v < len(p.strings) \
        else (v if isinstance(v, str) else "")


def frame_name(p, loc_id):
    loc = p.locations.get(loc_id, {})
    for ln in loc.get("lines", []):
        fn = p.functions.get(ln["function_id"])
        if fn:
            name = _s(p, fn.get("name", ""))
            if name:
                return name
    m = p.mappings.get(loc.get("mapping_id", 0))
    base = os.path.basename(_s(p, m.get("filename", ""))) if m else ""
    return f"{base or '??'}+0x{loc.get('address', 0):x}"


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("directory")
    ap.add_argument("--type", default="samples",
                    help="profile type suffix (samples, gpu_kernel_time, "
                         "wallclock, probe, ...)")
    ap.add_argument("-n", type=int, default=20)
    ap.add_argument("--cum", action="store_true",
                    help="cumulative (anywhere in stack) instead of leaf")
    ap.add_argument("--by-stack", action="store_true",
                    help="rank whole stacks instead of functions")
    ap.add_argument("--folded", action="store_true",
                    help="emit Brendan-Gregg folded stacks "
                         "(pipe into flamegraph.pl)")
    args = ap.parse_args()

    files = sorted(glob.glob(
        os.path.join(args.directory, f"*.{args.type}.pb.gz")))
    if not files:
        print(f"no *.{args.type}.pb.gz under {args.directory}",
              file=sys.stderr)
        return 1
    flat: Counter = Counter()
    unit = ""
    total = 0
    for f in files:
        p = decode_profile(open(f, "rb").read())
        if p.sample_types:
            unit = p.sample_types[0].unit
        for s in p.samples:
            v = s["values"][0]
            total += v
            ids = s["location_ids"]
            if args.by_stack or args.folded:
                key = ";".join(
                    frame_name(p, lid) for lid in reversed(ids))
                flat[key] += v
            elif args.cum:
                for name in {frame_name(p, lid) for lid in ids}:
                    flat[name] += v
            elif ids:
                flat[frame_name(p, ids[0])] += v
    if args.folded:
        for name, v in flat.most_common():
            print(f"{name} {v}")
        return 0
    mode = "stack" if args.by_stack else ("cum" if args.cum else "flat")
    print(f"{len(files)} files, total {total} {unit}; top {args.n} "
          f"({mode}):")
    width = len(str(flat.most_common(1)[0][1])) if flat else 1
    for name, v in flat.most_common(args.n):
        pct = 100.0 * v / total if total else 0.0
        print(f"{v:{width}d}  {pct:5.1f}%  {name}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
