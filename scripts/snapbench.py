#!/usr/bin/env python3
"""BASELINE config 4: snapshot diff+merge of an N-GiB random-byte region on
the gfx950 kernels. Usage: python scripts/snapbench.py [GiB] [dirty_pct]"""

import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from faabric_amd import _core


def main():
    gib = float(sys.argv[1]) if len(sys.argv) > 1 else 4.0
    dirty_pct = float(sys.argv[2]) if len(sys.argv) > 2 else 25.0
    iters = int(sys.argv[3]) if len(sys.argv) > 3 else 5
    size = int(gib * (1 << 30))

    res = _core.bench_snapshot_pipeline(
        size, iters=iters, warmup=2, dirty_pct=dirty_pct
    )
    out = {k: res[k] for k in (
        "bytes", "n_pages", "dirty_pages", "diff_ms", "apply_ms",
        "diff_gbps", "apply_gbps", "pipeline_gbps",
    )}
    out["gib"] = gib
    out["dirty_pct"] = dirty_pct
    print(json.dumps(out))


if __name__ == "__main__":
    main()
