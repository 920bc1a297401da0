"""Edge-list text → Vite binary converter.

The reference's CMake references a `converters/` directory that is absent
from its snapshot (CMakeLists.txt:106-107; the binary-format converters live
in upstream Vite). This minimal converter fills that gap so real-world
edge-list data can feed `-f`:

    python -m cuvite_amd.convert edges.txt graph.bin [--one-based]
        [--no-symmetrize] [--unit-weights]

Input: one `<u> <v> [w]` per line; `#`/`%` comment lines are skipped.
Output: the Vite binary layout (io.py): nv, ne, index[nv+1], {tail,w}[ne].
"""

from __future__ import annotations

import argparse

import numpy as np
import torch

from .graph import Graph
from .io import write_graph


def convert_edge_list(in_path: str, out_path: str, one_based: bool = False,
                      symmetrize: bool = True, unit_weights: bool = False,
                      nv: int = 0) -> Graph:
    rows = []
    with open(in_path) as f:
        for line in f:
            line = line.strip()
            if not line or line[0] in "#%":
                continue
            parts = line.split()
            u, v = int(parts[0]), int(parts[1])
            w = float(parts[2]) if len(parts) > 2 and not unit_weights else 1.0
            rows.append((u, v, w))
    if not rows:
        raise ValueError(f"no edges found in {in_path}")
    arr = np.array(rows, dtype=np.float64)
    u = arr[:, 0].astype(np.int64)
    v = arr[:, 1].astype(np.int64)
    w = arr[:, 2]
    if one_based:
        u -= 1
        v -= 1
    if u.min() < 0 or v.min() < 0:
        raise ValueError("negative vertex ids (forgot --one-based?)")
    n = int(max(u.max(), v.max())) + 1
    n = max(n, nv)
    if symmetrize:
        src = np.concatenate([u, v])
        dst = np.concatenate([v, u])
        ww = np.concatenate([w, w])
    else:
        src, dst, ww = u, v, w
    g = Graph.from_edge_tuples(n, torch.from_numpy(src),
                               torch.from_numpy(dst), torch.from_numpy(ww))
    write_graph(out_path, g)
    return g


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(
        prog="cuvite_amd.convert",
        description="Convert a text edge list to the Vite binary format")
    ap.add_argument("input")
    ap.add_argument("output")
    ap.add_argument("--one-based", action="store_true",
                    help="input vertex ids start at 1")
    ap.add_argument("--no-symmetrize", action="store_true",
                    help="input already lists both directions of every edge")
    ap.add_argument("--unit-weights", action="store_true",
                    help="ignore weight columns; all weights 1.0")
    ap.add_argument("--nv", type=int, default=0,
                    help="force the vertex count (isolated trailing vertices)")
    args = ap.parse_args(argv)
    g = convert_edge_list(args.input, args.output, one_based=args.one_based,
                          symmetrize=not args.no_symmetrize,
                          unit_weights=args.unit_weights, nv=args.nv)
    print(f"Wrote {args.output}: nv={g.nv} ne(directed)={g.ne}")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
