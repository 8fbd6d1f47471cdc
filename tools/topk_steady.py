"""Top individual kernels (name x grid) within the steady graph-replay window.

Usage: python tools/topk_steady.py <kernel_trace.csv> [steps] [topn]

Same Adam-boundary windowing as analyze_trace.py, but aggregated per
(kernel name, grid/workgroup dims) so the dominant *shapes* are visible —
the per-category averages hide bimodal mixes (e.g. 16-row scan LayerNorms
vs 1024-row imagination ones).
"""

import collections
import csv
import sys


def main() -> None:
    path = sys.argv[1]
    want_steps = int(sys.argv[2]) if len(sys.argv) > 2 else 8
    topn = int(sys.argv[3]) if len(sys.argv) > 3 else 30
    rows = []
    with open(path) as fh:
        rd = csv.DictReader(fh)
        name_k = next((k for k in rd.fieldnames if "name" in k.lower()), None)
        start_k = next((k for k in rd.fieldnames if "start" in k.lower()), None)
        end_k = next((k for k in rd.fieldnames if "end" in k.lower()), None)
        grid_ks = [k for k in rd.fieldnames if "grid" in k.lower()]
        wg_ks = [k for k in rd.fieldnames if "workgroup" in k.lower() or "block" in k.lower()]
        for r in rd:
            grid = "x".join(r[k] for k in grid_ks) if grid_ks else "?"
            wg = "x".join(r[k] for k in wg_ks) if wg_ks else "?"
            rows.append((int(r[start_k]), int(r[end_k]), r[name_k], grid, wg))
    rows.sort()
    adam_ts = [s for s, _, n, _, _ in rows if "adam" in n]
    bounds = adam_ts[2::3]
    if len(bounds) < want_steps + 1:
        print(f"only {len(bounds)} steps found; using all")
        want_steps = max(1, len(bounds) - 1)
    t0, t1 = bounds[-want_steps - 1], bounds[-1]
    agg = collections.Counter()
    cnt = collections.Counter()
    for s, e, n, grid, wg in rows:
        if t0 < s <= t1:
            nn = n.replace("(anonymous namespace)::", "").replace("void ", "")
            key = (nn.split("(")[0][:70], grid, wg)
            agg[key] += e - s
            cnt[key] += 1
    tot = sum(agg.values())
    print(f"steady window: {want_steps} steps, {tot/want_steps/1e6:.2f} ms kernel/step")
    for (n, grid, wg), v in agg.most_common(topn):
        c = cnt[(n, grid, wg)]
        print(
            f"{v/want_steps/1e6:8.3f} ms/step  {100*v/tot:5.1f}%  n/step={c/want_steps:6.1f}  "
            f"avg={v/c/1e3:7.2f} us  grid={grid:>16s} wg={wg:>12s}  {n}"
        )


if __name__ == "__main__":
    main()
