"""Steady-state kernel-time breakdown from a rocprofv3 kernel trace CSV.

Usage: python tools/analyze_trace.py <kernel_trace.csv> [steps]

Segments the timeline by the device-side Adam optimizer kernels (3 calls per
DV3 train step: world model / actor / critic), takes the last `steps` full
steps, and prints per-category kernel time scaled to ms/step — excluding the
MIOpen find phase and warmup that dominate whole-process --stats output.
"""

import collections
import csv
import sys


def category(n: str) -> str:
    if "naive_conv" in n:
        return "miopen_naive(find)"
    if "Cijk" in n:
        return "gemm(hipblaslt)"
    if ("conv" in n.lower() and ("ck" in n or "xdl" in n or "igemm" in n)) or "GridwiseGemm_xdl_cshuffle_conv" in n:
        return "conv(CK)"
    if "gemm" in n.lower():
        return "gemm(other)"
    for k in ("ln_act", "gru_gates", "cat_st", "masked_lerp", "adam", "ema",
              "obs_norm", "symlog", "symexp", "lambda_scan", "gae", "step_inc"):
        if k in n:
            return "ours:" + k
    for k in ("elementwise", "reduce", "copy", "fill", "index", "philox", "cat"):
        if k in n.lower():
            return k
    return "other"


def main() -> None:
    path = sys.argv[1]
    want_steps = int(sys.argv[2]) if len(sys.argv) > 2 else 8
    rows = []
    with open(path) as fh:
        rd = csv.DictReader(fh)
        name_k = next((k for k in rd.fieldnames if "name" in k.lower()), None)
        start_k = next((k for k in rd.fieldnames if "start" in k.lower()), None)
        end_k = next((k for k in rd.fieldnames if "end" in k.lower()), None)
        if not (name_k and start_k and end_k):
            print("columns:", rd.fieldnames)
            return
        for r in rd:
            rows.append((int(r[start_k]), int(r[end_k]), r[name_k]))
    rows.sort()
    adam_ts = [s for s, _, n in rows if "adam" in n]
    # 3 adam launches per train step; boundaries at every 3rd
    bounds = adam_ts[2::3]
    if len(bounds) < want_steps + 1:
        print(f"only {len(bounds)} steps found; using all")
        want_steps = max(1, len(bounds) - 1)
    t0, t1 = bounds[-want_steps - 1], bounds[-1]
    agg = collections.Counter()
    cnt = collections.Counter()
    for s, e, n in rows:
        if t0 < s <= t1:
            c = category(n)
            agg[c] += e - s
            cnt[c] += 1
    tot = sum(agg.values())
    span = t1 - t0
    print(f"steady window: {want_steps} steps, {span/1e6:.2f} ms wall, "
          f"{tot/1e6:.2f} ms kernel time ({span/want_steps/1e6:.2f} ms/step wall)")
    for k, v in agg.most_common(25):
        print(f"{k:22s} {v/want_steps/1e6:8.3f} ms/step  {100*v/tot:5.1f}%  calls/step={cnt[k]/want_steps:.0f}")


if __name__ == "__main__":
    main()
