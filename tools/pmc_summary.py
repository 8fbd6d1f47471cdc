"""Summarize a rocprofv3 counter_collection.csv: per-kernel wait/active split
and MFMA instruction share (usage: pmc_summary.py <csv> [out])."""
import csv as _csv
import collections
import sys

path = sys.argv[1]
agg = collections.defaultdict(lambda: collections.defaultdict(float))
cnt = collections.Counter()
with open(path) as fh:
    for row in _csv.DictReader(fh):
        k = row["Kernel_Name"][:70]
        agg[k][row["Counter_Name"]] += float(row["Counter_Value"])
        if row["Counter_Name"] == "SQ_WAVE_CYCLES":
            cnt[k] += 1
out = sys.stdout if len(sys.argv) < 3 else open(sys.argv[2], "w")
print("kernel | dispatches | wait_parked% | issue_stall% | active_inst% | mfma_insts/disp", file=out)
rows = sorted(agg.items(), key=lambda kv: -kv[1]["SQ_WAVE_CYCLES"])
for k, c in rows[:28]:
    wc = c["SQ_WAVE_CYCLES"] or 1.0
    print(f"{k} | {cnt[k]} | {100*c['SQ_WAIT_ANY']/wc:.0f} | {100*c['SQ_WAIT_INST_ANY']/wc:.0f} | "
          f"{100*c['SQ_ACTIVE_INST_ANY']/wc:.0f} | {c['SQ_INSTS_MFMA']/max(cnt[k],1):.0f}", file=out)

print("\n-- steady-state kernels of interest --", file=out)
import re
pat = re.compile(r"Cijk|ln_act|gru_gates|cat_st|scan3|adam_mt|replay_gather|nll_|reinforce|vloss")
for k, c in rows:
    if pat.search(k):
        wc = c["SQ_WAVE_CYCLES"] or 1.0
        print(f"{k} | {cnt[k]} | {100*c['SQ_WAIT_ANY']/wc:.0f} | {100*c['SQ_WAIT_INST_ANY']/wc:.0f} | "
              f"{100*c['SQ_ACTIVE_INST_ANY']/wc:.0f} | {c['SQ_INSTS_MFMA']/max(cnt[k],1):.0f}", file=out)
