"""Summarize a rocprofv3 counter_collection CSV: per-kernel mean of each
counter + dispatch count (profiles/ evidence helper)."""
import glob
import sys

import pandas as pd

pattern, out = sys.argv[1], sys.argv[2]
files = glob.glob(pattern)
assert files, f"no counter files match {pattern}"
df = pd.concat([pd.read_csv(f) for f in files])
kcol = "Kernel_Name" if "Kernel_Name" in df.columns else "Kernel Name"
ccol = "Counter_Name" if "Counter_Name" in df.columns else "Counter Name"
vcol = "Counter_Value" if "Counter_Value" in df.columns else "Counter Value"
df["kernel"] = df[kcol].str.slice(0, 100)
g = (
    df.groupby(["kernel", ccol])[vcol]
    .agg(["mean", "count"])
    .reset_index()
    .rename(columns={ccol: "counter"})
)
g["mean"] = g["mean"].round(2)
g.sort_values(["kernel", "counter"]).to_csv(out, index=False)
print(f"wrote {out}: {len(g)} rows, kernels={df['kernel'].nunique()}")
