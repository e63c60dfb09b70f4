"""Summarize rocprofv3 kernel_stats CSVs into a readable hot-kernel table."""
import csv
import re
import sys


def classify(name):
    if "flush_icache" in name:
        return "tunableop-tuning-flush"
    if name.startswith("Cijk"):
        mt = re.search(r"MT(\d+x\d+x\d+)", name)
        return f"hipBLASLt GEMM {mt.group(1) if mt else ''}"
    if name.startswith("npf_"):
        return name.split("(")[0] + " [npf HIP]"
    for pat, label in [
        ("rocblas", "rocBLAS"),
        ("miopen", "MIOpen conv"),
        ("Miopen", "MIOpen conv"),
        ("elementwise", "torch elementwise"),
        ("reduce_kernel", "torch reduce"),
        ("CatArrayBatched", "torch cat"),
        ("index_elementwise", "torch index"),
        ("adam", "fused Adam"),
        ("multi_tensor", "multi-tensor apply"),
        ("philox", "RNG"),
        ("distribution_", "RNG"),
    ]:
        if pat in name:
            return label
    return name[:60]


def summarize(path, topn=30):
    rows = list(csv.DictReader(open(path)))
    agg = {}
    for r in rows:
        key = classify(r["Name"])
        a = agg.setdefault(key, [0, 0.0])
        a[0] += int(r["Calls"])
        a[1] += float(r["TotalDurationNs"])
    total = sum(v[1] for v in agg.values())
    out = [f"== {path}", f"   {len(rows)} distinct kernels, total GPU time {total/1e6:.1f} ms"]
    for key, (calls, ns) in sorted(agg.items(), key=lambda kv: -kv[1][1])[:topn]:
        out.append(f"  {100*ns/total:5.2f}%  {calls:8d}x  avg {ns/calls/1e3:7.2f}us  {key}")
    return "\n".join(out)


if __name__ == "__main__":
    for p in sys.argv[1:]:
        print(summarize(p))
        print()
