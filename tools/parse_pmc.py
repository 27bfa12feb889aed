import csv, glob, sys
agg, ns = {}, 0
for fn in sorted(glob.glob(sys.argv[1] + "/**/*counter_collection.csv", recursive=True)):
    for row in csv.DictReader(open(fn)):
        kn = row.get("Kernel_Name", "")
        if "gemm" in kn or "fill" not in kn:
            agg[row["Counter_Name"]] = agg.get(row["Counter_Name"], 0) + float(row["Counter_Value"])
            ns += 1
for k, v in sorted(agg.items()):
    print(f"{k:30s} {v:.4e}")
w = agg.get("SQ_WAVE_CYCLES")
if w:
    for k in ("SQ_WAIT_ANY", "SQ_WAIT_INST_ANY", "SQ_ACTIVE_INST_ANY", "SQ_LDS_BANK_CONFLICT"):
        if k in agg: print(f"{k}/WAVE_CYCLES = {agg[k]/w*100:.1f}%")
    if "SQ_VALU_MFMA_BUSY_CYCLES" in agg:
        print(f"MFMA_BUSY/(4*WAVE_CYCLES quad) = {agg['SQ_VALU_MFMA_BUSY_CYCLES']/(4*w)*100:.1f}%")
