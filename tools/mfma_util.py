#!/usr/bin/env python3
"""Compute the MFMA-busy ratio for rank_gemm dispatches from two rocprofv3
PMC passes (SQ_VALU_MFMA_BUSY_CYCLES and SQ_WAVE_CYCLES — separate passes,
deterministic kernel so dispatch k matches across passes; SQ_WAVE_CYCLES
counts quad-cycles, MICROARCH guide §counters). Prints one JSON line."""
import csv
import json
import sys


def per_kernel(path, kernel, counter):
    vals = []
    with open(path) as f:
        for row in csv.DictReader(f):
            if kernel in row["Kernel_Name"] and \
                    row["Counter_Name"] == counter:
                vals.append(float(row["Counter_Value"]))
    return vals


def main():
    busy_csv, wave_csv = sys.argv[1], sys.argv[2]
    kernel = sys.argv[3] if len(sys.argv) > 3 else "rank_gemm"
    busy = per_kernel(busy_csv, kernel, "SQ_VALU_MFMA_BUSY_CYCLES")
    wave = per_kernel(wave_csv, kernel, "SQ_WAVE_CYCLES")
    if not busy or not wave:
        print(json.dumps({"error": "no dispatches",
                          "busy_n": len(busy), "wave_n": len(wave)}))
        return
    mb = sum(busy) / len(busy)
    mw = sum(wave) / len(wave)
    out = {
        "kernel": kernel,
        "dispatches": [len(busy), len(wave)],
        "mean_mfma_busy_cycles": round(mb),
        "mean_wave_quad_cycles": round(mw),
        # the r01-comparable ratio (3.8% baseline): busy / wave_cycles
        "mfma_busy_over_wave_cycles": round(mb / mw, 4),
        # wave counter is quad-cycles; busy is cycles -> cycle-true ratio
        "mfma_busy_over_wave_cycles_x4": round(mb / (mw * 4), 4),
    }
    print(json.dumps(out))


if __name__ == "__main__":
    main()
