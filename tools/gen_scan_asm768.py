#!/usr/bin/env python3
"""Generator for matrixone_amd/csrc/scan_asm768.hip — the hand-scheduled
flagship list-scan kernel (METRIC=L2SQ, dpad=768, QT=16).

The kernel's 192-quad inner loop is one inline-asm block owning both
hardware counters (8-deep global-load ring at `s_waitcnt vmcnt(7)`,
double-buffered LDS query reads at `s_waitcnt lgkmcnt(1)`); its ~2700 asm
lines are emitted by this printer so the schedule stays editable at the
rotation/step level. Regenerate with:

    python tools/gen_scan_asm768.py > matrixone_amd/csrc/scan_asm768.hip

and diff before committing; the checked-in file is the source of truth the
build uses (no codegen at build time)."""

ROT_OFFSETS = [6144, 9216, 12288, 15360, 18432, 21504, 24576, 27648,
               30720, 33792, 36864, 39936, 43008, 46080]


def st(acc, buf, x0, x1, x2, x3, tail):
    q = ["v72", "v73", "v74", "v75"] if buf == 0 else ["v76", "v77", "v78",
                                                       "v79"]
    lines = ['        "s_waitcnt lgkmcnt(1)\\n\\t"']
    for t, qq, xx in zip(["v80", "v81", "v82", "v83"], q, [x0, x1, x2, x3]):
        lines.append(f'        "v_sub_f32 {t}, {xx}, {qq}\\n\\t"')
    for t in ["v80", "v81", "v82", "v83"]:
        lines.append(f'        "v_fmac_f32 %[{acc}], {t}, {t}\\n\\t"')
    lines += tail
    return lines


def rot(waitc, regs, reissue):
    x0, x1, x2, x3 = regs
    out = [f'        "s_waitcnt vmcnt({waitc})\\n\\t"']
    for t in range(16):
        buf = t % 2
        if t < 14:
            nxt = ROT_OFFSETS[t]
            tgt = "v[72:75]" if buf == 0 else "v[76:79]"
            tail = [f'        "ds_read_b128 {tgt}, v86 offset:{nxt}\\n\\t"']
            if t == 13:
                tail.append('        "v_add_u32 v86, 16, v86\\n\\t"')
        elif t == 14:
            tail = ['        "ds_read_b128 v[72:75], v86\\n\\t"']
        else:
            tail = ['        "ds_read_b128 v[76:79], v86 offset:3072\\n\\t"']
        out += st(f"a{t}", buf, x0, x1, x2, x3, tail)
    if reissue:
        rng = f"v[{x0[1:]}:{x3[1:]}]"
        out += [f'        "global_load_dwordx4 {rng}, v[84:85], off\\n\\t"',
                '        "v_add_co_u32 v84, vcc, 0x400, v84\\n\\t"',
                '        "v_addc_co_u32 v85, vcc, 0, v85, vcc\\n\\t"']
    return out


SLOTS = [("v40", "v41", "v42", "v43"), ("v44", "v45", "v46", "v47"),
         ("v48", "v49", "v50", "v51"), ("v52", "v53", "v54", "v55"),
         ("v56", "v57", "v58", "v59"), ("v60", "v61", "v62", "v63"),
         ("v64", "v65", "v66", "v67"), ("v68", "v69", "v70", "v71")]


def emit_asm():
    asm = []
    asm.append('        /* drain compiler-issued traffic so counted waits '
               'start at 0 */')
    asm.append('        "s_waitcnt vmcnt(0) lgkmcnt(0)\\n\\t"')
    asm.append('        "v_mov_b32 v84, %[alo]\\n\\t"')
    asm.append('        "v_mov_b32 v85, %[ahi]\\n\\t"')
    asm.append('        "v_mov_b32 v86, %[lbase]\\n\\t"')
    for s4 in SLOTS:
        rng = f"v[{s4[0][1:]}:{s4[3][1:]}]"
        asm.append(f'        "global_load_dwordx4 {rng}, v[84:85], off\\n\\t"')
        asm.append('        "v_add_co_u32 v84, vcc, 0x400, v84\\n\\t"')
        asm.append('        "v_addc_co_u32 v85, vcc, 0, v85, vcc\\n\\t"')
    asm.append('        "ds_read_b128 v[72:75], v86\\n\\t"')
    asm.append('        "ds_read_b128 v[76:79], v86 offset:3072\\n\\t"')
    asm.append('        "MOANN_L%=:\\n\\t"')
    for s4 in SLOTS:
        asm += rot(7, s4, True)
    asm.append('        "s_sub_u32 %[cnt], %[cnt], 1\\n\\t"')
    asm.append('        "s_cmp_lg_u32 %[cnt], 0\\n\\t"')
    asm.append('        "s_cbranch_scc1 MOANN_L%=\\n\\t"')
    for w, s4 in enumerate(SLOTS):
        asm += rot(7 - w, s4, False)
    asm.append('        "s_waitcnt vmcnt(0) lgkmcnt(0)\\n\\t"')
    return "\n".join(asm)


def main():
    import os
    here = os.path.dirname(os.path.abspath(__file__))
    path = os.path.join(here, "..", "matrixone_amd", "csrc",
                        "scan_asm768.hip")
    cur = open(path).read()
    body = emit_asm()
    start = cur.index('        /* drain compiler-issued traffic')
    end = cur.index('            : [a0] "+v"(a0)')
    new = cur[:start] + body + "\n" + cur[end:]
    if new == cur:
        print("scan_asm768.hip asm body is up to date")
    else:
        open(path, "w").write(new)
        print("scan_asm768.hip asm body regenerated — rebuild and re-run "
              "the asm768 parity test")


if __name__ == "__main__":
    main()
