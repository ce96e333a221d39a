"""Synthetic BAM corpus generators (shared by benches and the yield
harness): chem2.2-shaped subread stacks with mutations + insertions so
the gap-aware spacing machinery does real work."""
from __future__ import annotations

import os

import numpy as np

from deepconsensus_amd.dcio import bam as bam_lib


def make_synth_bams(out_dir, n_zmws, length, n_subreads, seed,
                    err_rate=0.005, n_ins=1, ccs_err_rate=0.0):
    """err_rate: per-base substitution rate per subread; n_ins: short
    insertions per subread; ccs_err_rate: substitution rate of the
    draft CCS vs truth (a nonzero value makes the consensus task
    require more than copying the CCS input row)."""
    rng = np.random.default_rng(seed)
    refs, zmw_seqs = [], {}
    for z in range(n_zmws):
        name = f"m000/{z + 10}/ccs"
        seq = "".join(rng.choice(list("ATCG"), size=length))
        refs.append((name, length))
        zmw_seqs[name] = seq
    header = bam_lib.BamHeader(text="@HD\tVN:1.6", references=refs)

    sub_path = os.path.join(out_dir, "subreads_to_ccs.bam")
    with bam_lib.BamWriter(sub_path, header) as w:
        for rid, (name, ln) in enumerate(refs):
            zm = int(name.split("/")[1])
            seq = zmw_seqs[name]
            for i in range(n_subreads):
                # Mutate err_rate of bases and add n_ins short
                # insertions so the multi-read spacing has actual gap
                # columns to create.
                s = list(seq)
                n_mut = max(int(ln * err_rate), 1)
                for p in rng.integers(0, ln, n_mut):
                    s[p] = rng.choice(list("ATCG"))
                cuts = np.sort(rng.integers(1, ln - 1, size=n_ins))
                full_parts, cig, prev = [], [], 0
                for cpos in cuts:
                    cpos = int(cpos)
                    if cpos <= prev:
                        continue
                    ins_len = int(rng.integers(1, 4))
                    ins = "".join(rng.choice(list("ATCG"), size=ins_len))
                    full_parts.append("".join(s[prev:cpos]))
                    cig.append((0, cpos - prev))
                    full_parts.append(ins)
                    cig.append((1, ins_len))
                    prev = cpos
                full_parts.append("".join(s[prev:]))
                cig.append((0, ln - prev))
                full = "".join(full_parts)
                n = len(full)
                w.write(bam_lib.BamRead(
                    qname=f"m000/{zm}/{i * (ln + 50)}_{i * (ln + 50) + n}",
                    flag=16 if i % 2 else 0,
                    ref_id=rid, pos=0, mapq=60, cigartuples=cig, seq=full,
                    query_qualities=[30] * n,
                    tags={
                        "zm": zm,
                        "pw": rng.integers(0, 60, n).astype(np.uint8),
                        "ip": rng.integers(0, 60, n).astype(np.uint8),
                        "sn": np.array([6.0, 7.0, 5.5, 9.1], np.float32),
                    },
                ))

    ccs_path = os.path.join(out_dir, "ccs.bam")
    with bam_lib.BamWriter(ccs_path, header) as w:
        for rid, (name, ln) in enumerate(refs):
            zm = int(name.split("/")[1])
            ccs_seq = zmw_seqs[name]
            if ccs_err_rate > 0:
                cs = list(ccs_seq)
                for p in rng.integers(0, ln, max(int(ln * ccs_err_rate),
                                                 1)):
                    cs[p] = rng.choice(list("ATCG"))
                ccs_seq = "".join(cs)
            w.write(bam_lib.BamRead(
                qname=name, flag=4, ref_id=-1, pos=-1, cigartuples=[],
                seq=ccs_seq,
                query_qualities=rng.integers(20, 40, ln),
                tags={"zm": zm, "ec": 11.5, "np": n_subreads, "rq": 0.998,
                      "RG": "rg0"},
            ))
    return sub_path, ccs_path, zmw_seqs


