"""`deepconsensus filter_reads`: FASTQ/BAM -> FASTQ filtered by avg quality.

Parity with reference quality_calibration/filter_reads.py:84-131 including
the 5-decimal rounding of the probability-space average Phred.
"""
from __future__ import annotations

import argparse
import logging
import math
from typing import List, Optional, Sequence

log = logging.getLogger(__name__)


def avg_phred(base_qualities: Sequence[float]) -> float:
    if base_qualities is None or len(base_qualities) == 0:
        return 0
    return -10 * math.log10(
        sum(10 ** (i / -10) for i in base_qualities)
        / int(len(base_qualities))
    )


def filter_bam_or_fastq_by_quality(
    input_seq: str, output_fastq: str, quality_threshold: int
) -> None:
    from deepconsensus_amd.dcio import bam as bam_lib
    from deepconsensus_amd.dcio import fastq as fastq_lib
    from deepconsensus_amd.utils import phred as phred_lib

    out = open(output_fastq, "w")
    total_reads = 0
    total_reads_above_q = 0
    if input_seq.endswith(".bam"):
        for read in bam_lib.BamReader(input_seq):
            total_reads += 1
            p = round(avg_phred(read.query_qualities), 5)
            if p >= quality_threshold:
                total_reads_above_q += 1
                qual = phred_lib.quality_scores_to_string(
                    read.query_qualities
                )
                out.write(
                    "\n".join(
                        ["@" + read.qname, read.query_sequence, "+", qual]
                    )
                    + "\n"
                )
    else:
        for rec in fastq_lib.read_fastq(input_seq):
            total_reads += 1
            p = round(avg_phred(rec.get_quality_array()), 5)
            if p >= quality_threshold:
                total_reads_above_q += 1
                out.write(str(rec) + "\n")
    out.close()
    log.info("TOTAL READS IN INPUT: %d", total_reads)
    log.info("TOTAL READS IN OUTPUT: %d", total_reads_above_q)
    log.info("TOTAL FILTERED READS: %d", total_reads - total_reads_above_q)


def main(argv: Optional[List[str]] = None) -> None:
    ap = argparse.ArgumentParser("deepconsensus filter_reads")
    ap.add_argument("--input_seq", "-i", required=True)
    ap.add_argument("--output_fastq", "-o", required=True)
    ap.add_argument("--quality_threshold", "-q", type=int, required=True)
    args = ap.parse_args(argv)
    filter_bam_or_fastq_by_quality(
        args.input_seq, args.output_fastq, args.quality_threshold
    )


if __name__ == "__main__":
    logging.basicConfig(level=logging.INFO)
    main()
