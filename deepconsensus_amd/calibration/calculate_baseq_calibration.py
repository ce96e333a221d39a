"""`deepconsensus calibrate`: empirical base-quality calibration stats.

Parity with reference quality_calibration/calculate_baseq_calibration.py:
per-interval match/mismatch counts per predicted base quality (0..MAX_BASEQ)
from aligned reads vs the reference via a cigar walk, parallelized over
interval stripes, CSV output (baseq,total_match,total_mismatch).

Deviation from the reference's mechanics (not its math): without an indexed
BAM, reads are bucketed by contig in one sequential scan and intervals then
select by reference_start overlap.
"""
from __future__ import annotations

import argparse
import collections
import logging
import multiprocessing
from typing import Dict, List, Optional

import numpy as np

from deepconsensus_amd.calibration import calibration as calibration_lib
from deepconsensus_amd.utils import constants

log = logging.getLogger(__name__)

MAX_BASEQ = 100

# Flags bits mirroring the reference's pysam skip conditions.
FDUP = 0x400
FQCFAIL = 0x200
FSECONDARY = 0x100


class RegionRecord:
    def __init__(self, contig: str, start: int, stop: int):
        self.contig = contig
        self.start = start
        self.stop = stop

    def __str__(self):
        return "[REGION: Contig= %s, Start= %d, Stop= %d]" % (
            self.contig, self.start, self.stop,
        )


def process_region_string(region_string: str, fasta_file: str) -> RegionRecord:
    from deepconsensus_amd.dcio.fasta import FastaFile

    if ":" in region_string:
        if len(region_string.split(":")) != 2:
            raise ValueError(f"Malformed region string {region_string}")
        contig, start_stop = region_string.split(":")
        if len(start_stop.split("-")) != 2:
            raise ValueError(f"Malformed region string {region_string}")
        start, stop = start_stop.split("-")
        rec = RegionRecord(contig, int(start), int(stop))
        if rec.start > rec.stop:
            raise ValueError(f"Malformed region string {region_string}")
        return rec
    fasta = FastaFile(fasta_file)
    if region_string not in fasta.references:
        raise ValueError(f"Contig {region_string} not found in fasta")
    return RegionRecord(
        region_string, 0, fasta.get_reference_length(region_string)
    )


def split_regions_in_intervals(
    regions: List[RegionRecord], region_length: int
) -> List[RegionRecord]:
    all_intervals = []
    for region in regions:
        for pos in range(region.start, region.stop, region_length):
            all_intervals.append(
                RegionRecord(
                    region.contig,
                    max(region.start, pos),
                    min(region.stop, pos + region_length),
                )
            )
    return all_intervals


def get_contig_regions(
    bam_file: str, fasta_file: str, region: Optional[str],
    interval_length: int,
) -> List[RegionRecord]:
    from deepconsensus_amd.dcio.bam import BamReader
    from deepconsensus_amd.dcio.fasta import FastaFile

    if region:
        regions = [process_region_string(region, fasta_file)]
    else:
        fasta = FastaFile(fasta_file)
        bam = BamReader(bam_file)
        bam_contigs = {name for name, _l in bam.header.references}
        regions = [
            RegionRecord(name, 0, fasta.get_reference_length(name))
            for name in fasta.references
            if name in bam_contigs
        ]
        bam.close()
    return split_regions_in_intervals(regions, interval_length)


def get_quality_calibration_stats(
    reads,
    ref_sequence: str,
    region_interval: RegionRecord,
    min_mapq: int,
    dc_calibration: calibration_lib.QualityCalibrationValues,
) -> List[Dict[str, int]]:
    """Cigar-walk M/X counting (reference :303-375)."""
    counts = [{"M": 0, "X": 0} for _ in range(MAX_BASEQ)]
    for read in reads:
        if read.flag & (FDUP | FQCFAIL | FSECONDARY) or read.is_unmapped:
            continue
        if read.is_supplementary or read.mapq < min_mapq:
            continue
        current_ref_pos = read.reference_start
        current_read_index = 0
        quals = np.asarray(read.query_qualities)
        if dc_calibration.enabled:
            fit = calibration_lib.calibrate_quality_scores(
                quals.astype(np.float64), dc_calibration
            )
            fit = np.round(fit, decimals=0).astype(np.int32)
        else:
            fit = quals
        for cigar_op, cigar_len in read.cigartuples:
            if current_ref_pos > region_interval.stop:
                break
            if cigar_op in (constants.CMATCH, constants.CDIFF,
                            constants.CEQUAL):
                for _ in range(cigar_len):
                    if (
                        region_interval.start
                        <= current_ref_pos
                        <= region_interval.stop
                    ):
                        idx = current_ref_pos - region_interval.start
                        if idx < len(ref_sequence):
                            ref_base = ref_sequence[idx].upper()
                            read_base = read.query_sequence[
                                current_read_index
                            ].upper()
                            q = int(fit[current_read_index])
                            if 0 <= q < MAX_BASEQ and ref_base in "ACGT":
                                key = "M" if ref_base == read_base else "X"
                                counts[q][key] += 1
                    current_read_index += 1
                    current_ref_pos += 1
            elif cigar_op in (constants.CSOFT_CLIP, constants.CINS):
                for _ in range(cigar_len):
                    if (
                        region_interval.start
                        <= current_ref_pos
                        <= region_interval.stop
                    ):
                        q = int(fit[current_read_index])
                        if 0 <= q < MAX_BASEQ:
                            counts[q]["X"] += 1
                    current_read_index += 1
            elif cigar_op in (constants.CREF_SKIP, constants.CDEL):
                current_ref_pos += cigar_len
    return counts


def calculate_quality_calibration(
    bam_file: str,
    fasta_file: str,
    process_intervals: List[RegionRecord],
    min_mapq: int,
    dc_calibration: str,
) -> List[Dict[str, int]]:
    from deepconsensus_amd.dcio.bam import BamReader
    from deepconsensus_amd.dcio.fasta import FastaFile

    fasta = FastaFile(fasta_file)
    # Bucket reads by contig (one sequential scan; no .bai needed).
    by_contig: Dict[str, list] = collections.defaultdict(list)
    for read in BamReader(bam_file):
        name = read.reference_name
        if name is not None:
            by_contig[name].append(read)
    calib = calibration_lib.parse_calibration_string(dc_calibration)
    main_dict = [{"M": 0, "X": 0} for _ in range(MAX_BASEQ)]
    for interval in process_intervals:
        ref_seq = fasta.fetch(
            interval.contig, interval.start, interval.stop + 5
        )
        reads = [
            r
            for r in by_contig.get(interval.contig, [])
            if r.reference_start <= interval.stop
        ]
        counts = get_quality_calibration_stats(
            reads, ref_seq, interval, min_mapq, calib
        )
        for i in range(MAX_BASEQ):
            main_dict[i]["M"] += counts[i]["M"]
            main_dict[i]["X"] += counts[i]["X"]
    return main_dict


def main(argv: Optional[List[str]] = None) -> None:
    ap = argparse.ArgumentParser("deepconsensus calibrate")
    ap.add_argument("--bam", required=True,
                    help="reads aligned to the reference")
    ap.add_argument("--ref", required=True, help="reference FASTA")
    ap.add_argument("--output_csv", required=True)
    ap.add_argument("--region", default=None, help="contig or contig:a-b")
    ap.add_argument("--interval_length", type=int, default=100000)
    ap.add_argument("--min_mapq", type=int, default=60)
    ap.add_argument("--cpus", type=int, default=multiprocessing.cpu_count())
    ap.add_argument("--dc_calibration", default="skip")
    args = ap.parse_args(argv)
    if args.cpus == 0:
        raise ValueError("Must set cpus to >=1 for processing.")

    all_intervals = get_contig_regions(
        args.bam, args.ref, args.region, args.interval_length
    )
    global_stats = [{"M": 0, "X": 0} for _ in range(MAX_BASEQ)]

    # Stripe intervals across workers (reference :250-267).
    stripes = [all_intervals[i::args.cpus] for i in range(args.cpus)]
    stripes = [s for s in stripes if s]
    if args.cpus == 1 or len(stripes) <= 1:
        results = [
            calculate_quality_calibration(
                args.bam, args.ref, s, args.min_mapq, args.dc_calibration
            )
            for s in stripes
        ]
    else:
        with multiprocessing.Pool(len(stripes)) as pool:
            results = pool.starmap(
                calculate_quality_calibration,
                [
                    (args.bam, args.ref, s, args.min_mapq,
                     args.dc_calibration)
                    for s in stripes
                ],
            )
    for res in results:
        for i in range(MAX_BASEQ):
            global_stats[i]["M"] += res[i]["M"]
            global_stats[i]["X"] += res[i]["X"]

    with open(args.output_csv, "w") as f:
        f.write("baseq,total_match,total_mismatch\n")
        for q in range(MAX_BASEQ):
            f.write(
                f"{q},{global_stats[q]['M']},{global_stats[q]['X']}\n"
            )
    print("Processing complete.")


if __name__ == "__main__":
    logging.basicConfig(level=logging.INFO)
    main()
