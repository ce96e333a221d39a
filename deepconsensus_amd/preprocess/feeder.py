"""ZMW feeding: subread grouping, CCS construction, truth labels.

Behavioral parity with reference pre_lib.py:50-91 (SubreadGrouper),
:965-998 (construct_ccs_read), :1001-1058 (fetch_label_alignment,
read_truth_bedfile, read_truth_split), :1279-1367 (create_proc_feeder) and
:1370-1384 (subreads_to_dc_example), on the in-repo BAM reader.
"""
from __future__ import annotations

import collections
import functools
import logging
from typing import Any, Dict, List, Optional, Union

import numpy as np

from deepconsensus_amd.dcio import bam
from deepconsensus_amd.preprocess.expand import expand_clip_indent
from deepconsensus_amd.preprocess.read import Read, space_out_subreads
from deepconsensus_amd.preprocess.windows import DcConfig, DcExample
from deepconsensus_amd.utils import constants

Issue = constants.Issue


class SubreadGrouper:
    """Yields all mapped subreads of one ZMW (zm-tag grouped, streaming)."""

    def __init__(self, subreads_to_ccs: str, reader_threads: int = 1):
        self.bam_reader = iter(
            bam.BamReader(subreads_to_ccs, reader_threads=reader_threads)
        )
        self.keep_iter = True
        self.subread_group: List[bam.BamRead] = []
        first_read = next(self.bam_reader)
        self.zmw = first_read.get_tag("zm")
        if not first_read.is_unmapped:
            self.subread_group.append(first_read)

    def __iter__(self):
        return self

    def __next__(self) -> List[bam.BamRead]:
        if not self.keep_iter:
            raise StopIteration
        while self.keep_iter:
            try:
                read = next(self.bam_reader)
                if read.is_unmapped:
                    continue
            except StopIteration:
                self.keep_iter = False
                break
            read_zmw = read.get_tag("zm")
            if read_zmw == self.zmw:
                self.subread_group.append(read)
            else:
                subreads_set = self.subread_group
                self.subread_group = [read]
                self.zmw = read_zmw
                if subreads_set:
                    return subreads_set
        if self.subread_group:
            out = self.subread_group
            self.subread_group = []
            return out
        raise StopIteration


def construct_ccs_read(ccs_bam_read: bam.BamRead) -> Read:
    """CCS read with quality scores + aux tags (pre_lib.py:965-998)."""
    ccs_seq = np.array(list(ccs_bam_read.seq), dtype="<U1")

    def get_tag(read, tag_name):
        try:
            return read.get_tag(tag_name)
        except KeyError:
            return None

    return Read(
        name=ccs_bam_read.qname,
        bases=ccs_seq,
        cigar=np.repeat(np.uint8(constants.CMATCH), len(ccs_seq)),
        pw=np.repeat(np.uint8(0), len(ccs_seq)),
        ip=np.repeat(np.uint8(0), len(ccs_seq)),
        sn=np.repeat(0, 4),
        ec=get_tag(ccs_bam_read, "ec"),
        np_num_passes=get_tag(ccs_bam_read, "np"),
        rq=get_tag(ccs_bam_read, "rq"),
        rg=get_tag(ccs_bam_read, "RG"),
        strand=constants.Strand.UNKNOWN,
        base_quality_scores=np.array(ccs_bam_read.query_qualities),
        ccs_idx=np.arange(len(ccs_seq)),
    )


def fetch_label_alignment(
    ccs_seqname: str,
    truth_index: Dict[str, List[bam.BamRead]],
    truth_range: Dict[str, Any],
) -> Union[constants.Issue, Read]:
    """Fetches the label aligned to a ccs sequence (pre_lib.py:1001-1014)."""
    alns = truth_index.get(ccs_seqname)
    if not alns:
        return Issue.TRUTH_ALIGNMENT_NOT_FOUND
    truth_alignment = alns[0]
    if truth_alignment.is_supplementary:
        return Issue.SUPP_TRUTH_ALIGNMENT
    return expand_clip_indent(truth_alignment, truth_range)


def read_truth_bedfile(truth_bed: str) -> Dict[str, Dict[str, Any]]:
    bed_coords = {}
    with open(truth_bed) as bedfile:
        for line in bedfile:
            contig, begin, end, ccs_seqname = line.strip().split("\t")[:4]
            bed_coords[ccs_seqname] = {
                "contig": contig,
                "begin": int(begin),
                "end": int(end),
            }
    return bed_coords


def read_truth_split(split_fname: str) -> Dict[str, str]:
    contig_split = {}
    split_regions = {}
    lower = split_fname.lower()
    if any(x in lower for x in ["chm13", "hg00", "human"]):
        genome = "HUMAN"
    elif "maize" in lower:
        genome = "MAIZE"
    else:
        raise ValueError(
            f"{split_fname} does not correspond to any genome specified in "
            "constants.py. Please either change the file name or add new "
            "train/eval/test regions."
        )
    for i in constants.TRAIN_REGIONS[genome]:
        split_regions[i] = "train"
    for i in constants.EVAL_REGIONS[genome]:
        split_regions[i] = "eval"
    for i in constants.TEST_REGIONS[genome]:
        split_regions[i] = "test"
    with open(split_fname) as f:
        for line in f:
            contig, chrom = line.split()
            if chrom in split_regions:
                contig_split[contig] = split_regions[chrom]
    return contig_split


class ZmwJob:
    """Deferred per-ZMW expansion: carries the raw BAM records so the
    CPU-heavy expand_clip_indent runs in the worker pool instead of the
    serial feeder thread (inference mode only — training filters on the
    expanded label feeder-side)."""

    __slots__ = ("read_set", "ccs_bam_read", "ins_trim")

    def __init__(self, read_set, ccs_bam_read, ins_trim):
        self.read_set = read_set
        self.ccs_bam_read = ccs_bam_read
        self.ins_trim = ins_trim

    def __len__(self):
        return len(self.read_set) + 1

    def materialize(self, counter) -> List[Read]:
        expand = functools.partial(
            expand_clip_indent,
            truth_range=None,
            ins_trim=self.ins_trim,
            counter=counter,
        )
        subreads = list(map(expand, self.read_set))
        subreads.append(construct_ccs_read(self.ccs_bam_read))
        return subreads


class RawZmwJob(ZmwJob):
    """ZmwJob over UNDECODED record buffers: the worker decodes the BAM
    records too, leaving only a zm/flag/qname peek per record on the
    serial feeder thread. Raw bytes also pickle ~2x smaller than decoded
    BamRead objects (4-bit packed seq, packed tags)."""

    __slots__ = ("header",)

    def __init__(self, read_set, ccs_bam_read, ins_trim, header):
        super().__init__(read_set, ccs_bam_read, ins_trim)
        self.header = header

    def materialize(self, counter) -> List[Read]:
        records = [bam.decode_record(b, self.header) for b in self.read_set]
        ccs_rec = bam.decode_record(self.ccs_bam_read, self.header)
        return ZmwJob(records, ccs_rec, self.ins_trim).materialize(counter)


class RawSubreadGrouper:
    """SubreadGrouper over raw record buffers (zm-tag peek grouping).

    ``start`` is a (compressed_offset, within_block_offset) record
    boundary from bam.build_zmw_index; ``max_groups`` stops after that
    many ZMW groups — together they let shard i/N stream only its own
    contiguous byte range.
    """

    def __init__(self, subreads_to_ccs: str, decompress_threads: int = 4,
                 start=None, max_groups: Optional[int] = None):
        self.reader = bam.RawBamReader(
            subreads_to_ccs, decompress_threads=decompress_threads,
            start=start,
        )
        self.header = self.reader.header
        self._iter = iter(self.reader)
        self.keep_iter = True
        self.max_groups = max_groups
        self.groups_out = 0
        self.subread_group: List[bytes] = []
        if max_groups is not None and max_groups <= 0:
            self.keep_iter = False
            self.zmw = None
            return
        first = next(self._iter)
        self.zmw = bam.raw_tag(first, "zm")
        if not bam.raw_flag(first) & bam.FUNMAP:
            self.subread_group.append(first)

    def __iter__(self):
        return self

    def _count_run(self) -> None:
        """Counts one consumed zm-run against max_groups.

        Counts EVERY run (emitted or all-unmapped/empty) so the
        accounting matches bam.build_zmw_index, which records every
        zm-run boundary: a shard limited to max_groups=k must stop at
        the k-th index entry even when some runs emit nothing,
        otherwise it reads past its byte range and duplicates the next
        shard's first ZMWs."""
        self.groups_out += 1
        if self.max_groups is not None and self.groups_out >= self.max_groups:
            self.keep_iter = False

    def __next__(self) -> List[bytes]:
        if not self.keep_iter:
            raise StopIteration
        exhausted = False
        while self.keep_iter:
            try:
                buf = next(self._iter)
            except StopIteration:
                self.keep_iter = False
                exhausted = True
                break
            read_zmw = bam.raw_tag(buf, "zm")
            if read_zmw == self.zmw:
                if not bam.raw_flag(buf) & bam.FUNMAP:
                    self.subread_group.append(buf)
                continue
            # zm-run boundary: count the finished run, stage the new one.
            subreads_set = self.subread_group
            self.subread_group = (
                [] if bam.raw_flag(buf) & bam.FUNMAP else [buf]
            )
            self.zmw = read_zmw
            self._count_run()
            if subreads_set:
                return subreads_set
            # All-unmapped run: counted but nothing to emit; keep
            # scanning (the while condition honors a max_groups stop).
        if exhausted and self.subread_group:
            out = self.subread_group
            self.subread_group = []
            self._count_run()
            return out
        raise StopIteration


def create_proc_feeder(
    subreads_to_ccs: str,
    ccs_bam: str,
    dc_config: DcConfig,
    ins_trim: int = 0,
    use_ccs_smart_windows: bool = False,
    truth_bed: Optional[str] = None,
    truth_to_ccs: Optional[str] = None,
    truth_split: Optional[str] = None,
    limit: int = 0,
    bam_reader_threads: int = 1,
    defer_expansion: bool = False,
    raw_records: bool = True,
    shard_index: int = 0,
    shard_count: int = 1,
):
    """Generator feeding per-ZMW jobs (pre_lib.py:1279-1367).

    With defer_expansion and raw_records (the inference default), the
    serial thread never decodes a BAM record: BGZF blocks decompress on
    a thread pool, subreads group by a zm-tag peek on raw buffers, and
    full record decode + expansion happen in the worker pool.

    With shard_count > 1, this feeder yields only shard_index's ZMWs.
    If ZMW index sidecars exist for both BAMs (bam.build_zmw_index /
    `deepconsensus index`), the shard is a contiguous byte range and the
    feeder decompresses only ~1/N of each stream — otherwise it streams
    everything and keeps every Nth group (each rank then pays the full
    decompress, like the reference's `ccs --chunk` sharding).
    """
    main_counter = collections.Counter()
    is_training = truth_bed and truth_to_ccs and truth_split
    assert not (defer_expansion and is_training), (
        "defer_expansion is an inference-mode optimization"
    )
    raw = defer_expansion and raw_records
    modulo_shard = shard_count > 1
    if raw:
        sub_start = None
        sub_max_groups = None
        ccs_start = None
        if shard_count > 1:
            sub_idx = bam.load_zmw_index(subreads_to_ccs)
            ccs_idx = bam.load_zmw_index(ccs_bam)
            if (
                sub_idx is not None and ccs_idx is not None
                and sub_idx["sorted_flag"][0] and ccs_idx["sorted_flag"][0]
            ):
                n_groups = len(sub_idx["zmw"])
                lo = n_groups * shard_index // shard_count
                hi = n_groups * (shard_index + 1) // shard_count
                sub_max_groups = hi - lo
                if lo < n_groups:
                    sub_start = (int(sub_idx["coffset"][lo]),
                                 int(sub_idx["uoffset"][lo]))
                    j = int(np.searchsorted(ccs_idx["zmw"],
                                            sub_idx["zmw"][lo]))
                    j = min(j, max(len(ccs_idx["zmw"]) - 1, 0))
                    ccs_start = (int(ccs_idx["coffset"][j]),
                                 int(ccs_idx["uoffset"][j]))
                modulo_shard = False
                logging.info(
                    "byte-range shard %d/%d: ZMW groups [%d, %d) of %d",
                    shard_index, shard_count, lo, hi, n_groups,
                )
        subread_grouper = RawSubreadGrouper(
            subreads_to_ccs, start=sub_start, max_groups=sub_max_groups
        )
        ccs_bam_h = iter(bam.RawBamReader(ccs_bam, start=ccs_start))
        header = subread_grouper.header
    else:
        subread_grouper = SubreadGrouper(subreads_to_ccs, bam_reader_threads)
        ccs_bam_h = iter(bam.BamReader(ccs_bam))

    if is_training:
        truth_index = bam.fetch_index(truth_to_ccs)
        truth_ref_coords = read_truth_bedfile(truth_bed)
        truth_split_dict = read_truth_split(truth_split)

    def proc_feeder():
        group_idx = -1
        for read_set in subread_grouper:
            main_counter["n_zmw_processed"] += 1
            group_idx += 1
            if modulo_shard and group_idx % shard_count != shard_index:
                continue
            if raw:
                ccs_seqname = header.references[
                    bam.raw_ref_id(read_set[0])
                ][0]
                while True:
                    try:
                        ccs_bam_read = next(ccs_bam_h)
                    except StopIteration:
                        # PEP 479 would surface this as an opaque
                        # RuntimeError; name the missing ZMW instead
                        # (ccs filtering or a stale/mismatched index).
                        raise ValueError(
                            f"ccs bam does not contain {ccs_seqname} "
                            "(exhausted while matching qnames — ccs "
                            "filtering or stale ZMW index?)"
                        ) from None
                    if bam.raw_qname(ccs_bam_read) == ccs_seqname:
                        break
                window_widths = None
                if use_ccs_smart_windows:
                    window_widths = np.array(
                        bam.raw_tag(ccs_bam_read, "wl")
                    )
                main_counter["n_zmw_inference"] += 1
                main_counter["n_zmw_pass"] += 1
                yield (RawZmwJob(read_set, ccs_bam_read, ins_trim, header),
                       ccs_seqname, dc_config, "inference", window_widths)
                if limit and main_counter["n_zmw_pass"] >= limit:
                    break
                continue
            ccs_seqname = read_set[0].reference_name
            while True:
                try:
                    ccs_bam_read = next(ccs_bam_h)
                except StopIteration:
                    raise ValueError(
                        f"ccs bam does not contain {ccs_seqname} "
                        "(exhausted while matching qnames)"
                    ) from None
                if ccs_bam_read.qname == ccs_seqname:
                    break
            window_widths = None
            if use_ccs_smart_windows:
                window_widths = np.array(ccs_bam_read.get_tag("wl"))

            if defer_expansion:
                main_counter["n_zmw_inference"] += 1
                main_counter["n_zmw_pass"] += 1
                yield (ZmwJob(read_set, ccs_bam_read, ins_trim),
                       ccs_seqname, dc_config, "inference", window_widths)
                if limit and main_counter["n_zmw_pass"] >= limit:
                    break
                continue

            expand = functools.partial(
                expand_clip_indent,
                truth_range=None,
                ins_trim=ins_trim,
                counter=main_counter,
            )
            subreads = list(map(expand, read_set))
            ccs_read = construct_ccs_read(ccs_bam_read)
            subreads.append(ccs_read)

            if is_training:
                truth_range = truth_ref_coords.get(ccs_seqname, None)
                if not truth_range:
                    logging.info(
                        "No truth_range defined for %s.", ccs_seqname
                    )
                    main_counter["n_zmw_missing_truth_range"] += 1
                    continue
                label = fetch_label_alignment(
                    ccs_seqname, truth_index, truth_range
                )
                if label == Issue.TRUTH_ALIGNMENT_NOT_FOUND:
                    logging.info(
                        "Unable to fetch label alignment for %s.",
                        ccs_seqname,
                    )
                    main_counter["n_zmw_no_label_alignment"] += 1
                    continue
                elif label == Issue.SUPP_TRUTH_ALIGNMENT:
                    main_counter["n_zmw_truth_label_supp_alignment"] += 1
                    continue
                subreads.append(label)
                split = truth_split_dict.get(truth_range["contig"], None)
                if not split:
                    logging.info("No split defined for %s.", ccs_seqname)
                    main_counter["n_zmw_missing_contig_split"] += 1
                    continue
            else:
                split = "inference"
            main_counter[f"n_zmw_{split}"] += 1
            main_counter["n_zmw_pass"] += 1
            yield (subreads, ccs_seqname, dc_config, split, window_widths)
            if limit and main_counter["n_zmw_pass"] >= limit:
                break

    return proc_feeder, main_counter


def subreads_to_dc_example(
    subreads: List[Read],
    ccs_seqname: str,
    dc_config: DcConfig,
    window_widths: Optional[np.ndarray] = None,
) -> DcExample:
    """Spaces reads and wraps them in a DcExample (pre_lib.py:1370-1384)."""
    aln_reads = space_out_subreads(subreads)
    return DcExample(
        name=ccs_seqname,
        reads=aln_reads,
        config=dc_config,
        window_widths=window_widths,
    )
