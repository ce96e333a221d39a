"""Alignment expansion: BAM record -> gap-expanded Read.

Behavioral parity with reference pre_lib.py:1061-1239:
* trim_insertions: removes insertions longer than ins_trim from sequence,
  cigar and pw/ip tags (tag masks applied in original-strand orientation);
* expand_clip_indent: gap-expands deletions, strips soft/hard clips, indents
  by reference start, reverses pw/ip for reverse-strand reads, and adjusts
  truth ranges for soft-clipped label alignments.
"""
from __future__ import annotations

import collections
from typing import Any, Dict, Optional, Union

import numpy as np

from deepconsensus_amd.dcio.bam import BamRead
from deepconsensus_amd.preprocess.read import Read
from deepconsensus_amd.utils import constants


def trim_insertions(
    read: BamRead,
    ins_trim: int,
    counter: Optional[collections.Counter] = None,
) -> BamRead:
    """Removes insertions > ins_trim bp (pre_lib.py:1061-1125)."""
    if ins_trim <= 0:
        return read
    pw_vals = read.get_tag("pw") if read.has_tag("pw") else []
    ip_vals = read.get_tag("ip") if read.has_tag("ip") else []

    trimmed_cigar = []
    trimmed_seq = ""
    seq_pos = 0
    mask = [True] * len(read.seq)
    for cigar_op, op_len in read.cigartuples:
        if cigar_op == constants.CINS and op_len > ins_trim:
            mask[seq_pos : seq_pos + op_len] = [False] * op_len
            seq_pos += op_len
            if counter is not None:
                counter["zmw_trimmed_insertions"] += 1
                counter["zmw_trimmed_insertions_bp"] += op_len
        else:
            trimmed_cigar.append((cigar_op, op_len))
            if cigar_op != constants.CDEL:
                trimmed_seq += read.query_sequence[
                    seq_pos : seq_pos + op_len
                ]
                seq_pos += op_len
        if counter is not None:
            counter["zmw_total_bp"] += op_len

    if len(pw_vals):
        m = np.array(mask[::-1] if read.is_reverse else mask)
        read.set_tag("pw", np.array(pw_vals)[m])
    if len(ip_vals):
        m = np.array(mask[::-1] if read.is_reverse else mask)
        read.set_tag("ip", np.array(ip_vals)[m])

    read.seq = trimmed_seq
    read.cigartuples = trimmed_cigar
    return read


def _spacing_ext():
    global _SPACING
    if _SPACING is None:
        try:
            from deepconsensus_amd.ops.build import build_spacing

            _SPACING = build_spacing()
        except Exception:  # pragma: no cover - no compiler at runtime
            _SPACING = False
    return _SPACING


_SPACING = None


def _expand_native(read: BamRead, ext) -> Read:
    """C++ single-pass expand (inference path: no truth_range)."""
    cig = np.asarray(read.cigartuples, dtype=np.int32).reshape(-1, 2)
    pw_vals = np.minimum(np.asarray(read.get_tag("pw")), 255).astype(
        np.uint8
    )
    ip_vals = np.minimum(np.asarray(read.get_tag("ip")), 255).astype(
        np.uint8
    )
    bases_u32, cigar_u8, pw, ip, ccs_idx = ext.expand_read(
        read.seq.encode("ascii"), cig, pw_vals, ip_vals,
        int(read.pos) if read.pos and read.pos > 0 else 0,
        bool(read.is_reverse), True,
    )
    strand = (
        constants.Strand.REVERSE
        if read.is_reverse
        else constants.Strand.FORWARD
    )
    return Read(
        name=read.qname,
        bases=bases_u32.view("<U1"),
        cigar=cigar_u8,
        pw=pw,
        ip=ip,
        sn=np.array(read.get_tag("sn")),
        strand=strand,
        ccs_idx=ccs_idx,
        truth_range=None,
    )


def expand_clip_indent(
    read: BamRead,
    truth_range: Union[Dict[str, Any], None] = None,
    ins_trim: int = 0,
    counter: Optional[collections.Counter] = None,
) -> Read:
    """Expands an alignment into CCS space (pre_lib.py:1128-1239)."""
    if ins_trim > 0:
        read = trim_insertions(read, ins_trim, counter)
    if truth_range is None:
        ext = _spacing_ext()
        if ext:
            return _expand_native(read, ext)

    read_idx, ccs_idx = read.aligned_index_arrays()
    aln_len = len(read_idx)

    new_seq = np.full(aln_len, constants.GAP, dtype="<U1")
    new_pw = np.zeros(aln_len, dtype=np.uint8)
    new_ip = np.zeros(aln_len, dtype=np.uint8)

    new_seq[read_idx >= 0] = np.frombuffer(
        read.seq.encode("ascii"), dtype="S1"
    ).astype("<U1")

    strand = (
        constants.Strand.REVERSE
        if read.is_reverse
        else constants.Strand.FORWARD
    )

    if not truth_range:
        pw_vals = np.asarray(read.get_tag("pw"))
        ip_vals = np.asarray(read.get_tag("ip"))
        if strand == constants.Strand.REVERSE:
            pw_vals = pw_vals[::-1]
            ip_vals = ip_vals[::-1]
        # Values are clipped into uint8 range downstream (format_rows clips
        # to PW_MAX/IP_MAX); store clamped here like pysam's uint8 tag array.
        new_pw[read_idx >= 0] = np.minimum(pw_vals, 255).astype(np.uint8)
        new_ip[read_idx >= 0] = np.minimum(ip_vals, 255).astype(np.uint8)
        sn = np.array(read.get_tag("sn"))
    else:
        sn = np.empty(0, dtype=np.uint8)

    cigar_ops = []
    for op, n in read.cigartuples:
        cigar_ops.extend([op] * n)
    new_cigar = np.array(cigar_ops, dtype=np.uint8)
    new_cigar = new_cigar[new_cigar != constants.CHARD_CLIP]

    # Trim soft-clipped segments.
    if np.sum(new_cigar == constants.CSOFT_CLIP) > 0:
        new_seq[new_cigar == constants.CSOFT_CLIP] = constants.GAP
        qstart = np.where(read_idx == read.query_alignment_start)[0][0]
        qend = (
            np.where(read_idx == read.query_alignment_end - 1)[0][0] + 1
        )
        if truth_range:
            op, op_len = read.cigartuples[0]
            if op == constants.CSOFT_CLIP:
                truth_range["begin"] = truth_range["begin"] + op_len
            op, op_len = read.cigartuples[-1]
            if op == constants.CSOFT_CLIP:
                truth_range["end"] = truth_range["end"] - op_len
        new_seq = new_seq[qstart:qend]
        new_pw = new_pw[qstart:qend]
        new_ip = new_ip[qstart:qend]
        new_cigar = new_cigar[qstart:qend]
        ccs_idx = ccs_idx[qstart:qend]

    # Indent by reference start.
    if read.pos:
        new_seq = np.insert(new_seq, 0, [constants.GAP] * read.pos)
        new_cigar = np.insert(
            new_cigar, 0, np.repeat(int(constants.CREF_SKIP), read.pos)
        )
        new_pw = np.insert(new_pw, 0, np.repeat(0, read.pos))
        new_ip = np.insert(new_ip, 0, np.repeat(0, read.pos))
        ccs_idx = np.insert(ccs_idx, 0, np.repeat(-1, read.pos))

    return Read(
        name=read.qname,
        bases=new_seq,
        cigar=new_cigar,
        pw=new_pw,
        ip=new_ip,
        sn=sn,
        strand=strand,
        ccs_idx=ccs_idx,
        truth_range=truth_range,
    )
