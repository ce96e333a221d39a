"""Window stitching: sorted per-window predictions -> full read FASTQ.

Behavioral parity with reference stitch_utils.py:39-189: missing windows
discard the read (or N-fill), gaps are removed with their quality scores,
average-Phred quality filter (rounded to 5 decimals to dodge float drift),
minimum length filter, and the OutcomeCounter outcome taxonomy.
"""
from __future__ import annotations

import dataclasses
import logging
from typing import Iterable, Optional, Tuple

import numpy as np

from deepconsensus_amd.utils import constants, phred


@dataclasses.dataclass
class DCModelOutput:
    molecule_name: str
    window_pos: int
    ec: Optional[float] = None
    np_num_passes: Optional[int] = None
    rq: Optional[float] = None
    rg: Optional[str] = None
    sequence: Optional[str] = None
    quality_string: Optional[str] = None


def get_full_sequence(
    deepconsensus_outputs: Iterable[DCModelOutput],
    max_length: int,
    fill_n: bool = False,
):
    """Concatenates sorted windows; None on missing window unless fill_n."""
    full_sequence_parts = []
    quality_string_parts = []
    start = 0
    for dc_output in deepconsensus_outputs:
        while dc_output.window_pos > start:
            if not fill_n:
                return None, ""
            full_sequence_parts.append("N" * max_length)
            empty = np.array([constants.EMPTY_QUAL] * max_length)
            quality_string_parts.append(phred.quality_scores_to_string(empty))
            start += max_length
        full_sequence_parts.append(dc_output.sequence)
        quality_string_parts.append(dc_output.quality_string)
        start += max_length
    return "".join(full_sequence_parts), "".join(quality_string_parts)


def remove_gaps(sequence: str, quality_string: str) -> Tuple[str, str]:
    """Drops gap positions and their quality characters."""
    final_sequence = []
    final_quality = []
    for base, quality in zip(sequence, quality_string):
        if base != constants.GAP:
            final_sequence.append(base)
            final_quality.append(quality)
    out_seq = "".join(final_sequence)
    out_qual = "".join(final_quality)
    assert len(out_seq) == len(out_qual)
    return out_seq, out_qual


def is_quality_above_threshold(quality_string: str, min_quality: int) -> bool:
    quality_score_array = phred.quality_string_to_array(quality_string)
    # Round to dodge float drift: all-Q10 reads average to 9.99999...
    rounded_avg_phred = round(phred.avg_phred(quality_score_array), 5)
    return rounded_avg_phred >= min_quality


def format_as_fastq(
    molecule_name: str, sequence: str, quality_string: str
) -> str:
    return f"@{molecule_name}\n{sequence}\n+\n{quality_string}\n"


@dataclasses.dataclass
class OutcomeCounter:
    empty_sequence: int = 0
    only_gaps: int = 0
    failed_quality_filter: int = 0
    failed_length_filter: int = 0
    success: int = 0

    @property
    def total(self) -> int:
        return (
            self.empty_sequence + self.only_gaps
            + self.failed_quality_filter + self.failed_length_filter
            + self.success
        )


def stitch_to_fastq(
    molecule_name: str,
    predictions: Iterable[DCModelOutput],
    max_length: int,
    min_quality: int,
    min_length: int,
    outcome_counter: OutcomeCounter,
) -> Optional[str]:
    """Stitches, filters, and formats one read (stitch_utils.py:131-189)."""
    full_sequence, full_quality_string = get_full_sequence(
        deepconsensus_outputs=predictions, max_length=max_length
    )
    if not full_sequence:
        outcome_counter.empty_sequence += 1
        logging.debug("empty after stitching: %s", molecule_name)
        return None

    final_sequence, final_quality_string = remove_gaps(
        full_sequence, full_quality_string
    )
    if not final_sequence:
        outcome_counter.only_gaps += 1
        return None
    if not is_quality_above_threshold(final_quality_string, min_quality):
        outcome_counter.failed_quality_filter += 1
        return None
    if len(final_sequence) < min_length:
        outcome_counter.failed_length_filter += 1
        return None
    outcome_counter.success += 1
    return format_as_fastq(
        molecule_name, final_sequence, final_quality_string
    )
