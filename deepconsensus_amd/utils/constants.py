"""Shared constants for DeepConsensus-AMD.

Behavioral parity with the reference's constants module
(deepconsensus/utils/dc_constants.py:38-130), re-declared without the pysam /
TF dependencies: cigar opcodes are the BAM-spec integer codes directly.
"""
from __future__ import annotations

import enum

import numpy as np

# Vocab. Gap sorts first so that id 0 is the gap/pad token.
GAP = " "
ALLOWED_BASES = "ATCG"
SEQ_VOCAB = GAP + ALLOWED_BASES
SEQ_VOCAB_SIZE = len(SEQ_VOCAB)
GAP_INT = SEQ_VOCAB.index(GAP)  # == 0

# BAM/SAM cigar opcodes (SAM spec section 4.2; same integer codes pysam uses).
CMATCH = 0  # M
CINS = 1  # I
CDEL = 2  # D
CREF_SKIP = 3  # N
CSOFT_CLIP = 4  # S
CHARD_CLIP = 5  # H
CPAD = 6  # P
CEQUAL = 7  # =
CDIFF = 8  # X
CBACK = 9  # B

CIGAR_OPS = {
    "M": CMATCH,
    "I": CINS,
    "D": CDEL,
    "N": CREF_SKIP,
    "S": CSOFT_CLIP,
    "H": CHARD_CLIP,
    "P": CPAD,
    "=": CEQUAL,
    "X": CDIFF,
    "B": CBACK,
}
CIGAR_CHARS = "MIDNSHP=XB"

# Ops that consume query (read) bases.
READ_ADVANCING_OPS = (CMATCH, CINS, CEQUAL, CDIFF)
# Ops that consume reference bases.
REF_ADVANCING_OPS = (CMATCH, CDEL, CREF_SKIP, CEQUAL, CDIFF)


class Issue(int, enum.Enum):
    TRUTH_ALIGNMENT_NOT_FOUND = 1
    SUPP_TRUTH_ALIGNMENT = 2


class Strand(int, enum.Enum):
    UNKNOWN = 0
    FORWARD = 1
    REVERSE = 2


NP_DATA_TYPE = np.float32

EMPTY_QUAL = 0

# Train/eval/test genomic splits (reference dc_constants.py:90-111).
ECOLI_REGIONS = {
    "TRAIN": (464253, 4178270),
    "EVAL": (0, 464252),
    "TEST": (4178271, 4642522),
}
TRAIN_REGIONS = {
    "HUMAN": (
        [str(i) for i in range(1, 19)]
        + ["chr%d" % i for i in range(1, 19)]
        + ["X", "Y", "chrX", "chrY"]
    ),
    "MAIZE": [str(i) for i in range(1, 9)] + ["chr%d" % i for i in range(1, 9)],
}
EVAL_REGIONS = {
    "HUMAN": ["21", "22", "chr21", "chr22"],
    "MAIZE": ["9", "chr9"],
}
TEST_REGIONS = {
    "HUMAN": ["19", "20", "chr19", "chr20"],
    "MAIZE": ["10", "chr10"],
}

# Features present in DC examples (reference dc_constants.py:114-125).
DC_FEATURES = [
    "rows",
    "label",
    "num_passes",
    "window_pos",
    "name",
    "ccs_base_quality_scores",
    "ec",
    "np_num_passes",
    "rq",
    "rg",
]

MAIN_EVAL_METRIC_NAME = "eval/per_example_accuracy"

# Maximum Phred quality emitted (quick_inference caps at 93; chr(93+33)='~').
MAX_QUAL = 93
