"""Phred/utils tests mirroring reference utils_test.py."""
import numpy as np

from deepconsensus_amd.utils import constants, phred


def test_vocab():
    assert constants.SEQ_VOCAB == " ATCG"
    assert constants.GAP_INT == 0
    assert constants.SEQ_VOCAB_SIZE == 5


def test_encoded_sequence_round_trip():
    s = "ATCG ATT"
    enc = phred.string_to_encoded_sequence(s)
    assert phred.encoded_sequence_to_string(enc) == s


def test_quality_conversions():
    assert phred.quality_score_to_string(0) == "!"
    assert phred.quality_score_to_string(93) == "~"
    assert phred.quality_scores_to_string(np.array([0, 40, 93])) == "!I~"
    assert phred.quality_string_to_array("!I~") == [0, 40, 93]


def test_avg_phred_prob_space():
    # avg of two equal quals is that qual.
    assert abs(phred.avg_phred([30, 30]) - 30.0) < 1e-6
    # prob-space averaging pulls toward the worse qual.
    v = phred.avg_phred([10, 50])
    assert 12 < v < 14
    # -1 spacer values ignored.
    assert abs(phred.avg_phred([30, -1, 30]) - 30.0) < 1e-6
    assert phred.avg_phred([0, 0]) == 0.0
    assert phred.avg_phred([-1, -1]) == 0.0


def test_left_shift():
    seq = np.array([0, 1, 0, 2, 3, 0])
    np.testing.assert_array_equal(
        phred.left_shift_seq(seq), [1, 2, 3, 0, 0, 0]
    )
    batch = np.array([[0, 1, 0, 2], [4, 0, 0, 1]])
    np.testing.assert_array_equal(
        phred.left_shift(batch), [[1, 2, 0, 0], [4, 1, 0, 0]]
    )
