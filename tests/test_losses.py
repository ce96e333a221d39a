"""Loss/metric tests porting the reference losses_and_metrics_test.py
hand-computed expectations (the correctness oracle for the wavefront
kernels)."""
import numpy as np
import pytest
import torch

from deepconsensus_amd.models import losses as L
from deepconsensus_amd.utils import phred


def seq_to_array(s):
    return phred.string_to_encoded_sequence(s.replace(" ", " ")).astype(
        np.float32
    )


def multiseq_to_array(seqs):
    return np.stack([seq_to_array(s) for s in seqs])


def seq_to_one_hot(seqs):
    arr = multiseq_to_array(seqs).astype(np.int64)
    return np.eye(5, dtype=np.float32)[arr]


def convert_seqs(sequences):
    y_true_str, y_pred_str = sequences
    y_true = torch.from_numpy(multiseq_to_array(y_true_str))
    y_pred = torch.from_numpy(seq_to_one_hot(y_pred_str))
    return y_true, y_pred


ALIGNMENT_LOSS_CASES = [
    # (name, (true, pred), del_cost, loss_reg, width, expected)
    ("identical", (["TTAGGC", "AGCTGG"], ["TTAGGC", "AGCTGG"]),
     1.0, None, None, 0.0),
    ("identical_same_pad",
     (["TTAGGC    ", "AGCTGG    "], ["TTAGGC    ", "AGCTGG    "]),
     1.0, None, None, 0.0),
    ("identical_diff_pad",
     (["TTAGGCAT", "AGCTGG  "], ["TTAGGCAT  ", "AGCTGG    "]),
     1.0, None, None, 0.0),
    ("correct_ins_no_pad",
     (["TTAGGC", "AGCTGG"], ["T TA G G C", "AGC    TGG"]),
     1.0, None, None, 0.0),
    ("correct_ins_pad",
     (["TTAGGC    ", "AGCTGG    "], ["TTA G GC  ", "AGC    TGG"]),
     1.0, None, None, 0.0),
    ("one_del_cost1", (["TTAGGC", "AGCTGG"], ["TTAGG ", "GCTGG "]),
     1.0, None, None, 1.0),
    ("one_del_cost2", (["TTAGGC", "AGCTGG"], ["TAGGC ", "AGCGG "]),
     2.0, None, None, 2.0),
    ("two_del_cost1", (["TTAGGC", "AGCTGG"], ["TTAG  ", "GCGG  "]),
     1.0, None, None, 2.0),
    ("one_error", (["TTAGGC", "AGCTGG"], ["ATAGGC", "TGCTGG"]),
     1.0, None, None, 16.118),
    ("two_errors", (["TTAGGC", "AGCTGG"], ["AAAGGC", "TGCTGC"]),
     1.0, None, None, 32.236),
    ("one_bad_ins",
     (["TTAGGC", "ATCGAC", "AGCTGG"],
      ["TTAGGCA", "ATCCGAC", "CAGCTGG"]),
     1.0, None, None, 16.118),
    ("del_small_cost", (["ATCG ", "ATCG "], ["TCG  ", "TCG  "]),
     1.0, None, None, 1.0),
    ("del_large_cost", (["ATCG ", "ATCG "], ["TCG  ", "TCG  "]),
     1e9, None, None, 64.472),
    # banded
    ("band_identical", (["TTAGGC", "AGCTGG"], ["TTAGGC", "AGCTGG"]),
     1.0, None, 2, 0.0),
    ("band_one_del", (["TTAGGC", "AGCTGG"], ["TTAGG ", "GCTGG "]),
     1.0, None, 2, 1.0),
    ("band_identical_pad",
     (["TTAGGC    ", "AGCTGG    "], ["TTAGGC    ", "AGCTGG    "]),
     1.0, None, 1, 0.0),
    ("band_correct_ins",
     (["TTAGGC   ", "AGCTG   G"], ["T TAG G C", "AGC   TGG"]),
     1.0, None, 8, 0.0),
    ("band_correct_ins_pad",
     (["TTAGGC    ", "AGCTGG    "], ["TTA G GC  ", "AGC    TGG"]),
     1.0, None, 8, 0.0),
    ("band_two_errors", (["TTAGGC", "AGCTGG"], ["AAAGGC", "TGCTGC"]),
     1.0, None, 4, 32.236),
    ("band2_two_dels", (["TTA", "GGC"], ["A  ", "C  "]),
     1.0, None, 2, 2.0),
    ("band1_del_align", (["TTA", "GGC"], ["A  ", "C  "]),
     1.0, None, 1, 18.118),
]


@pytest.mark.parametrize(
    "name,sequences,del_cost,loss_reg,width,expected",
    ALIGNMENT_LOSS_CASES,
    ids=[c[0] for c in ALIGNMENT_LOSS_CASES],
)
def test_alignment_loss(name, sequences, del_cost, loss_reg, width,
                        expected):
    y_true, y_pred = convert_seqs(sequences)
    loss = L.AlignmentLoss(del_cost=del_cost, loss_reg=loss_reg,
                           width=width)(y_true, y_pred)
    assert abs(float(loss) - expected) < 0.01, float(loss)


def test_alignment_loss_soft_close_to_hard():
    """Small loss_reg approaches the hard min."""
    y_true, y_pred = convert_seqs(
        (["TTAGGC", "AGCTGG"], ["TTAGG ", "GCTGG "])
    )
    hard = L.AlignmentLoss(del_cost=1.0, loss_reg=None)(y_true, y_pred)
    soft = L.AlignmentLoss(del_cost=1.0, loss_reg=0.01)(y_true, y_pred)
    assert abs(float(hard) - float(soft)) < 0.05


def test_alignment_loss_differentiable():
    torch.manual_seed(0)
    y_true = torch.from_numpy(multiseq_to_array(["TTAGGC", "AGCTGG"]))
    logits = torch.randn(2, 8, 5, requires_grad=True)
    probs = torch.softmax(logits, -1)
    loss = L.AlignmentLoss(del_cost=10.0, loss_reg=0.1)(y_true, probs)
    loss.backward()
    assert logits.grad is not None
    assert torch.isfinite(logits.grad).all()
    assert logits.grad.abs().sum() > 0


def test_left_shift_sequence():
    y = torch.tensor([[0, 1, 0, 2, 3, 0], [4, 0, 0, 1, 0, 2]])
    out = L.left_shift_sequence(y)
    np.testing.assert_array_equal(
        out.numpy(), [[1, 2, 3, 0, 0, 0], [4, 1, 2, 0, 0, 0]]
    )


ALIGNMENT_METRIC_CASES = [
    ("identical", (["TTAGGC", "AGCTGG"], ["TTAGGC", "AGCTGG"]),
     (1.0, 1.0)),
    ("two_errors", (["TTAGGC", "AGCTGG"], ["AAAGGC", "TGCTGC"]),
     (0.667, 0.667)),
    ("correct_ins", (["TTAGGC", "AGCTGG"], ["T TA G G C", "AGC    TGG"]),
     (1.0, 1.0)),
    ("one_del", (["TTAGGC", "AGCTGG"], ["TTAGG ", "GCTGG "]),
     (0.833, 0.833)),
    ("bad_ins",
     (["TTAGGC", "ATCGAC", "AGCTGG"],
      ["TTAGGCA", "ATCCGAC", "CAGCTGG"]),
     (0.857, 0.857, 0.857)),
    ("one_del_shorter", (["ATCG ", "ATCG "], ["TCG  ", "TCG  "]),
     (0.75, 0.75)),
    ("empty_pred", (["ATCG ", "ATCG "], ["     ", "     "]),
     (0.0, 0.0)),
    ("empty_truth", (["     ", "     "], ["ATCG ", "ATCG "]),
     (0.0, 0.0)),
    ("empty_pred_len1_truth", (["A    ", "T    "], ["     ", "     "]),
     (0.0, 0.0)),
    ("empty_truth_len1_pred", (["     ", "     "], ["A    ", "T    "]),
     (0.0, 0.0)),
    ("both_empty", (["     ", "     "], ["     ", "     "]),
     (1.0, 1.0)),
]


@pytest.mark.parametrize(
    "name,sequences,expected_pid", ALIGNMENT_METRIC_CASES,
    ids=[c[0] for c in ALIGNMENT_METRIC_CASES],
)
def test_alignment_metric(name, sequences, expected_pid):
    y_true, y_pred = convert_seqs(sequences)
    metric = L.AlignmentMetric()
    _, _, mv = metric.alignment(y_true, y_pred)
    for i, exp in enumerate(expected_pid):
        assert abs(float(mv["pid"][i]) - exp) < 0.01, (
            i, float(mv["pid"][i]), exp
        )


def test_per_example_accuracy():
    acc = L.PerExampleAccuracy()
    y_true = multiseq_to_array(["A T C G", "T T T T", "A A A A"])
    y_pred = seq_to_one_hot(["   ATCG", "   GGGG", "   AAAA"])
    acc.update_state(y_true, y_pred)
    assert abs(acc.result() - 2 / 3) < 1e-6


def test_per_example_accuracy_multiple_updates():
    acc = L.PerExampleAccuracy()
    y_true = multiseq_to_array(["A T C G"] * 3)
    y_pred = seq_to_one_hot(["   ATCG", "ATCG   ", "  ATCG "])
    acc.update_state(y_true, y_pred)
    assert acc.result() == 1.0
    y_true = multiseq_to_array(["C C C C", "A T C G", "C C C C"])
    y_pred = seq_to_one_hot(["   ATCG", "ATCG   ", "  CCCC "])
    acc.update_state(y_true, y_pred)
    assert abs(acc.result() - 5 / 6) < 1e-6


def test_batch_identity_and_yield():
    labels = torch.from_numpy(multiseq_to_array(["TTAGGC", "AGCTGG"]))
    preds = torch.from_numpy(seq_to_one_hot(["CCCCCC", "TGCTGG"]))
    ccs = multiseq_to_array(["CCAGGC", "TGCTGG"])
    metric = L.AlignmentMetric()
    identity_ccs, identity_pred = L.get_batch_identity_ccs_pred(
        ccs, preds, labels, metric
    )
    assert abs(identity_pred - 0.5) < 0.01
    assert abs(identity_ccs - 0.75) < 0.01

    y = L.YieldOverCCSMetric(quality_threshold=0.7)
    y.update_state(identity_ccs, identity_pred)  # ccs yes, dc no
    y.update_state(1.0, 1.0)  # both
    assert y.result() == 0.5


def test_distillation_loss():
    torch.manual_seed(0)
    t = torch.randn(4, 10, 5)
    assert float(L.DistillationLoss()(t, t)) < 1e-9
    s = torch.randn(4, 10, 5)
    assert float(L.DistillationLoss()(t, s)) > 0
    assert float(L.DistillationLoss(logit_loss="mean_squared_error")(t, s)) > 0
