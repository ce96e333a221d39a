"""E2E training tests (CPU), mirroring reference
model_train_custom_loop_test.py: full train() on tiny synthetic data,
asserting checkpoints, metrics TSV, params.json, best_checkpoint.txt; plus
distillation and eval binary smokes and a world_size=2 gloo DP test."""
import glob
import json
import os

import pytest
import torch

from deepconsensus_amd.dcio import bam as bam_lib
from deepconsensus_amd.models import config as cfg

from test_io_and_pipeline import make_test_bams


def make_training_data(tmp_path, n_zmws=4, length=220):
    """Synthetic training TFRecords via the real preprocess pipeline."""
    sub, ccs = make_test_bams(tmp_path, n_zmws=n_zmws, length=length)
    # Truth: perfect alignments of the ccs sequence itself.
    ccs_reads = list(bam_lib.BamReader(ccs))
    refs = [(r.qname, len(r.seq)) for r in ccs_reads]
    header = bam_lib.BamHeader(text="@HD\tVN:1.6", references=refs)
    truth_path = str(tmp_path / "truth_to_ccs.bam")
    with bam_lib.BamWriter(truth_path, header) as w:
        for rid, r in enumerate(ccs_reads):
            w.write(
                bam_lib.BamRead(
                    qname=f"truth_{rid}", flag=0, ref_id=rid, pos=0,
                    mapq=60, cigartuples=[(0, len(r.seq))], seq=r.seq,
                    query_qualities=[40] * len(r.seq), tags={},
                )
            )
    bed_path = str(tmp_path / "truth.bed")
    with open(bed_path, "w") as f:
        for rid, r in enumerate(ccs_reads):
            f.write(f"chr1\t0\t{len(r.seq)}\t{r.qname}\n")
    split_path = str(tmp_path / "human_split.txt")
    with open(split_path, "w") as f:
        f.write("chr1 chr1\n")

    from deepconsensus_amd.preprocess import preprocess_cli

    out = str(tmp_path / "tfex" / "ex-@split.tfrecord.gz")
    preprocess_cli.main([
        "--subreads_to_ccs", sub, "--ccs_bam", ccs, "--output", out,
        "--truth_to_ccs", truth_path, "--truth_bed", bed_path,
        "--truth_split", split_path, "--cpus", "0",
    ])
    train_file = str(tmp_path / "tfex" / "ex-train.tfrecord.gz")
    assert os.path.exists(train_file)
    summary = json.load(open(str(tmp_path / "tfex" / "ex-summary.training.json")))
    assert summary["n_zmw_train"] == n_zmws
    return train_file, summary


@pytest.fixture(scope="module")
def train_data(tmp_path_factory):
    tmp = tmp_path_factory.mktemp("traindata")
    return make_training_data(tmp)


def _tiny_params(train_file, model="transformer_learn_values"):
    params = cfg.get_config(f"{model}+test")
    params.train_path = [train_file]
    params.eval_path = [train_file]
    params.batch_size = 4
    params.num_epochs = 1
    params.buffer_size = 8
    params.n_examples_train = 12
    params.n_examples_eval = 12
    params.warmup_steps = 2
    cfg.modify_params(params)
    return params


def test_train_e2e_transformer(tmp_path, train_data):
    from deepconsensus_amd.models import train as train_lib

    train_file, _ = train_data
    params = _tiny_params(train_file)
    out_dir = str(tmp_path / "model")
    summary = train_lib.train_model(
        out_dir, params, device="cpu", eval_every=2, limit_steps=2,
    )
    assert "eval/per_example_accuracy" in summary
    assert os.path.exists(os.path.join(out_dir, "params.json"))
    assert glob.glob(os.path.join(out_dir, "checkpoint-*.pt"))
    assert os.path.exists(os.path.join(out_dir, "checkpoint_metrics.tsv"))
    assert os.path.exists(os.path.join(out_dir, "best_checkpoint.txt"))
    assert os.path.exists(os.path.join(out_dir, "training_summary.json"))

    # Checkpoint loads back into a fresh model.
    from deepconsensus_amd.models import checkpoint as ckpt_lib
    from deepconsensus_amd.models.model import get_model

    p2 = ckpt_lib.load_params(out_dir)
    cfg.modify_params(p2, is_training=False)
    m2 = get_model(p2)
    ckpt_lib.load_checkpoint(out_dir, m2)

    # Resume: second call restores epoch/step from sidecars.
    path, epoch, step = ckpt_lib.get_checkpoint_and_initial_epoch(out_dir)
    assert path is not None and step >= 2


def test_train_e2e_fc(tmp_path, train_data):
    from deepconsensus_amd.models import train as train_lib

    train_file, _ = train_data
    params = _tiny_params(train_file, model="fc")
    out_dir = str(tmp_path / "fc_model")
    summary = train_lib.train_model(
        out_dir, params, device="cpu", eval_every=100, limit_steps=2,
    )
    assert summary["steps"] >= 2


def test_distill_e2e(tmp_path, train_data):
    from deepconsensus_amd.models import distill as distill_lib
    from deepconsensus_amd.models import train as train_lib

    train_file, _ = train_data
    teacher_params = _tiny_params(train_file)
    teacher_dir = str(tmp_path / "teacher")
    train_lib.train_model(teacher_dir, teacher_params, device="cpu",
                          eval_every=100, limit_steps=1)

    params = cfg.get_config("transformer_learn_values_distill+test")
    params.train_path = [train_file]
    params.eval_path = [train_file]
    params.batch_size = 4
    params.num_epochs = 1
    params.n_examples_train = 12
    cfg.modify_params(params)
    out_dir = str(tmp_path / "student")
    summary = distill_lib.train_model(
        out_dir, teacher_dir, params, device="cpu", limit_steps=1,
    )
    assert summary["steps"] == 1
    assert glob.glob(os.path.join(out_dir, "checkpoint-*.pt"))


def test_eval_binary(tmp_path, train_data):
    from deepconsensus_amd.models import infer_eval

    train_file, _ = train_data
    params = _tiny_params(train_file)
    out_dir = str(tmp_path / "eval_out")
    metrics = infer_eval.run_inference(
        out_dir, "random", [train_file], params=params, device="cpu",
    )
    assert os.path.exists(os.path.join(out_dir, "inference.csv"))
    assert 0 <= metrics["per_example_accuracy"] <= 1


def _dp_worker(rank, world, train_file, out_dir, port):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    from deepconsensus_amd.models import train as train_lib

    params = _tiny_params(train_file, model="fc")
    train_lib.train_model(out_dir, params, device="cpu", eval_every=100,
                          limit_steps=2)
    import torch.distributed as dist

    if dist.is_initialized():
        dist.destroy_process_group()


def test_train_dp2_gloo(tmp_path, train_data):
    """DP=2 on gloo: fused-bucket all-reduce path, world_size 2."""
    train_file, _ = train_data
    out_dir = str(tmp_path / "dp2")
    ctx = torch.multiprocessing.spawn(
        _dp_worker, args=(2, train_file, out_dir, 29876),
        nprocs=2, join=True,
    )
    assert glob.glob(os.path.join(out_dir, "checkpoint-*.pt"))


def test_lamb_capturable_matches_standard():
    """capturable=True (device-tensor lr/step) matches the standard step."""
    import torch

    from deepconsensus_amd.models import lamb as lamb_lib

    torch.manual_seed(0)
    net_a = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.Linear(16, 4))
    net_b = __import__("copy").deepcopy(net_a)
    opt_a = lamb_lib.LAMB(net_a.parameters(), lr=0.01, weight_decay=0.02)
    opt_b = lamb_lib.LAMB(net_b.parameters(), lr=0.01, weight_decay=0.02,
                          capturable=True)
    sched_a = lamb_lib.PolynomialWarmupSchedule(0.01, 0.001, 20, 5)
    sched_b = lamb_lib.PolynomialWarmupSchedule(0.01, 0.001, 20, 5)
    x = torch.randn(16, 8)
    y = torch.randn(16, 4)
    for step in range(6):
        for net, opt, sched in ((net_a, opt_a, sched_a),
                                (net_b, opt_b, sched_b)):
            sched.apply(opt, step)
            opt.zero_grad()
            ((net(x) - y) ** 2).mean().backward()
            opt.step()
    for pa, pb in zip(net_a.parameters(), net_b.parameters()):
        assert torch.allclose(pa, pb, atol=1e-6), (pa - pb).abs().max()


def test_train_resume_from_checkpoint(tmp_path, train_data):
    """A second train_model in the same out_dir resumes from the latest
    checkpoint and continues the step counter."""
    from deepconsensus_amd.models import checkpoint as ckpt_lib
    from deepconsensus_amd.models import train as train_lib

    train_file, _ = train_data
    params = _tiny_params(train_file)
    out_dir = str(tmp_path / "model")
    s1 = train_lib.train_model(out_dir, params, device="cpu",
                               eval_every=2, limit_steps=2)
    path1, epoch1, step1 = ckpt_lib.get_checkpoint_and_initial_epoch(out_dir)
    assert path1 and step1 == s1["steps"] >= 2
    s2 = train_lib.train_model(out_dir, params, device="cpu",
                               eval_every=2, limit_steps=2)
    assert s2["steps"] >= s1["steps"] + 2  # continued, not restarted
    _, _, step2 = ckpt_lib.get_checkpoint_and_initial_epoch(out_dir)
    assert step2 == s2["steps"]


def test_training_converges_with_fused_embed_backward(tmp_path, train_data):
    """Loss drops substantially over a few dozen steps — a guard on the
    fused embedding-backward gradients actually pointing downhill."""
    from deepconsensus_amd.models import config as cfg
    from deepconsensus_amd.models import lamb as lamb_lib
    from deepconsensus_amd.models import losses as losses_lib
    from deepconsensus_amd.models.model import get_model
    from deepconsensus_amd.models import data as data_lib
    from deepconsensus_amd.models.train import _prepare_batch

    train_file, _ = train_data
    params = _tiny_params(train_file)
    torch.manual_seed(1)
    model = get_model(params)
    model.train()
    opt = lamb_lib.LAMB(model.parameters(), lr=3e-3)
    loss_fn = losses_lib.AlignmentLoss(del_cost=params.del_cost,
                                       loss_reg=params.loss_reg,
                                       reduction="mean")
    ds = data_lib.DatasetIterator([train_file], params, 4, seed=3)
    losses = []
    for epoch in range(40):
        for batch in ds.iterate(epoch):
            rows, label = _prepare_batch(batch, "cpu")
            opt.zero_grad()
            probs = model(rows, training=True)
            loss = loss_fn(label, probs.float())
            loss.backward()
            # The fused path must be ACTIVE (gate: training + grad).
            assert model.bases_embedding.table.grad is not None
            opt.step()
            losses.append(float(loss))
            break  # one batch per epoch
        if len(losses) >= 40:
            break
    first = sum(losses[:5]) / 5
    last = sum(losses[-5:]) / 5
    assert last < first * 0.8, (first, last)
    # Embedding tables actually moved.
    assert float(model.pw_embedding.table.grad.abs().sum()) > 0
