"""GPU end-to-end pipeline tests: full train loop and full `run` inference
on the native MI355X path."""
import glob
import json
import os

import pytest
import torch

from deepconsensus_amd.models import config as cfg

from test_io_and_pipeline import make_test_bams
from test_train import make_training_data, _tiny_params

pytestmark = pytest.mark.gpu


def test_train_e2e_gpu(tmp_path):
    """Full training loop on cuda: HIP alignment loss + LAMB + checkpoints."""
    from deepconsensus_amd.models import train as train_lib

    train_file, _ = make_training_data(tmp_path)
    params = _tiny_params(train_file)  # batch 4 -> 3 steps per epoch
    out_dir = str(tmp_path / "model")
    summary = train_lib.train_model(
        out_dir, params, device="cuda", eval_every=2, limit_steps=2,
    )
    assert summary["steps"] >= 2
    assert glob.glob(os.path.join(out_dir, "checkpoint-*.pt"))
    assert 0.0 <= summary["eval/per_example_accuracy"] <= 1.0


def test_quick_inference_e2e_gpu(tmp_path):
    """`run` end-to-end on cuda with the native kernel path."""
    from deepconsensus_amd.inference import quick_inference as qi

    sub, ccs = make_test_bams(tmp_path, n_zmws=4, length=300)
    out = str(tmp_path / "out.fastq")
    options = qi.InferenceOptions(
        batch_size=64, batch_zmws=2, cpus=0, min_quality=0,
        skip_windows_above=0,
    )
    counter = qi.run(
        subreads_to_ccs=sub, ccs_bam=ccs, checkpoint="random",
        output=out, options=options, device="cuda",
    )
    assert counter.total == 4
    stats = json.load(open(tmp_path / "out.inference.json"))
    assert stats["n_zmw_processed"] == 4
    from deepconsensus_amd.dcio.fastq import read_fastq

    recs = list(read_fastq(out))
    assert len(recs) == 4


def test_trained_checkpoint_serves_gpu(tmp_path):
    """Train -> checkpoint -> quick_inference with that checkpoint (cuda)."""
    from deepconsensus_amd.inference import quick_inference as qi
    from deepconsensus_amd.models import train as train_lib

    train_file, _ = make_training_data(tmp_path, n_zmws=3)
    params = _tiny_params(train_file)
    cfg.modify_params(params)
    out_dir = str(tmp_path / "model")
    train_lib.train_model(out_dir, params, device="cuda", eval_every=100,
                          limit_steps=1)

    infer_dir = tmp_path / "infer"
    infer_dir.mkdir()
    sub, ccs = make_test_bams(infer_dir, n_zmws=2, length=150)
    out = str(tmp_path / "served.fastq")
    options = qi.InferenceOptions(batch_size=32, batch_zmws=2, cpus=0,
                                  min_quality=0, skip_windows_above=0)
    counter = qi.run(
        subreads_to_ccs=sub, ccs_bam=ccs, checkpoint=out_dir,
        output=out, options=options, device="cuda",
    )
    assert counter.total == 2


def test_train_bf16_gpu(tmp_path):
    """bf16-autocast training step runs and converges sanely on cuda."""
    from deepconsensus_amd.models import train as train_lib

    train_file, _ = make_training_data(tmp_path)
    params = _tiny_params(train_file)
    out_dir = str(tmp_path / "model_bf16")
    summary = train_lib.train_model(
        out_dir, params, device="cuda", eval_every=100, limit_steps=2,
        use_bf16=True,
    )
    assert summary["steps"] >= 2
    import math
    assert math.isfinite(summary["eval/loss"])


@pytest.mark.gpu
def test_native_int16_staging_matches_fp32(tmp_path):
    """int16 feature staging is bit-identical to fp32 through the native
    path (the embed kernel casts to int either way)."""
    import numpy as np

    from deepconsensus_amd.models import config as cfg
    from deepconsensus_amd.models.model import get_model
    from deepconsensus_amd.models.runner import InferenceRunner

    params = cfg.get_config("transformer_learn_values+custom")
    cfg.modify_params(params, is_training=False)
    torch.manual_seed(11)
    runner = InferenceRunner(params, get_model(params), device="cuda")
    assert runner.native
    rng = np.random.default_rng(0)
    rows = np.zeros((64, params.total_rows, 100), dtype=np.float32)
    mp = params.max_passes
    rows[:, 0:mp] = rng.integers(0, 5, size=(64, mp, 100))
    rows[:, mp:3 * mp] = rng.integers(0, 256, size=(64, 2 * mp, 100))
    rows[:, 3 * mp:4 * mp] = rng.integers(0, 3, size=(64, mp, 100))
    rows[:, 4 * mp] = rng.integers(0, 5, size=(64, 100))
    # fractional SN values exercise the truncation equivalence
    rows[:, -4:] = rng.uniform(3.2, 29.8, size=(64, 4, 1))
    b32, q32 = runner.forward_windows(torch.from_numpy(rows))
    b16, q16 = runner.forward_windows(
        torch.from_numpy(rows.astype(np.int16))
    )
    assert torch.equal(b32.cpu(), b16.cpu())
    assert torch.equal(q32.cpu(), q16.cpu())


@pytest.mark.gpu
def test_embed_grad_kernel_matches_cpu_fallback():
    """The HIP embed_grad kernel produces the same table gradients as the
    CPU index_add fallback."""
    import numpy as np

    from deepconsensus_amd.models import config as cfg
    from deepconsensus_amd.models.embed_stack import EmbedStackFunction, EmbedMeta
    from deepconsensus_amd.models.model import get_model

    params = cfg.get_config("transformer_learn_values+custom")
    cfg.modify_params(params)
    torch.manual_seed(5)
    model = get_model(params)
    meta = EmbedMeta(model)
    tables = [getattr(model, a).table for a in meta.table_attrs]
    rng = np.random.default_rng(6)
    R, L = params.total_rows, params.max_length
    rows = np.zeros((8, R, L), dtype=np.float32)
    mp = params.max_passes
    rows[:, 0:mp] = rng.integers(0, 5, size=(8, mp, L))
    rows[:, mp:3 * mp] = rng.integers(0, 256, size=(8, 2 * mp, L))
    rows[:, 3 * mp:4 * mp] = rng.integers(0, 3, size=(8, mp, L))
    rows[:, 4 * mp] = rng.integers(0, 5, size=(8, L))
    rows[:, -4:] = rng.uniform(3.0, 30.0, size=(8, 4, 1))
    x_cpu = torch.from_numpy(rows)

    def grads_on(device):
        tabs = [t.detach().to(device).requires_grad_(True) for t in tables]
        # use_gather=False on both sides: this test pins the BACKWARD
        # kernel against the CPU index_add fallback with an identical
        # torch forward.
        out = EmbedStackFunction.apply(
            x_cpu.to(device), meta, False, model, *tabs
        )
        out.square().sum().backward()
        return [t.grad.cpu() for t in tabs]

    g_cpu = grads_on("cpu")
    g_gpu = grads_on("cuda:0")
    for a, b in zip(g_cpu, g_gpu):
        torch.testing.assert_close(a, b, atol=2e-3, rtol=1e-3)


def test_distill_bf16_e2e_gpu(tmp_path):
    """Distillation loop on cuda with bf16 autocast: teacher + student
    forwards run the fused MFMA attention path; checkpoint written."""
    from deepconsensus_amd.models import distill as distill_lib
    from deepconsensus_amd.models import train as train_lib

    train_file, _ = make_training_data(tmp_path)
    params = _tiny_params(train_file)
    teacher_dir = str(tmp_path / "teacher")
    train_lib.train_model(
        teacher_dir, params, device="cuda", eval_every=100,
        limit_steps=1,
    )
    sparams = _tiny_params(train_file)
    sparams.num_hidden_layers = max(params.num_hidden_layers - 1, 1)
    from deepconsensus_amd.models.config import modify_params

    modify_params(sparams)
    out = str(tmp_path / "student")
    summary = distill_lib.train_model(
        out, teacher_dir, sparams, device="cuda", limit_steps=2,
        use_bf16=True,
    )
    assert summary["steps"] >= 2
    assert glob.glob(os.path.join(out, "checkpoint-*.pt"))
