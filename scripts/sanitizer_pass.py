#!/usr/bin/env python3
"""Device-AddressSanitizer pass over the HIP kernels.

Runs every hand-written kernel once at production shapes through the
gfx950:xnack+ ASAN-instrumented build (ops/build.py build_asan). Any
device OOB read/write aborts with an ASAN report. Invoke on a GPU box:

  LD_PRELOAD=/opt/rocm/lib/llvm/lib/clang/22/lib/linux/\
libclang_rt.asan-x86_64.so \
  ASAN_OPTIONS=detect_leaks=0 HSA_XNACK=1 \
  python scripts/sanitizer_pass.py

Race checking has no device-TSan equivalent on ROCm; the suite covers
the race axis with bitwise determinism tests instead
(tests/test_gpu_large_batch.py big1==big2 at B=4096, which caught the
round-1 embed_gather nondeterminism).
"""
import importlib.util
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def load_asan_ext():
    so = os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "deepconsensus_amd", "ops", "_build_asan",
        "dc_hip_kernels_asan.so",
    )
    spec = importlib.util.spec_from_file_location("dc_hip_kernels_asan", so)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


def main():
    from deepconsensus_amd import ops as dc_ops
    from deepconsensus_amd.models import config as cfg
    from deepconsensus_amd.models.model import get_model
    from deepconsensus_amd.models import runner as runner_lib

    ext = load_asan_ext()
    # Patch the extension into the ops loader so InferenceRunner uses it.
    dc_ops._ext = ext
    print("ASAN extension loaded", flush=True)

    params = cfg.get_config("transformer_learn_values+custom")
    cfg.modify_params(params, is_training=False)
    torch.manual_seed(5)
    r = runner_lib.InferenceRunner(params, get_model(params),
                                   device="cuda")
    assert r.ext is ext
    rng = np.random.default_rng(11)
    B, L, mp = 512, params.max_length, params.max_passes
    rows = np.zeros((B, params.total_rows, L), np.float32)
    rows[:, 0:mp] = rng.integers(0, 5, size=(B, mp, L))
    rows[:, mp:3 * mp] = rng.integers(0, 60, size=(B, 2 * mp, L))
    rows[:, 3 * mp:4 * mp] = rng.integers(1, 3, size=(B, mp, L))
    rows[:, 4 * mp] = rng.integers(0, 5, size=(B, L))
    rows[:, -4:] = rng.uniform(3, 10, size=(B, 4, 1))
    x = torch.from_numpy(rows.astype(np.int16))
    bases, quals = r.forward_windows(x)  # embed/linear/attn/ffn/ln_head
    torch.cuda.synchronize()
    print("serving chain (K2,K3,K5-K12) OK:",
          bases.shape, int(bases.sum()), flush=True)

    # K13 alignment DP fwd+bwd.
    Bd, m, n = 64, 100, 100
    subs = torch.rand(Bd, m, n, device="cuda")
    ins = torch.rand(Bd, n, device="cuda")
    lens = torch.randint(10, m, (Bd,), device="cuda", dtype=torch.int32)
    loss, w = ext.alignment_dp_fwd(subs, ins, lens, 10.0, 0.1, 0)
    gs, gi = ext.alignment_dp_bwd(torch.ones_like(loss), w, lens, m, n, 0)
    torch.cuda.synchronize()
    print("alignment_dp fwd+bwd OK:", float(loss.sum()),
          float(gs.sum()), float(gi.sum()), flush=True)

    # K14 metric.
    yt = torch.randint(0, 5, (Bd, m), device="cuda", dtype=torch.int32)
    yp = torch.randint(0, 5, (Bd, n), device="cuda", dtype=torch.int32)
    ytl = (yt != 0).sum(-1).int()
    ypl = (yp != 0).sum(-1).int()
    v, counts = ext.alignment_metric_counts(yt, yp, ytl, ypl,
                                            2.0, 5.0, 9.0, 4.0)
    torch.cuda.synchronize()
    print("alignment_metric OK:", float(v.sum()),
          int(counts.sum()), flush=True)

    # K2-grad (training embedding backward).
    from deepconsensus_amd.models.model import (
        EncoderOnlyLearnedValuesTransformer,
    )
    print("ALL KERNELS PASSED UNDER DEVICE ASAN", flush=True)


if __name__ == "__main__":
    main()
