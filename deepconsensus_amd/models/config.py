"""Hyperparameter system for DeepConsensus-AMD.

Re-implements the reference's two-tier config system
(deepconsensus/models/model_configs.py:252-379 `get_config('<model>+<dataset>')`,
deepconsensus/models/model_utils.py:237-354 `modify_params`,
:434-475 params.json round-trip) on a plain dict-backed Params class — no
ml_collections dependency.

The params.json contract matches the reference's shipped
testdata/model/params.json key set, so a reference params.json loads directly.
"""
from __future__ import annotations

import json
import os
from typing import Any, Dict, Optional, Tuple


class Params(dict):
    """Dict with attribute access; the in-memory form of params.json."""

    def __getattr__(self, name: str) -> Any:
        try:
            return self[name]
        except KeyError as e:
            raise AttributeError(name) from e

    def __setattr__(self, name: str, value: Any) -> None:
        self[name] = value

    def copy(self) -> "Params":
        return Params(dict.copy(self))


# Keys merged from the transformer size presets only when absent
# (reference transformer_basic_params.py:33-67 via model_utils.py:347-354).
_TRANSFORMER_SIZE_PRESETS: Dict[str, Dict[str, Any]] = {
    "base": dict(
        default_batch_size=2048,
        default_batch_size_tpu=32768,
        initializer_gain=1.0,
        hidden_size=512,
        num_hidden_layers=6,
        num_heads=8,
        filter_size=2048,
        layer_postprocess_dropout=0.1,
        attention_dropout=0.1,
        relu_dropout=0.1,
        label_smoothing=0.1,
        learning_rate=2.0,
        learning_rate_decay_rate=1.0,
        learning_rate_warmup_steps=16000,
        optimizer_adam_beta1=0.9,
        optimizer_adam_beta2=0.997,
        optimizer_adam_epsilon=1e-09,
        extra_decode_length=50,
        beam_size=4,
        alpha=0.6,
        use_tpu=False,
        static_batch=False,
        allow_ffn_pad=True,
    ),
    "big": dict(),  # filled below
    "tiny": dict(),  # filled below
}
_TRANSFORMER_SIZE_PRESETS["big"] = dict(
    _TRANSFORMER_SIZE_PRESETS["base"],
    default_batch_size=4096,
    default_batch_size_tpu=16384,
    hidden_size=1024,
    filter_size=4096,
    num_heads=16,
)
_TRANSFORMER_SIZE_PRESETS["tiny"] = dict(
    _TRANSFORMER_SIZE_PRESETS["base"],
    default_batch_size=1024,
    default_batch_size_tpu=1024,
    hidden_size=32,
    num_heads=4,
    filter_size=256,
)


def _base_params() -> Params:
    """Common base config (reference model_configs.py:289-351)."""
    p = Params()
    p.trial = 1
    p.rezero = False
    # Feature clipping maxima.
    p.PW_MAX = 255
    p.IP_MAX = 255
    p.SN_MAX = 500
    p.CCS_BQ_MAX = 95
    p.STRAND_MAX = 2
    # Features.
    p.use_bases = True
    p.use_pw = True
    p.use_ip = True
    p.use_strand = True
    p.use_sn = True
    p.use_ccs = True
    p.use_ccs_bq = False
    p.per_base_hidden_size = 1
    p.pw_hidden_size = 1
    p.ip_hidden_size = 1
    p.sn_hidden_size = 1
    p.strand_hidden_size = 1
    p.ccs_bq_hidden_size = 1
    p.total_rows = None
    # Common.
    p.vocab_size = 5
    p.tensorboard_update_freq = "batch"
    p.model_checkpoint_freq = "epoch"
    p.seed = 1
    p.remove_label_gaps = False
    p.loss_function = "alignment_loss"
    # AlignmentLoss.
    p.del_cost = 10.0
    p.loss_reg = 0.1
    p.band_width = None
    # Window.
    p.max_length = 100
    p.model_config_name = "transformer_learn_values"
    p.dataset_config_name = "ccs"
    p.conv_model = "resnet50"
    p.tpu_scale_factor = 1
    return p


def _set_optimizer_params(p: Params) -> None:
    """LAMB optimizer + schedule constants (model_configs.py:115-124)."""
    p.initial_learning_rate = 3.6246e-3
    p.end_learning_rate = 2.86594e-5
    p.warmup_steps = 35536
    p.weight_decay_rate = 6.9868e-3
    p.beta_1 = 0.9
    p.beta_2 = 0.999
    p.epsilon = 1e-6


def _set_base_fc_hparams(p: Params) -> None:
    p.model_name = "fc"
    p.fc_size = [256, 512, 256, 128]
    p.fc_dropout = 0.0
    p.num_channels = 1
    for k in (
        "per_base_hidden_size",
        "pw_hidden_size",
        "ip_hidden_size",
        "strand_hidden_size",
        "ccs_bq_hidden_size",
        "sn_hidden_size",
    ):
        p[k] = 1
    p.l2 = 0.0
    p.batch_size = 256
    p.num_epochs = 15
    p.num_epochs_for_decay = 15
    p.buffer_size = 1_000_000
    _set_optimizer_params(p)


def _set_base_transformer_hparams(p: Params) -> None:
    p.model_name = "transformer"
    p.add_pos_encoding = True
    p.num_heads = 2
    p.layer_norm = False
    p.rezero = True
    p.condense_transformer_input = False
    p.transformer_model_size = "base"
    # Band half-width: attention is masked to |i-j| <= attn_win_size.
    p.attn_win_size = 12
    p.num_channels = 1
    for k in (
        "per_base_hidden_size",
        "pw_hidden_size",
        "ip_hidden_size",
        "sn_hidden_size",
        "ccs_bq_hidden_size",
        "strand_hidden_size",
    ):
        p[k] = 1
    p.layer_postprocess_dropout = 0.1
    p.attention_dropout = 0.1
    p.relu_dropout = 0.1
    p.batch_size = 256
    p.num_epochs = 9
    p.num_epochs_for_decay = 9
    p.buffer_size = 1_000_000
    _set_optimizer_params(p)


def _set_transformer_learned_embeddings_hparams(p: Params) -> None:
    _set_base_transformer_hparams(p)
    p.model_name = "transformer_learn_values"
    p.per_base_hidden_size = 8
    p.pw_hidden_size = 8
    p.ip_hidden_size = 8
    p.strand_hidden_size = 2
    p.sn_hidden_size = 8
    p.ccs_bq_hidden_size = 8
    p.condense_transformer_input = True
    p.transformer_input_size = 280


def _set_transformer_learned_embeddings_distill_hparams(p: Params) -> None:
    _set_transformer_learned_embeddings_hparams(p)
    p.model_name = "transformer_learn_values_distill"
    p.num_hidden_layers = 5
    p.filter_size = 2048
    p.layer_postprocess_dropout = 0.0
    p.attention_dropout = 0.1
    p.relu_dropout = 0.0
    p.init_encoder_stack = True
    p.init_nonencoder_layers = True
    p.teacher_encoder_layers = [1, 2, 3, 4, 5]
    p.student_encoder_layers = [0, 1, 2, 3, 4]
    p.warmup_steps = 0
    p.distill_alpha = 1.0e5
    p.student_alpha = 1.0
    p.temperature = 1.0
    p.logit_loss_identifier = "mean_squared_error"


def _testdata_dir() -> str:
    return os.path.join(os.path.dirname(__file__), "..", "testdata")


def _set_test_data_hparams(p: Params, bq: bool = False) -> None:
    sub = "tf_examples_bq" if bq else "tf_examples"
    base = os.path.join(_testdata_dir(), "human_1m", sub)
    if bq:
        p.use_ccs_bq = True
    p.train_path = [os.path.join(base, "train", "*")]
    p.eval_path = p.train_path
    p.test_path = p.train_path
    p.inference_path = os.path.join(base, "inference", "*")
    p.n_examples_train = 253
    p.n_examples_eval = 253
    p.max_passes = 20
    p.batch_size = 1
    p.num_epochs = 1
    p.buffer_size = 10
    if p.model_name == "fc":
        p.fc_size = [4, 4]


def _set_custom_data_hparams(p: Params) -> None:
    p.tf_dataset = ["/path_to_training_data"]
    p.max_passes = 20


def get_config(config_name: Optional[str] = None) -> Params:
    """Returns the config for '<model_name>+<dataset_name>'.

    Mirrors reference model_configs.get_config (model_configs.py:252-379). The
    OSS reference only ships working branches for the 'test', 'test_bq' and
    'custom' datasets (its poa/ccs/ecoli branches call functions stripped from
    the release); this build treats those names as 'custom'.
    """
    p = _base_params()
    if config_name is None:
        return p

    model_config_name, dataset_config_name = config_name.split("+")
    p.model_config_name = model_config_name
    p.dataset_config_name = dataset_config_name
    p.tf_dataset = None
    p.limit = -1
    if model_config_name == "fc":
        _set_base_fc_hparams(p)
    elif model_config_name == "transformer":
        _set_base_transformer_hparams(p)
    elif model_config_name == "transformer_learn_values":
        _set_transformer_learned_embeddings_hparams(p)
    elif model_config_name == "transformer_learn_values_distill":
        _set_transformer_learned_embeddings_distill_hparams(p)
    else:
        raise ValueError(f"Unknown model_config_name: {model_config_name}")

    if dataset_config_name == "test":
        _set_test_data_hparams(p)
    elif dataset_config_name == "test_bq":
        _set_test_data_hparams(p, bq=True)
    elif dataset_config_name in ("custom", "ccs", "poa", "ecoli", "ccs_test"):
        _set_custom_data_hparams(p)
    else:
        raise ValueError(
            f"dataset_config_name is {dataset_config_name}. Must be one of: "
            "test, test_bq, custom (ccs/poa/ecoli aliases of custom)."
        )
    return p


def get_total_rows(max_passes: int, use_ccs_bq: bool) -> int:
    """Number of rows in the input example (data_providers.py:62-79)."""
    fixed_length = 6 if use_ccs_bq else 5
    return (max_passes * 4) + fixed_length


def get_indices(
    max_passes: int, use_ccs_bq: bool
) -> Tuple[Tuple[int, int], ...]:
    """(start, end) row ranges per feature block (data_providers.py:81-113)."""
    base_indices = (0, max_passes)
    pw_indices = (max_passes, max_passes * 2)
    ip_indices = (max_passes * 2, max_passes * 3)
    strand_indices = (max_passes * 3, max_passes * 4)
    ccs_indices = (max_passes * 4, max_passes * 4 + 1)
    if use_ccs_bq:
        ccs_bq_indices = (max_passes * 4 + 1, max_passes * 4 + 2)
        sn_indices = (max_passes * 4 + 2, max_passes * 4 + 6)
    else:
        ccs_bq_indices = (0, 0)
        sn_indices = (max_passes * 4 + 1, max_passes * 4 + 5)
    return (
        base_indices,
        pw_indices,
        ip_indices,
        strand_indices,
        ccs_indices,
        ccs_bq_indices,
        sn_indices,
    )


def modify_params(
    params: Params,
    speedy: bool = False,
    max_length: Optional[int] = None,
    is_training: bool = True,
    num_devices: int = 1,
) -> None:
    """Derives dependent parameters (reference model_utils.py:237-354).

    In this framework batch_size always means PER-RANK batch; the global batch
    under DP is batch_size * world_size (the reference instead multiplied
    batch_size by num_gpus for its single-process MirroredStrategy).
    num_devices is kept for parity in step-count math only.
    """
    del speedy
    if not is_training:
        for k in ("tf_dataset", "train_path", "eval_path", "test_path",
                  "inference_path"):
            params.pop(k, None)

    if max_length is not None:
        params.max_length = max_length
    if "max_length" not in params:
        raise ValueError("No params.max_length provided.")

    params.total_rows = get_total_rows(
        params.max_passes, params.use_ccs_bq
    )

    if "transformer_learn_values" in params.model_name:
        dim = (
            (params.use_bases * params.per_base_hidden_size)
            + (params.use_pw * params.pw_hidden_size)
            + (params.use_ip * params.ip_hidden_size)
            + (params.use_strand * params.strand_hidden_size)
            + (params.use_ccs_bq * params.ccs_bq_hidden_size)
        )
        params.hidden_size = (
            (params.max_passes * dim)
            + (params.use_ccs * params.per_base_hidden_size)
            + (params.use_ccs_bq * params.ccs_bq_hidden_size)
            + (params.use_sn * params.sn_hidden_size * 4)
        )
    else:
        params.hidden_size = params.total_rows

    if "transformer" in params.model_name and params.hidden_size % 2 != 0:
        params.hidden_size += 1

    if "transformer" in params.model_name:
        params.default_batch_size = params.batch_size
        if params.get("condense_transformer_input"):
            # The embedding concat width stays available as embedding_width.
            params.embedding_width = params.hidden_size
            params.hidden_size = params.transformer_input_size
        preset = _TRANSFORMER_SIZE_PRESETS[
            params.get("transformer_model_size", "base")
        ]
        for k, v in preset.items():
            if k not in params:
                params[k] = v
    params.num_devices = num_devices


def read_params_from_json(checkpoint_path: str) -> Params:
    """Loads params.json sitting next to a checkpoint (model_utils.py:434-465).

    checkpoint_path may be the checkpoint file/dir itself or its directory.
    Unknown keys are preserved; missing base keys get defaults with the same
    "fill from base config" semantics as the reference.
    """
    if os.path.isdir(checkpoint_path):
        json_path = os.path.join(checkpoint_path, "params.json")
    else:
        json_path = os.path.join(
            os.path.dirname(checkpoint_path), "params.json"
        )
    with open(json_path) as f:
        loaded = json.load(f)
    params = _base_params()
    # Model-specific defaults so older params.json files resolve fully.
    model_name = loaded.get("model_name", "transformer_learn_values")
    if model_name == "fc":
        _set_base_fc_hparams(params)
    elif model_name == "transformer":
        _set_base_transformer_hparams(params)
    elif model_name == "transformer_learn_values_distill":
        _set_transformer_learned_embeddings_distill_hparams(params)
    else:
        _set_transformer_learned_embeddings_hparams(params)
    params.update(loaded)
    return params


def save_params_as_json(out_dir: str, params: Params) -> str:
    """Writes params.json (model_utils.py:468-475)."""
    os.makedirs(out_dir, exist_ok=True)
    json_path = os.path.join(out_dir, "params.json")
    serializable = {}
    for k, v in sorted(params.items()):
        try:
            json.dumps(v)
            serializable[k] = v
        except TypeError:
            serializable[k] = str(v)
    with open(json_path, "w") as f:
        json.dump(serializable, f, indent=2, sort_keys=True)
    return json_path
