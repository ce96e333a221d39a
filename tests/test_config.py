"""Config-system tests mirroring reference model_utils_test.py param plumbing."""
import json
import os


from deepconsensus_amd.models import config as cfg


def test_get_config_production_model():
    p = cfg.get_config("transformer_learn_values+test")
    assert p.model_name == "transformer_learn_values"
    assert p.attn_win_size == 12
    assert p.rezero is True
    assert p.num_heads == 2
    assert p.condense_transformer_input is True
    assert p.transformer_input_size == 280
    assert p.max_passes == 20
    assert p.del_cost == 10.0
    assert p.loss_reg == 0.1
    assert p.max_length == 100


def test_modify_params_derived_hidden_size():
    # Production config: dim=26/pass, hidden = 20*26 + 8 + 32 = 560 -> condensed 280.
    p = cfg.get_config("transformer_learn_values+test")
    cfg.modify_params(p)
    assert p.total_rows == 85
    assert p.embedding_width == 560
    assert p.hidden_size == 280
    assert p.num_hidden_layers == 6
    assert p.filter_size == 2048


def test_modify_params_transformer_hidden_size():
    # Plain transformer: hidden = total_rows (85) padded to even (86).
    p = cfg.get_config("transformer+test")
    cfg.modify_params(p)
    assert p.total_rows == 85
    assert p.hidden_size == 86


def test_modify_params_fc_hidden_size():
    p = cfg.get_config("fc+test")
    cfg.modify_params(p)
    assert p.hidden_size == 85


def test_modify_params_ccs_bq():
    p = cfg.get_config("transformer_learn_values+test_bq")
    cfg.modify_params(p)
    assert p.total_rows == 86
    # dim = 8+8+8+2+8 = 34; hidden = 20*34 + 8 + 8 + 32 = 728 -> condensed 280.
    assert p.embedding_width == 728
    assert p.hidden_size == 280


def test_params_json_round_trip(tmp_path):
    p = cfg.get_config("transformer_learn_values+test")
    cfg.modify_params(p)
    out = cfg.save_params_as_json(str(tmp_path), p)
    assert os.path.exists(out)
    loaded = cfg.read_params_from_json(str(tmp_path))
    assert loaded.hidden_size == 280
    assert loaded.max_passes == 20
    assert loaded.attn_win_size == 12


def test_read_reference_format_params_json(tmp_path):
    """A params.json with the reference's exact key set loads correctly."""
    ref_like = {
        "CCS_BQ_MAX": 95, "IP_MAX": 255, "PW_MAX": 255, "SN_MAX": 500,
        "STRAND_MAX": 2, "add_pos_encoding": True, "attn_win_size": 12,
        "batch_size": 1, "condense_transformer_input": True,
        "del_cost": 10, "filter_size": 2048, "hidden_size": 280,
        "loss_function": "alignment_loss", "loss_reg": 0.1,
        "max_length": 100, "max_passes": 20,
        "model_name": "transformer_learn_values", "num_heads": 2,
        "num_hidden_layers": 6, "per_base_hidden_size": 8,
        "pw_hidden_size": 8, "ip_hidden_size": 8, "sn_hidden_size": 8,
        "strand_hidden_size": 2, "ccs_bq_hidden_size": 8,
        "rezero": True, "total_rows": 85,
        "transformer_input_size": 280, "use_bases": True, "use_ccs": True,
        "use_ccs_bq": False, "use_ip": True, "use_pw": True, "use_sn": True,
        "use_strand": True, "vocab_size": 5,
        "dc_calibration": "0,1.197654,-0.99781",
    }
    with open(tmp_path / "params.json", "w") as f:
        json.dump(ref_like, f)
    p = cfg.read_params_from_json(str(tmp_path))
    assert p.hidden_size == 280
    assert p.dc_calibration == "0,1.197654,-0.99781"
    # Missing keys filled with defaults.
    assert p.layer_postprocess_dropout == 0.1


def test_get_indices():
    idx = cfg.get_indices(20, False)
    assert idx[0] == (0, 20)
    assert idx[3] == (60, 80)
    assert idx[4] == (80, 81)
    assert idx[5] == (0, 0)
    assert idx[6] == (81, 85)
    idx_bq = cfg.get_indices(20, True)
    assert idx_bq[5] == (81, 82)
    assert idx_bq[6] == (82, 86)
