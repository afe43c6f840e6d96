import json

import pytest

from metis_amd.planner.validate import CostValidator, plan_key


def test_plan_key():
    assert plan_key(8, 1, 1, 2, 16) == "dp8_tp1_pp1_mbs2_gbs16"


def test_validation_math(tmp_path):
    doc = {"runs": [
        {"plan": {"dp": 8, "tp": 1, "pp": 1, "mbs": 2, "gbs": 16}, "measured_ms": 100.0},
        {"plan": {"dp": 4, "tp": 2, "pp": 1, "mbs": 2, "gbs": 16}, "measured_ms": 200.0},
    ]}
    path = tmp_path / "measured.json"
    path.write_text(json.dumps(doc))

    v = CostValidator(str(path), error_threshold_pct=10.0)
    result = v.validate({
        "dp8_tp1_pp1_mbs2_gbs16": 105.0,   # 5% error
        "dp4_tp2_pp1_mbs2_gbs16": 260.0,   # 30% error
        "dp2_tp4_pp1_mbs2_gbs16": 50.0,    # no measurement -> ignored
    })
    assert result.num_validated == 2
    assert result.mean_abs_error_pct == pytest.approx((5 + 30) / 2)
    assert result.max_abs_error_pct == pytest.approx(30.0)
    assert result.num_within_tolerance == 1
