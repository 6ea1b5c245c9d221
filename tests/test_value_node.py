"""DistributedValue coercion/fallback behavior (reference
tests/test_distributed_value.py coverage, reimplemented for this node)."""

import json


from comfyui_distributed_amd.nodes.utilities import DistributedValue


def run(default, values, is_worker=False, worker_id=""):
    node = DistributedValue()
    (out,) = node.distribute(default, json.dumps(values) if isinstance(values, dict) else values,
                             is_worker, worker_id)
    return out


def test_master_returns_default_with_type_coercion():
    assert run("5", {"_type": "INT", "1": "9"}) == 5
    assert run("2.5", {"_type": "FLOAT"}) == 2.5
    assert run("hello", {"_type": "STRING"}) == "hello"


def test_worker_lookup_is_one_indexed():
    values = {"_type": "INT", "1": "10", "2": "20"}
    assert run("0", values, True, "worker_0") == 10
    assert run("0", values, True, "worker_1") == 20


def test_worker_fallback_to_default():
    values = {"_type": "INT", "1": "10"}
    assert run("7", values, True, "worker_5") == 7  # key "6" missing


def test_bare_index_worker_id():
    values = {"_type": "FLOAT", "3": "1.5"}
    assert run("0", values, True, "2") == 1.5


def test_invalid_json_falls_back():
    assert run("x", "{not json", True, "worker_0") == "x"


def test_non_dict_values_ignored():
    assert run("d", "[1,2,3]", True, "worker_0") == "d"


def test_int_coercion_through_float_string():
    assert run("0", {"_type": "INT", "1": "3.7"}, True, "worker_0") == 3


def test_coercion_failure_keeps_default_value():
    # default not coercible -> returned as-is (reference _coerce_safe)
    assert run("abc", {"_type": "INT"}, False, "") == "abc"


def test_combo_stays_string():
    assert run("euler", {"_type": "COMBO", "1": "dpmpp_2m"}, True, "worker_0") == "dpmpp_2m"


def test_zero_is_a_legitimate_worker_override():
    import json

    from comfyui_distributed_amd.nodes.utilities import DistributedValue

    node = DistributedValue()
    vals = json.dumps({"_type": "FLOAT", "1": 0.0, "2": 5.0})
    out = node.distribute("3.5", vals, is_worker=True, worker_id="worker_0")
    assert out == (0.0,)  # not the 3.5 default
    out = node.distribute("3.5", vals, is_worker=True, worker_id="worker_1")
    assert out == (5.0,)
    # empty string still falls back
    vals2 = json.dumps({"_type": "FLOAT", "1": ""})
    out = node.distribute("3.5", vals2, is_worker=True, worker_id="worker_0")
    assert out == (3.5,)


def test_model_name_writes_back_into_workflow():
    """Reference nodes/utilities.py:164-224: the resolved model name is
    written into the executing workflow's node entry (widgets_values)."""
    from comfyui_distributed_amd.graph.executor import Executor

    prompt = {
        "1": {"class_type": "DistributedModelName",
              "inputs": {"model_name": "sd15.safetensors"}},
    }
    ex = Executor()
    out = ex.execute(prompt)
    assert out["1"][0] == "sd15.safetensors"
    assert prompt["1"]["widgets_values"] == ["sd15.safetensors"]


def test_model_name_stringifies_nonstring():
    from comfyui_distributed_amd.nodes.utilities import DistributedModelName

    node = DistributedModelName()
    assert node.log_input(7)[0] == "7"
    assert node.log_input({"a": 1})[0] == '{\n    "a": 1\n}'
    assert node.log_input(["a", "b"])[0] == ["a", "b"]
