"""feast_exporter unit contracts (reference parity:
src/test/anovos/feature_store/test_feast_exporter.py — same generated
snippets, validation messages, and timestamp-column behavior)."""

import os
from copy import deepcopy

import pandas as pd
import pytest

from anovos_amd.feature_store import feast_exporter as fe


def _build_config(file_path):
    return {
        "entity": {"name": "test_entity", "id_col": "id column", "description": "test_description"},
        "file_source": {"owner": "test@owner.com", "description": "testcase description",
                         "timestamp_col": "eventtime", "create_timestamp_col": "test_create_column"},
        "feature_view": {"name": "test_view", "ttl_in_seconds": 1, "owner": "pytest@case"},
        "file_path": f"{file_path}",
    }


def test_generate_entity_definition():
    result = fe.generate_entity_definition(
        {"name": "entity", "id_col": "id column", "description": "test_description"})
    assert 'name="entity"' in result
    assert 'description="test_description"' in result
    assert 'join_keys=["id column"]' in result


def test_generate_feature_view():
    result = fe.generate_feature_view(
        types=[("field1", "string")], exclude_list=[],
        config={"name": "test_view", "ttl_in_seconds": 1, "owner": "pytest@case"},
        entity_name="test_entity")
    assert 'name="test_view"' in result
    assert 'entities=["test_entity"]' in result
    assert 'Field(name="field1", dtype=String)' in result
    assert "ttl=timedelta(seconds=1)" in result
    assert 'owner="pytest@case"' in result


def test_generate_field():
    assert fe.generate_field("field", "type").strip() == 'Field(name="field", dtype=type),'


def test_generate_file_source():
    result = fe.generate_file_source(
        {"owner": "test@owner.com", "description": "testcase description",
         "timestamp_col": "eventtime", "create_timestamp_col": "test_create_column"},
        "testfile")
    assert 'path="testfile"' in result
    assert 'timestamp_field="eventtime"' in result
    assert 'created_timestamp_column="test_create_column"' in result
    assert 'description="testcase description"' in result
    assert 'owner="test@owner.com"' in result


def test_generate_feature_service():
    result = fe.generate_feature_service("income_service", "view_name")
    assert "income_service" in result
    assert "view_name" in result
    assert "FeatureService" in result


def test_feature_description_integration(tmp_path):
    config = _build_config(tmp_path)
    fe.generate_feature_description([("field1", "string")], config, file_name="/output/result.csv")
    defs = [f for f in os.listdir(tmp_path) if f.endswith(".py")]
    assert defs
    result = open(os.path.join(tmp_path, defs[0])).read()
    assert config["entity"]["name"] in result
    assert config["file_source"]["owner"] in result
    assert config["feature_view"]["owner"] in result
    assert "/output/result.csv" in result


def test_check_feast_configuration_happy(tmp_path):
    fe.check_feast_configuration(_build_config(tmp_path), 1)


@pytest.mark.parametrize("missing,msg", [
    ("file_source", "Please, provide a file source definition in your config yml!"),
    ("entity", "Please, provide an entity definition in your config yml!"),
    ("feature_view", "Please, provide a feature view definition in your config yml!"),
    ("file_path", "Please, provide a path to the anovos feature_store repository!"),
])
def test_missing_blocks_raise(tmp_path, missing, msg):
    cfg = deepcopy(_build_config(tmp_path))
    del cfg[missing]
    with pytest.raises(ValueError) as e:
        fe.check_feast_configuration(cfg, 1)
    assert e.value.args[0] == msg


def test_faulty_repartition_raises(tmp_path):
    with pytest.raises(ValueError) as e:
        fe.check_feast_configuration(_build_config(tmp_path), 2)
    assert e.value.args[0] == "Please, set repartition parameter to 1 in write_main block in your config yml!"


def test_add_timestamp_columns():
    from anovos_amd.core.frame import AnovosFrame

    idf = AnovosFrame.from_pandas(pd.DataFrame({"a": [1.0, 2.0]}), device="cpu")
    odf = fe.add_timestamp_columns(idf, {"timestamp_col": "eventtime",
                                          "create_timestamp_col": "test_create_column"})
    assert "eventtime" in odf.columns and "test_create_column" in odf.columns
    assert odf.col("eventtime").dtype == "timestamp"
