"""Feature recommender + feast exporter tests (reference parity:
src/test feature_recommender 16 tests, feast 10 tests — condensed)."""

import os

import pandas as pd
import pytest

from anovos_amd.feature_recommender import featrec_init as fi
from anovos_amd.feature_recommender import feature_explorer as fe
from anovos_amd.feature_recommender import feature_mapper as fm
from anovos_amd.feature_store import feast_exporter


def test_list_all_industry_usecase_pair():
    ind = fe.list_all_industry()
    uc = fe.list_all_usecase()
    pairs = fe.list_all_pair()
    assert "banking" in list(ind["Industry"])
    assert "customer churn prediction" in list(uc["Usecase"])
    assert len(pairs) >= len(ind)


def test_process_exact_and_semantic():
    assert fe.process_industry("banking", semantic=False) == "banking"
    assert fe.process_industry("  Banking ", semantic=True) == "banking"
    # semantic snap: "bank" should match "banking"
    assert fe.process_industry("bank", semantic=True) == "banking"


def test_list_feature_by_industry_and_pair():
    feats = fe.list_feature_by_industry("banking", num_of_feat=3)
    assert len(feats) == 3
    pair = fe.list_feature_by_pair("banking", "fraud detection")
    assert all(pair["Industry"] == "banking")
    assert all(pair["Usecase"] == "fraud detection")


def test_feature_mapper():
    df = pd.DataFrame(
        {
            "attr": ["num_late_payments", "data_usage_monthly_gb", "zzz_qqq_xxx"],
            "desc": ["count of late payments in last year", "gigabytes of mobile data used per month", "opaque code"],
        }
    )
    out = fm.feature_mapper(df, name_column="attr", desc_column="desc", top_n=1, threshold=0.2)
    assert len(out) == 3
    m = out[out["Input_Attribute_Name"] == "num_late_payments"].iloc[0]
    assert m["Recommended_Feature_Name"] == "num_late_payments_12m"
    z = out[out["Input_Attribute_Name"] == "zzz_qqq_xxx"].iloc[0]
    assert z["Recommended_Feature_Name"] == "Null"


def test_find_attr_by_relevance():
    df = pd.DataFrame({"attr": ["credit_utilization_ratio", "avg_session_duration"],
                       "desc": ["balance over limit ratio", "minutes played per session"]})
    out = fm.find_attr_by_relevance(df, ["credit risk features based on utilization"],
                                    name_column="attr", desc_column="desc", threshold=0.1)
    assert out["Recommended_Input_Attribute_Name"].iloc[0] == "credit_utilization_ratio"


def test_sankey_visualization():
    df = pd.DataFrame({"attr": ["num_late_payments"], "desc": ["late payments count"]})
    mapped = fm.feature_mapper(df, name_column="attr", desc_column="desc", top_n=1, threshold=0.1)
    fig = fm.sankey_visualization(mapped, industry_included=True, usecase_included=True)
    assert fig.data[0].type == "sankey"


def test_custom_corpus(tmp_path):
    p = tmp_path / "corpus.csv"
    pd.DataFrame({"Feature Name": ["f1"], "Feature Description": ["a thing"],
                  "Industry": ["Aerospace"], "Usecase": ["Anomaly Detection"]}).to_csv(p, index=False)
    fi.set_corpus_path(str(p))
    try:
        ind = fe.list_all_industry()
        assert list(ind["Industry"]) == ["aerospace"]
    finally:
        fi.set_corpus_path(None)


def test_feast_exporter(tmp_path):
    cfg = {
        "file_path": str(tmp_path / "feast_repo"),
        "entity": {"name": "customer", "id_col": "ifa", "description": "customer id"},
        "file_source": {"timestamp_col": "event_ts", "create_timestamp_col": "create_ts",
                         "description": "anovos output", "owner": "me@example.com"},
        "feature_view": {"name": "income_view", "ttl_in_seconds": 3600, "owner": "me@example.com"},
        "service_name": "income_service",
    }
    types = [("ifa", "string"), ("age", "int"), ("income", "float"),
             ("event_ts", "timestamp"), ("create_ts", "timestamp")]
    path = feast_exporter.generate_feature_description(types, cfg, "data.parquet")
    assert os.path.exists(path)
    code = open(path).read()
    assert 'name="income_view"' in code
    assert 'Field(name="age", dtype=Int64)' in code
    assert "ifa" not in code.split("schema=[")[1].split("]")[0]  # id excluded from schema
    assert "income_service" in code
    compile(code, path, "exec")  # syntactically valid python


def test_feast_add_timestamp_columns():
    import numpy as np

    from anovos_amd.core.frame import AnovosFrame

    idf = AnovosFrame.from_pandas(pd.DataFrame({"a": [1.0, 2.0]}), device="cpu")
    odf = feast_exporter.add_timestamp_columns(idf, {"timestamp_col": "event_ts", "create_timestamp_col": "create_ts"})
    assert "event_ts" in odf.columns and "create_ts" in odf.columns
    assert odf.col("event_ts").dtype == "timestamp"


def test_feast_config_validation():
    with pytest.raises(ValueError):
        feast_exporter.check_feast_configuration({}, 1)
    with pytest.raises(ValueError):
        feast_exporter.check_feast_configuration({"file_path": "x", "entity": {}, "file_source": {}, "feature_view": {}}, 2)
