"""YAML workflow orchestration (reference parity: ``anovos/workflow.py``
:45-889 — the YAML file IS the API: top-level keys are pipeline stages,
nested keys are function kwarg dicts dispatched by reflection, with
save/reread materialization between stages and pre-saved-stats reuse
wired through ``stats_args``).

MI355X-native differences: the execution substrate is the AnovosContext
(one process per GPU, RCCL partial-aggregate merges) instead of a
SparkSession; ``save(..., reread=True)`` keeps the per-stage
restartability contract but the frame stays HBM-resident between stages
unless a write is configured (the reference forced disk round-trips,
workflow.py:64-88).
"""

from __future__ import annotations

import copy
import os
import time
from typing import Dict

import logging

import yaml

from anovos_amd.data_ingest import data_ingest
from anovos_amd.data_analyzer import association_evaluator, quality_checker, stats_generator
from anovos_amd.data_analyzer import ts_analyzer as ts_analyzer_mod
from anovos_amd.data_ingest import ts_auto_detection
from anovos_amd.data_report import report_preprocessing
from anovos_amd.data_report.basic_report_generation import anovos_basic_report
from anovos_amd.data_report.report_generation import anovos_report
from anovos_amd.data_transformer import transformers
from anovos_amd.data_transformer import transformers_advanced
from anovos_amd.drift_stability import drift_detector as ddetector
from anovos_amd.drift_stability import stability as dstability
from anovos_amd.shared.context import init_context


logger = logging.getLogger("anovos_amd.workflow")
if not logger.handlers:
    _h = logging.StreamHandler()
    _h.setFormatter(logging.Formatter("[%(name)s] %(message)s"))
    logger.addHandler(_h)
    logger.setLevel(logging.INFO)


def _log(msg: str):
    logger.info(msg)


def ETL(ctx, args: Dict):
    """Reference workflow.py:45-61 — read_dataset then reflective
    data_ingest transforms."""
    read_args = args.get("read_dataset", None)
    if not read_args:
        raise TypeError("Invalid input for reading dataset")
    df = data_ingest.read_dataset(ctx, **read_args)
    for key, value in args.items():
        if key != "read_dataset" and value is not None:
            f = getattr(data_ingest, key)
            if isinstance(value, dict):
                df = f(df, **value)
            else:
                df = f(df, value)
    return df


def save(data, write_configs, folder_name, reread=False):
    """Reference workflow.py:64-88 — write the frame under
    <file_path>/<folder_name>, optionally re-read it (stage checkpoint).

    ANOVOS_AMD_INMEMORY_PIPELINE=1 skips the per-stage intermediate
    materialization and keeps the frame HBM-resident (SURVEY §7 'hard
    parts': the save/reread barrier must be optional without changing
    results — each stage's output is identical, only the disk round-trip
    and its restart point are elided). write_main / write_stats are
    unaffected."""
    if not write_configs:
        return data if reread else None
    if reread and os.environ.get("ANOVOS_AMD_INMEMORY_PIPELINE", "") in ("1", "true"):
        return data  # only stage checkpoints reread; final/stat writes don't
    if "file_path" not in write_configs:
        raise TypeError("file path missing for writing data")
    write = copy.deepcopy(write_configs)
    write.pop("mlflow_run_id", "")
    write.pop("log_mlflow", False)
    write["file_path"] = write["file_path"] + "/" + folder_name
    data_ingest.write_dataset(data, **write)
    if reread:
        read = copy.deepcopy(write)
        if "file_configs" in read:
            read["file_configs"].pop("repartition", None)
            read["file_configs"].pop("mode", None)
        from anovos_amd.shared.context import get_context

        return data_ingest.read_dataset(get_context(), **read)
    return None


def stats_args(all_configs: Dict, func: str) -> Dict:
    """Reference workflow.py:91-145 — wire pre-saved stats CSVs into
    functions that can reuse them."""
    stats_configs = all_configs.get("stats_generator", None)
    write_configs = all_configs.get("write_stats", None)
    report_configs = all_configs.get("report_preprocessing", None)
    report_input_path = ""
    if report_configs is not None:
        if "master_path" not in report_configs:
            raise TypeError("Master path missing for saving report statistics")
        report_input_path = report_configs.get("master_path")
    result = {}
    if stats_configs:
        mainfunc_to_args = {
            "biasedness_detection": ["stats_mode"],
            "IDness_detection": ["stats_unique"],
            "nullColumns_detection": ["stats_unique", "stats_mode", "stats_missing"],
            "variable_clustering": ["stats_mode"],
            "charts_to_objects": ["stats_unique"],
            "cat_to_num_unsupervised": ["stats_unique"],
            "PCA_latentFeatures": ["stats_missing"],
            "autoencoder_latentFeatures": ["stats_missing"],
        }
        args_to_statsfunc = {
            "stats_unique": "measures_of_cardinality",
            "stats_mode": "measures_of_centralTendency",
            "stats_missing": "measures_of_counts",
        }
        for arg in mainfunc_to_args.get(func, []):
            if report_input_path:
                p = report_input_path + "/" + args_to_statsfunc[arg] + ".csv"
                if not os.path.exists(p):
                    continue  # stats stage hasn't produced it (yet)
                result[arg] = {
                    "file_path": p,
                    "file_type": "csv",
                    "file_configs": {"header": True, "inferSchema": True},
                }
            elif write_configs:
                read = copy.deepcopy(write_configs)
                if "file_configs" in read:
                    read["file_configs"].pop("repartition", None)
                    read["file_configs"].pop("mode", None)
                read["file_path"] = read["file_path"] + "/data_analyzer/stats_generator/" + args_to_statsfunc[arg]
                result[arg] = read
    return result


def main(all_configs: Dict, run_type: str = "local", auth_key_val: Dict = {}, device=None):
    """Reference workflow.py:148-870 — the per-key dispatch loop."""
    from anovos_amd.shared import mlflow_utils

    ctx = init_context(device)
    start_main = time.time()
    # cloud credential side-channel (reference workflow.py:153-158):
    # the last value in auth_key_val is the auth key (e.g. azure SAS)
    auth_key = "NA"
    for _k, _v in (auth_key_val or {}).items():
        auth_key = _v
    mlflow_config = mlflow_utils.setup_mlflow(all_configs.get("mlflow"))
    write_main = all_configs.get("write_main", None)
    write_intermediate = all_configs.get("write_intermediate", None)
    write_stats = all_configs.get("write_stats", None)
    report_input_path = (all_configs.get("report_preprocessing", {}) or {}).get("master_path", "")

    df = ETL(ctx, all_configs.get("input_dataset"))
    report_df = None

    for key, args in all_configs.items():
        if args is None:
            continue
        start = time.time()

        if key == "concatenate_dataset":
            method = args.get("method", "name")
            idfs = [df]
            for k in sorted(x for x in args.keys() if x not in ("method",)):
                idfs.append(ETL(ctx, args[k]))
            df = data_ingest.concatenate_dataset(*idfs, method_type=method)
            new = save(df, write_intermediate, "data_ingest/concatenate_dataset", reread=bool(write_intermediate))
            if new is not None:
                df = new

        elif key == "join_dataset":
            join_cols = args.get("join_cols")
            join_type = args.get("join_type", "inner")
            idfs = [df]
            for k in sorted(x for x in args.keys() if x not in ("join_cols", "join_type")):
                idfs.append(ETL(ctx, args[k]))
            df = data_ingest.join_dataset(*idfs, join_cols=join_cols, join_type=join_type)
            new = save(df, write_intermediate, "data_ingest/join_dataset", reread=bool(write_intermediate))
            if new is not None:
                df = new

        elif key == "geospatial_controller":
            # reference workflow.py:272-421: auto-detection/analyzer plus
            # optional geo_transformations applied to the main frame
            from anovos_amd.data_analyzer.geospatial_analyzer import geospatial_autodetection
            from anovos_amd.data_ingest.geo_auto_detection import ll_gh_cols
            from anovos_amd.data_transformer.geospatial import (
                centroid as geo_centroid,
                geo_format_geohash,
                geo_format_latlon,
                location_in_country,
                rog_calculation,
            )

            ga = args.get("geospatial_analyzer", {}) or {}
            gt = args.get("geo_transformations", False)
            lat_cols, long_cols, gh_cols = [], [], []
            if ga.get("auto_detection_analyzer", True):
                def _grid(v, default):
                    if isinstance(v, str):
                        return [float(x) for x in v.split(",")]
                    return v if v is not None else default

                lat_cols, long_cols, gh_cols = geospatial_autodetection(
                    df,
                    ga.get("id_col", ""),
                    report_input_path or ".",
                    ga.get("max_analysis_records", ga.get("max_records", 100000)),
                    ga.get("top_geo_records", 100),
                    ga.get("max_cluster", 20),
                    _grid(ga.get("eps"), [0.2, 0.8, 0.2]),
                    _grid(ga.get("min_samples"), [25, 100, 25]),
                    ga.get("global_map_box_val", 0),
                    run_type,
                    auth_key_val,
                )
            if gt:
                id_col = gt.get("id_col")
                if not (lat_cols and long_cols) and not gh_cols:
                    lat_cols = gt.get("list_of_lat", []) or []
                    long_cols = gt.get("list_of_lon", []) or []
                    gh_cols = gt.get("list_of_geohash", []) or []
                if gt.get("location_in_country_detection") and lat_cols:
                    df = location_in_country(ctx, df, lat_cols, long_cols, gt.get("country", "US"),
                                             country_shapefile_path=gt.get("country_shapefile_path", ""),
                                             method_type=gt.get("method_type", "approx"),
                                             result_prefix=gt.get("result_prefix_lat_lon", []))
                if gt.get("geo_format_conversion"):
                    if lat_cols:
                        df = geo_format_latlon(df, lat_cols, long_cols, gt.get("loc_input_format", "dd"),
                                               gt.get("loc_output_format", "geohash"),
                                               result_prefix=gt.get("result_prefix_lat_lon", []))
                    if gh_cols:
                        df = geo_format_geohash(df, gh_cols, "dd",
                                                result_prefix=gt.get("result_prefix_geo", []))
                if gt.get("centroid_calculation") and lat_cols and id_col:
                    for idx in range(len(lat_cols)):
                        df_ = geo_centroid(df, lat_cols[idx], long_cols[idx], id_col)
                        df = data_ingest.join_dataset(df, df_, join_cols=id_col, join_type="inner")
                if gt.get("rog_calculation") and lat_cols and id_col:
                    for idx in range(len(lat_cols)):
                        df_ = rog_calculation(df, lat_cols[idx], long_cols[idx], id_col)
                        df = data_ingest.join_dataset(df, df_, join_cols=id_col, join_type="inner")

        elif key == "timeseries_analyzer":
            id_col = args.get("id_col", "")
            out_path = report_input_path or "."
            if args.get("auto_detection", True):
                df, ts_cols, num_cols, cat_cols = ts_auto_detection.ts_preprocess(
                    ctx, df, id_col, out_path, tz_offset=args.get("tz_offset", "local"), run_type=run_type, auth_key=auth_key)
            if args.get("inspection", True):
                # reference key analysis_level ∈ {daily, weekly, hourly}
                ts_analyzer_mod.ts_analyzer(
                    ctx, df, id_col, args.get("max_days", 90), out_path,
                    output_type=args.get("analysis_level", args.get("output_type", "daily")),
                    run_type=run_type, auth_key=auth_key)

        elif key == "anovos_basic_report" and args.get("basic_report", False):
            anovos_basic_report(ctx, df, **(args.get("report_args", {}) or {}), run_type=run_type, auth_key=auth_key)
            _log("anovos_basic_report completed — skipping remaining stages (reference workflow.py:468-486)")
            return df

        elif key == "stats_generator":
            for m in args["metric"]:
                f = getattr(stats_generator, m)
                stats = f(ctx, df, **args.get("metric_args", {}), print_impact=False)
                if report_input_path:
                    report_preprocessing.save_stats(ctx, stats, report_input_path, m, run_type=run_type, auth_key=auth_key)
                if write_stats:
                    save_df = stats
                    wc = copy.deepcopy(write_stats)
                    wc["file_path"] = wc["file_path"] + "/data_analyzer/stats_generator/" + m
                    import pandas as pd

                    os.makedirs(wc["file_path"], exist_ok=True)
                    save_df.to_csv(os.path.join(wc["file_path"], "part-00000.csv"), index=False)
                _log(f"stats_generator.{m}: {time.time() - start:.3f}s")
                start = time.time()

        elif key == "quality_checker":
            for subkey, value in args.items():
                if value is None:
                    continue
                f = getattr(quality_checker, subkey)
                extra = stats_args(all_configs, subkey)
                df, df_stats = f(ctx, df, **value, **extra)
                new = save(df, write_intermediate, "data_analyzer/quality_checker/" + subkey,
                           reread=bool(write_intermediate))
                if new is not None:
                    df = new
                if report_input_path:
                    report_preprocessing.save_stats(ctx, df_stats, report_input_path, subkey, run_type=run_type, auth_key=auth_key)
                _log(f"quality_checker.{subkey}: {time.time() - start:.3f}s")
                start = time.time()

        elif key == "association_evaluator":
            for subkey, value in args.items():
                if value is None:
                    continue
                f = getattr(association_evaluator, subkey)
                extra = stats_args(all_configs, subkey)
                if subkey == "correlation_matrix" and all_configs.get("cat_to_num_transformer"):
                    # reference workflow.py:600-602: encode categoricals
                    # on a frame copy before the correlation matrix
                    df_corr = transformers.cat_to_num_transformer(
                        ctx, df, **all_configs["cat_to_num_transformer"])
                    stats = f(ctx, df_corr, **value, **extra)
                else:
                    stats = f(ctx, df, **value, **extra)
                if report_input_path:
                    report_preprocessing.save_stats(ctx, stats, report_input_path, subkey, run_type=run_type, auth_key=auth_key)
                _log(f"association_evaluator.{subkey}: {time.time() - start:.3f}s")
                start = time.time()

        elif key == "drift_detector":
            for subkey, value in args.items():
                if value is None:
                    continue
                if subkey == "drift_statistics":
                    configs = copy.deepcopy(value.get("configs", {}))
                    source_args = value.get("source_dataset", None)
                    if not configs.get("pre_existing_source", False) and source_args:
                        idf_source = ETL(ctx, source_args)
                    else:
                        idf_source = None
                    stats = ddetector.statistics(ctx, df, idf_source, **configs)
                    if report_input_path:
                        report_preprocessing.save_stats(ctx, stats, report_input_path, "drift_statistics", run_type=run_type, auth_key=auth_key)
                elif subkey == "stability_index":
                    configs = copy.deepcopy(value.get("configs", {}))
                    idfs = []
                    for k in sorted(x for x in value.keys() if x.startswith("dataset")):
                        idfs.append(ETL(ctx, value[k]))
                    stats = dstability.stability_index_computation(ctx, *idfs, **configs)
                    if report_input_path:
                        report_preprocessing.save_stats(ctx, stats, report_input_path, "stability_index", run_type=run_type, auth_key=auth_key)
                        # per-snapshot metric history feeds the report's
                        # stability line charts (reference workflow.py:700-721)
                        amp = (configs or {}).get("appended_metric_path", "")
                        if amp:
                            try:
                                from anovos_amd.core.io import read_dataset as _rd

                                df_metrics = _rd(amp, "csv", {"header": True}).to_pandas()
                                report_preprocessing.save_stats(ctx, df_metrics, report_input_path,
                                                                "stabilityIndex_metrics", run_type=run_type, auth_key=auth_key)
                            except Exception as e:
                                logger.warning(f"stabilityIndex_metrics not persisted: {e}")
                _log(f"drift_detector.{subkey}: {time.time() - start:.3f}s")
                start = time.time()

        elif key == "transformers":
            for subkey, value in args.items():
                if value is None:
                    continue
                for subkey2, value2 in value.items():
                    if value2 is None:
                        continue
                    mod = transformers if hasattr(transformers, subkey2) else transformers_advanced
                    f = getattr(mod, subkey2)
                    extra = stats_args(all_configs, subkey2)
                    import inspect

                    first = next(iter(inspect.signature(f).parameters))
                    df = f(ctx, df, **value2, **extra) if first == "ctx" else f(df, **value2, **extra)
                    new = save(df, write_intermediate, "data_transformer/" + subkey2,
                               reread=bool(write_intermediate))
                    if new is not None:
                        df = new
                    _log(f"transformers.{subkey2}: {time.time() - start:.3f}s")
                    start = time.time()

        elif key == "report_preprocessing":
            for subkey, value in args.items():
                if subkey == "master_path" or value is None:
                    continue
                if subkey == "charts_to_objects":
                    extra = stats_args(all_configs, subkey)
                    report_preprocessing.charts_to_objects(
                        ctx, df, **value, **extra, master_path=args["master_path"], run_type=run_type, auth_key=auth_key)
                    _log(f"report_preprocessing.charts_to_objects: {time.time() - start:.3f}s")
                    start = time.time()

        elif key == "report_generation":
            anovos_report(**args, run_type=run_type, auth_key=auth_key)
            _log(f"report_generation: {time.time() - start:.3f}s")

        elif key == "write_feast_features":
            from anovos_amd.feature_store import feast_exporter

            file_source_config = args.get("file_source", {})
            df = feast_exporter.add_timestamp_columns(df, file_source_config)
            feast_exporter.generate_feature_description(df.dtypes, args)

    if write_main:
        save(df, write_main, "final_dataset")
        if mlflow_config is not None and mlflow_config.get("track_output"):
            mlflow_utils.log_artifacts(write_main["file_path"], mlflow_config, "final_dataset")
    if mlflow_config is not None and report_input_path and mlflow_config.get("track_reports"):
        mlflow_utils.log_artifacts(report_input_path, mlflow_config, "report_stats")
    mlflow_utils.end_run(mlflow_config)
    _log(f"workflow total: {time.time() - start_main:.3f}s")
    return df


def run(config_path: str, run_type: str = "local", auth_key_val: Dict = {}, device=None):
    """Reference workflow.py:873-889."""
    with open(config_path, "r") as f:
        all_configs = yaml.load(f, yaml.SafeLoader)
    return main(all_configs, run_type, auth_key_val, device=device)
