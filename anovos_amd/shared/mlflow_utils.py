"""Optional MLflow integration (reference parity: workflow.py:184-214,
report_preprocessing.py:94-95 — tracking URI/experiment from the YAML
``mlflow`` block, artifact logging for datasets/stats/reports).

mlflow is not part of this stack's base image; every hook degrades to a
no-op when the package is absent, so configs carrying mlflow keys run
unchanged."""

from __future__ import annotations

from typing import Dict, Optional

try:
    import mlflow  # type: ignore

    HAS_MLFLOW = True
except ImportError:  # pragma: no cover - absent in this image
    mlflow = None
    HAS_MLFLOW = False


def setup_mlflow(cfg: Optional[Dict]) -> Optional[Dict]:
    """Configure tracking URI / experiment and start a run. Returns the
    (augmented) mlflow_config with run_id, or None."""
    if not cfg or not HAS_MLFLOW:
        return None
    if cfg.get("tracking_uri"):
        mlflow.set_tracking_uri(cfg["tracking_uri"])
    if cfg.get("experiment"):
        mlflow.set_experiment(cfg["experiment"])
    run = mlflow.start_run()
    out = dict(cfg)
    out["run_id"] = run.info.run_id
    return out


def log_artifact(path: str, mlflow_config: Optional[Dict], artifact_path: Optional[str] = None):
    if mlflow_config is None or not HAS_MLFLOW:
        return
    try:
        mlflow.log_artifact(path, artifact_path=artifact_path)
    except Exception:
        pass


def log_artifacts(local_dir: str, mlflow_config: Optional[Dict], artifact_path: Optional[str] = None):
    if mlflow_config is None or not HAS_MLFLOW:
        return
    try:
        mlflow.log_artifacts(local_dir, artifact_path=artifact_path)
    except Exception:
        pass


def end_run(mlflow_config: Optional[Dict]):
    if mlflow_config is None or not HAS_MLFLOW:
        return
    try:
        mlflow.end_run()
    except Exception:
        pass
