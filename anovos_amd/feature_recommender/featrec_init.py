"""Feature-recommender init: corpus load + embedding model (reference
parity: ``anovos/feature_recommender/featrec_init.py`` :42-243).

The reference lazy-loads sentence-transformers ``all-mpnet-base-v2`` and
caches corpus embeddings; that model cannot be fetched in this offline
stack, so the semantic backend is a TF-IDF vectorizer (sklearn,
char+word n-grams) over the same prepared corpus text — deterministic,
offline, and API-compatible (cosine-similarity semantic search). A
custom corpus CSV (columns: "Feature Name", "Feature Description",
Industry, Usecase) can be supplied via set_corpus_path()."""

from __future__ import annotations

import os
import re
from typing import List, Optional, Tuple

import numpy as np
import pandas as pd

_CORPUS_PATH: Optional[str] = None
_STATE = {}

_SEED_CORPUS = [
    # (feature name, description, industry, usecase) — native seed corpus
    ("days_since_last_purchase", "Number of days since the customer last made a purchase", "retail", "customer churn prediction"),
    ("total_purchase_amount_90d", "Total amount spent by the customer in the last 90 days", "retail", "customer lifetime value"),
    ("avg_basket_size", "Average number of items per transaction", "retail", "demand forecasting"),
    ("coupon_redemption_rate", "Share of offered coupons the customer redeemed", "retail", "campaign response prediction"),
    ("sessions_per_week", "Average number of app sessions per week", "gaming", "customer churn prediction"),
    ("in_app_purchase_count", "Number of in-app purchases made by the player", "gaming", "monetization"),
    ("days_active_last_30d", "Days with at least one login in the last 30 days", "gaming", "engagement scoring"),
    ("avg_session_duration", "Average duration of a play session in minutes", "gaming", "engagement scoring"),
    ("credit_utilization_ratio", "Ratio of outstanding balance to total credit limit", "banking", "credit risk scoring"),
    ("num_late_payments_12m", "Number of late payments in the last 12 months", "banking", "credit risk scoring"),
    ("avg_monthly_balance", "Average end-of-month account balance", "banking", "customer lifetime value"),
    ("num_products_held", "Number of distinct banking products held by the customer", "banking", "cross sell prediction"),
    ("txn_amount_stddev", "Standard deviation of transaction amounts", "banking", "fraud detection"),
    ("intl_txn_share", "Share of transactions made outside the home country", "banking", "fraud detection"),
    ("claim_frequency_3y", "Number of insurance claims filed in the last three years", "insurance", "claim risk scoring"),
    ("policy_tenure_months", "Months since the policy was first issued", "insurance", "customer churn prediction"),
    ("premium_to_income_ratio", "Ratio of annual premium to declared annual income", "insurance", "underwriting"),
    ("data_usage_gb_month", "Mobile data consumed per month in gigabytes", "telecom", "customer churn prediction"),
    ("dropped_call_rate", "Share of calls that were dropped in the last month", "telecom", "customer churn prediction"),
    ("plan_upgrade_count", "Number of plan upgrades in the customer lifetime", "telecom", "cross sell prediction"),
    ("support_tickets_90d", "Number of customer-support tickets in the last 90 days", "telecom", "customer satisfaction"),
    ("time_on_site_minutes", "Total minutes spent on the site per visit", "ecommerce", "conversion prediction"),
    ("cart_abandonment_rate", "Share of carts created but not checked out", "ecommerce", "conversion prediction"),
    ("product_view_count_7d", "Number of product pages viewed in the last 7 days", "ecommerce", "recommendation"),
    ("review_sentiment_score", "Average sentiment score of reviews written by the user", "ecommerce", "customer satisfaction"),
    ("distinct_categories_bought", "Number of distinct product categories purchased", "ecommerce", "customer lifetime value"),
    ("avg_delivery_delay_days", "Average delay between promised and actual delivery", "logistics", "delivery performance"),
    ("ontime_pickup_indicator", "Indicator of on-time pickup, 1 true 0 false", "logistics", "delivery performance"),
    ("route_distance_km", "Planned route distance in kilometers", "logistics", "demand forecasting"),
    ("vehicle_idle_hours", "Hours the vehicle spent idle per day", "logistics", "fleet utilization"),
    ("readmission_within_30d", "Whether the patient was readmitted within 30 days", "healthcare", "readmission prediction"),
    ("num_chronic_conditions", "Number of chronic conditions on record", "healthcare", "risk stratification"),
    ("medication_adherence_rate", "Share of prescribed doses actually taken", "healthcare", "treatment adherence"),
    ("avg_lab_glucose", "Average blood glucose over recent lab tests", "healthcare", "risk stratification"),
    ("energy_usage_kwh_month", "Household energy consumption per month in kWh", "utilities", "demand forecasting"),
    ("payment_failure_count", "Number of failed bill payments in the last year", "utilities", "credit risk scoring"),
    ("smart_meter_flag", "Whether the household has a smart meter installed", "utilities", "customer segmentation"),
    ("viewing_hours_week", "Hours of content watched per week", "media", "customer churn prediction"),
    ("content_diversity_index", "Diversity of genres consumed by the viewer", "media", "recommendation"),
    ("ad_click_through_rate", "Share of served ads the user clicked", "media", "campaign response prediction"),
]


def detect_model_path() -> str:
    """Local sentence-transformers model path, if one was pre-downloaded
    (reference featrec_init.py:11-33). The engine itself embeds with the
    built-in TF-IDF embedder and never requires this model; the path is
    honored when a user drops the reference's `all-mpnet-base-v2` cache
    in the conventional location."""
    transformers_path = os.getenv("SENTENCE_TRANSFORMERS_HOME")
    if transformers_path is None:
        torch_home = os.path.expanduser(
            os.getenv("TORCH_HOME", os.path.join(os.getenv("XDG_CACHE_HOME", "~/.cache"), "torch"))
        )
        transformers_path = os.path.join(torch_home, "sentence_transformers")
    return os.path.join(transformers_path, "sentence-transformers_all-mpnet-base-v2")


def model_download():
    """Reference featrec_init.py:36 downloads `all-mpnet-base-v2`. This
    deployment has no package for it and typically no egress; the
    recommender runs on the built-in TF-IDF embedder instead, so the
    download is optional. Raises with that explanation when the
    sentence-transformers package is unavailable."""
    try:
        from sentence_transformers import SentenceTransformer  # noqa: F401
    except ImportError as e:
        raise RuntimeError(
            "sentence-transformers is not installed in this environment; "
            "the feature recommender uses its built-in TF-IDF embedder "
            "(TfidfEmbedder/semantic_search) and does not need the download."
        ) from e
    print("Starting the Semantic Model download")
    SentenceTransformer("all-mpnet-base-v2")
    print("Model downloading finished")


def set_corpus_path(path: Optional[str]):
    """Point the recommender at a custom corpus CSV; None resets to the
    built-in seed corpus."""
    global _CORPUS_PATH
    _CORPUS_PATH = path
    _STATE.clear()


def _default_corpus_path() -> Optional[str]:
    """The full 1,085-entry corpus shipped with the package (carried over
    from the reference's data/feature_recommender/flatten_fr_db.csv —
    see data/README.md for provenance)."""
    p = os.path.join(os.path.dirname(os.path.abspath(__file__)), "data", "flatten_fr_db.csv")
    return p if os.path.exists(p) else None


def init_input_fer() -> pd.DataFrame:
    """Reference featrec_init.py — load the flattened feature corpus."""
    if "df" in _STATE:
        return _STATE["df"]
    path = _CORPUS_PATH or _default_corpus_path()
    if path:
        df = pd.read_csv(path)
    else:
        df = pd.DataFrame(_SEED_CORPUS, columns=["Feature Name", "Feature Description", "Industry", "Usecase"])
    df["Industry"] = df["Industry"].astype(str).str.strip().str.lower()
    df["Usecase"] = df["Usecase"].astype(str).str.strip().str.lower()
    _STATE["df"] = df
    return df


def get_column_name(df: pd.DataFrame) -> Tuple[str, str, str, str]:
    """Reference featrec_init.py:get_column_name — corpus column names."""
    cols = list(df.columns)
    return cols[0], cols[1], cols[2], cols[3]


def camel_case_split(input) -> str:  # reference arg name (featrec_init.py)
    """Reference featrec_init.py:114-130 — split camelCase boundaries,
    each segment emitted with a trailing space (exact upstream output:
    'accountWeeks' → 'account Weeks ')."""
    out = ""
    for m in re.finditer(r".+?(?:(?<=[a-z])(?=[A-Z])|(?<=[A-Z])(?=[A-Z][a-z])|$)", str(input)):
        out += str(m.group(0)) + " "
    return out


def _clean_text(s: str) -> str:
    s = camel_case_split(s)
    s = re.sub(r"[_\-/\.]", " ", s)
    s = re.sub(r"[^0-9a-zA-Z ]+", " ", s)
    return re.sub(r"\s+", " ", s).strip().lower()


def recommendation_data_prep(df: pd.DataFrame, name_column: Optional[str], desc_column: Optional[str]) -> Tuple[pd.DataFrame, List[str]]:
    """Reference featrec_init.py:recommendation_data_prep — combine and
    clean name+description text for embedding."""
    out = df.copy()
    parts = []
    if name_column:
        parts.append(out[name_column].fillna("").map(_clean_text))
    if desc_column:
        parts.append(out[desc_column].fillna("").map(_clean_text))
    if not parts:
        raise ValueError("at least one of name_column/desc_column is required")
    text = parts[0]
    for p in parts[1:]:
        text = text + " " + p
    out["__text__"] = text
    return out, list(text)


class TfidfEmbedder:
    """Offline embedding model: fitted on the corpus + query batch each
    call (TF-IDF with word 1-2-grams and char 3-4-grams)."""

    def __init__(self):
        from sklearn.feature_extraction.text import TfidfVectorizer
        from sklearn.pipeline import FeatureUnion

        self._word = TfidfVectorizer(ngram_range=(1, 2), sublinear_tf=True)
        self._char = TfidfVectorizer(analyzer="char_wb", ngram_range=(3, 4), sublinear_tf=True)

    def fit(self, texts: List[str]):
        self._word.fit(texts)
        self._char.fit(texts)
        return self

    def encode(self, texts: List[str]) -> np.ndarray:
        import scipy.sparse as sp

        w = self._word.transform(texts)
        c = self._char.transform(texts)
        m = sp.hstack([w, c]).toarray()
        n = np.linalg.norm(m, axis=1, keepdims=True)
        return m / np.maximum(n, 1e-12)


class TransformersEmbedder:
    """Optional LOCAL semantic-model path: mean-pooled hidden states from
    a transformers checkpoint on disk (e.g. a pre-downloaded
    all-mpnet-base-v2). Used automatically when `detect_model_path()`
    exists (or ANOVOS_FR_MODEL_PATH points at a model dir); there is no
    egress in this stack so nothing is ever downloaded. TF-IDF remains
    the default offline backend."""

    def __init__(self, model_path: str):
        from transformers import AutoModel, AutoTokenizer  # local files only

        self._tok = AutoTokenizer.from_pretrained(model_path, local_files_only=True)
        self._model = AutoModel.from_pretrained(model_path, local_files_only=True)
        self._model.eval()

    def fit(self, texts):
        return self

    def encode(self, texts: List[str]) -> np.ndarray:
        import torch as _t

        outs = []
        with _t.no_grad():
            for i in range(0, len(texts), 64):
                batch = self._tok(texts[i : i + 64], padding=True, truncation=True,
                                  max_length=128, return_tensors="pt")
                h = self._model(**batch).last_hidden_state
                mask = batch["attention_mask"].unsqueeze(-1).float()
                emb = (h * mask).sum(1) / mask.sum(1).clamp(min=1e-9)
                outs.append(emb.cpu().numpy())
        m = np.concatenate(outs)
        n = np.linalg.norm(m, axis=1, keepdims=True)
        return m / np.maximum(n, 1e-12)


def _local_model_path() -> Optional[str]:
    p = os.getenv("ANOVOS_FR_MODEL_PATH") or detect_model_path()
    return p if p and os.path.isdir(p) else None


def get_embedder(corpus_texts: List[str], query_texts: List[str]):
    """Semantic model when a local checkpoint exists, else TF-IDF."""
    mp = _local_model_path()
    if mp:
        try:
            return TransformersEmbedder(mp)
        except Exception:
            pass
    return TfidfEmbedder().fit(list(corpus_texts) + list(query_texts))


def semantic_search(query_texts: List[str], corpus_texts: List[str], top_k: int = 5):
    """util.semantic_search equivalent: per query, top_k (idx, score)."""
    emb = get_embedder(corpus_texts, query_texts)
    q = emb.encode(list(query_texts))
    c = emb.encode(list(corpus_texts))
    sims = q @ c.T
    out = []
    for row in sims:
        idx = np.argsort(-row)[:top_k]
        out.append([{"corpus_id": int(i), "score": float(row[i])} for i in idx])
    return out


def feature_exploration_prep() -> pd.DataFrame:
    """Reference featrec_init.py:feature_exploration_prep."""
    return init_input_fer()


def feature_recommendation_prep():
    """Reference featrec_init.py:feature_recommendation_prep — corpus +
    prepared text for matching."""
    if "prep" in _STATE:
        return _STATE["prep"]
    df = init_input_fer()
    name_c, desc_c, ind_c, use_c = get_column_name(df)
    prepared, texts = recommendation_data_prep(df, name_c, desc_c)
    _STATE["prep"] = (prepared, texts)
    return _STATE["prep"]
