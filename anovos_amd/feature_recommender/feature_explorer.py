"""Explore the feature corpus by industry / usecase (reference parity:
``anovos/feature_recommender/feature_explorer.py`` :61-310 — same
function names and output shapes; semantic matching backed by the
offline TF-IDF embedder in featrec_init)."""

from __future__ import annotations

from typing import Tuple

import pandas as pd

from anovos_amd.feature_recommender.featrec_init import (
    get_column_name,
    init_input_fer,
    semantic_search,
)


def list_all_industry() -> pd.DataFrame:
    """Reference feature_explorer.py — distinct industries."""
    df = init_input_fer()
    _, _, ind_c, _ = get_column_name(df)
    return pd.DataFrame({"Industry": sorted(df[ind_c].dropna().unique())})


def list_all_usecase() -> pd.DataFrame:
    df = init_input_fer()
    _, _, _, use_c = get_column_name(df)
    return pd.DataFrame({"Usecase": sorted(df[use_c].dropna().unique())})


def list_all_pair() -> pd.DataFrame:
    df = init_input_fer()
    _, _, ind_c, use_c = get_column_name(df)
    pairs = df[[ind_c, use_c]].drop_duplicates().sort_values([ind_c, use_c]).reset_index(drop=True)
    pairs.columns = ["Industry", "Usecase"]
    return pairs


def _lcs_len(a: str, b: str) -> int:
    """Longest common subsequence length (both strings are short)."""
    dp = [0] * (len(b) + 1)
    for ch in a:
        prev = 0
        for j, cb in enumerate(b, 1):
            cur = dp[j]
            dp[j] = prev + 1 if ch == cb else max(dp[j], dp[j - 1])
            prev = cur
    return dp[len(b)]


def _best_match(value: str, options) -> Tuple[str, float]:
    """Nearest option by TF-IDF cosine PLUS a subsequence bonus: the
    reference's sentence model snaps abbreviations ('telco' ->
    'telecommunication', 'bank' -> 'banking ...') that character
    n-grams alone under-score; LCS(query, option)/len(query) recovers
    exactly that class of match while nonsense strings stay low."""
    value = str(value).strip().lower()
    options = list(options)
    hits = semantic_search([value], options, top_k=len(options))[0]
    if not hits:
        return value, 0.0
    best, best_score = value, 0.0
    for h in hits:
        opt = options[h["corpus_id"]]
        bonus = 0.3 * (_lcs_len(value, opt) / max(len(value), 1)) if len(value) >= 4 else 0.0
        s = h["score"] + bonus
        if s > best_score:
            best, best_score = opt, s
    return best, best_score


def process_usecase(usecase: str, semantic: bool = True) -> str:
    """Reference feature_explorer.py:61 — normalize a usecase string,
    optionally snapping to the nearest corpus usecase."""
    if not isinstance(usecase, str):
        raise TypeError("Invalid input for usecase")
    usecase = usecase.strip().lower()
    if not semantic:
        return usecase
    options = list(list_all_usecase()["Usecase"])
    if usecase in options:
        return usecase
    best, score = _best_match(usecase, options)
    if score >= 0.15:
        print(f"Matching usecase '{usecase}' -> '{best}'")
        return best
    return usecase


def process_industry(industry: str, semantic: bool = True) -> str:
    if not isinstance(industry, str):
        raise TypeError("Invalid input for industry")
    industry = industry.strip().lower()
    if not semantic:
        return industry
    options = list(list_all_industry()["Industry"])
    if industry in options:
        return industry
    best, score = _best_match(industry, options)
    if score >= 0.15:
        print(f"Matching industry '{industry}' -> '{best}'")
        return best
    return industry


def list_usecase_by_industry(industry: str, semantic: bool = True) -> pd.DataFrame:
    df = init_input_fer()
    _, _, ind_c, use_c = get_column_name(df)
    industry = process_industry(industry, semantic)
    out = df[df[ind_c] == industry][[use_c]].drop_duplicates().reset_index(drop=True)
    out.columns = ["Usecase"]
    return out


def list_industry_by_usecase(usecase: str, semantic: bool = True) -> pd.DataFrame:
    df = init_input_fer()
    _, _, ind_c, use_c = get_column_name(df)
    usecase = process_usecase(usecase, semantic)
    out = df[df[use_c] == usecase][[ind_c]].drop_duplicates().reset_index(drop=True)
    out.columns = ["Industry"]
    return out


def list_feature_by_industry(industry: str, num_of_feat: int = 100, semantic: bool = True) -> pd.DataFrame:
    """Reference feature_explorer.py:181."""
    df = init_input_fer()
    name_c, desc_c, ind_c, use_c = get_column_name(df)
    industry = process_industry(industry, semantic)
    out = df[df[ind_c] == industry].head(num_of_feat).reset_index(drop=True)
    return out


def list_feature_by_usecase(usecase: str, num_of_feat: int = 100, semantic: bool = True) -> pd.DataFrame:
    df = init_input_fer()
    name_c, desc_c, ind_c, use_c = get_column_name(df)
    usecase = process_usecase(usecase, semantic)
    return df[df[use_c] == usecase].head(num_of_feat).reset_index(drop=True)


def list_feature_by_pair(industry: str, usecase: str, num_of_feat: int = 100, semantic: bool = True) -> pd.DataFrame:
    df = init_input_fer()
    name_c, desc_c, ind_c, use_c = get_column_name(df)
    industry = process_industry(industry, semantic)
    usecase = process_usecase(usecase, semantic)
    return df[(df[ind_c] == industry) & (df[use_c] == usecase)].head(num_of_feat).reset_index(drop=True)
