"""Map user attributes to corpus features by semantic similarity
(reference parity: ``anovos/feature_recommender/feature_mapper.py``
:35-560 — feature_mapper, find_attr_by_relevance, sankey_visualization).

Column contract mirrors the reference exactly:
- feature_mapper → [Input_Attribute_Name, Input_Attribute_Description,
  Matched_Feature_Name, Matched_Feature_Description,
  Feature_Similarity_Score, Industry, Usecase] (input columns present
  only when name_column/desc_column were given); below-threshold
  matches are filled with "N/A" (reference feature_mapper.py:210-214).
- find_attr_by_relevance → [Input_Feature_Description,
  Recommended_Input_Attribute_Name,
  Recommended_Input_Attribute_Description,
  Input_Attribute_Similarity_Score].

Embeddings come from the local semantic model when one is installed,
else the offline TF-IDF embedder (featrec_init.get_embedder)."""

from __future__ import annotations

from typing import List, Optional

import numpy as np
import pandas as pd
import plotly.graph_objects as go

from anovos_amd.feature_recommender.featrec_init import (
    get_column_name,
    get_embedder,
    init_input_fer,
    recommendation_data_prep,
)
from anovos_amd.feature_recommender.feature_explorer import process_industry, process_usecase


def _mapper_columns(name_column, desc_column):
    cols = []
    if name_column is not None:
        cols.append("Input_Attribute_Name")
    if desc_column is not None:
        cols.append("Input_Attribute_Description")
    return cols + [
        "Matched_Feature_Name",
        "Matched_Feature_Description",
        "Feature_Similarity_Score",
        "Industry",
        "Usecase",
    ]


def feature_mapper(df: pd.DataFrame, name_column: Optional[str] = None, desc_column: Optional[str] = None,
                   suggested_industry: str = "all", suggested_usecase: str = "all",
                   semantic: bool = True, top_n: int = 2, threshold: float = 0.3) -> pd.DataFrame:
    """Reference feature_mapper.py:35 — for each input attribute, the
    top_n most similar corpus features (with industry/usecase), scored
    by cosine similarity; below-threshold matches become 'N/A'."""
    if not isinstance(df, pd.DataFrame):
        raise TypeError("Invalid input for df")
    if top_n < 1:
        raise TypeError("Invalid input for top_n")
    corpus = init_input_fer()
    c_name, c_desc, c_ind, c_use = get_column_name(corpus)
    if suggested_industry != "all":
        corpus = corpus[corpus[c_ind] == process_industry(suggested_industry, semantic)]
    if suggested_usecase != "all":
        corpus = corpus[corpus[c_use] == process_usecase(suggested_usecase, semantic)]
    corpus = corpus.reset_index(drop=True)
    out_cols = _mapper_columns(name_column, desc_column)
    if len(corpus) == 0:
        print("Industry/Usecase pair does not exist.")
        return pd.DataFrame(columns=out_cols)
    corpus_prep, corpus_texts = recommendation_data_prep(corpus, c_name, c_desc)
    attr_prep, attr_texts = recommendation_data_prep(df, name_column, desc_column)

    emb = get_embedder(corpus_texts, attr_texts)
    A = emb.encode(attr_texts)
    C = emb.encode(corpus_texts)
    sims = A @ C.T

    rows = []
    for i in range(len(attr_prep)):
        order = np.argsort(-sims[i])[:top_n]
        for j in order:
            score = float(sims[i][j])
            prefix = []
            if name_column is not None:
                prefix.append(attr_prep[name_column].iloc[i])
            if desc_column is not None:
                prefix.append(attr_prep[desc_column].iloc[i])
            if score >= threshold:
                rows.append(prefix + [
                    corpus[c_name].iloc[j], corpus[c_desc].iloc[j],
                    round(score, 4),
                    corpus[c_ind].iloc[j], corpus[c_use].iloc[j],
                ])
            else:
                rows.append(prefix + ["N/A", "N/A", "N/A", "N/A", "N/A"])
    return pd.DataFrame(rows, columns=out_cols)


def find_attr_by_relevance(df: pd.DataFrame, building_corpus: List[str],
                           name_column: Optional[str] = None, desc_column: Optional[str] = None,
                           threshold: float = 0.3) -> pd.DataFrame:
    """Reference feature_mapper.py:322 — for each goal text, the input
    attributes ranked by relevance (≥ threshold)."""
    if not isinstance(building_corpus, list):
        raise TypeError("Invalid input for building_corpus")
    cols = ["Input_Feature_Description"]
    if name_column is not None:
        cols.append("Recommended_Input_Attribute_Name")
    if desc_column is not None:
        cols.append("Recommended_Input_Attribute_Description")
    cols.append("Input_Attribute_Similarity_Score")
    attr_prep, attr_texts = recommendation_data_prep(df, name_column, desc_column)
    goals = [str(g).strip().lower() for g in building_corpus]
    emb = get_embedder(attr_texts, goals)
    G = emb.encode(goals)
    A = emb.encode(attr_texts)
    sims = G @ A.T
    rows = []
    for gi, g in enumerate(building_corpus):
        hits = np.argsort(-sims[gi])
        any_hit = False
        for ai in hits:
            score = float(sims[gi][ai])
            if score < threshold:
                break
            any_hit = True
            row = [g]
            if name_column is not None:
                row.append(attr_prep[name_column].iloc[ai])
            if desc_column is not None:
                row.append(attr_prep[desc_column].iloc[ai])
            row.append(round(score, 4))
            rows.append(row)
        if not any_hit:
            rows.append([g] + ["N/A"] * (len(cols) - 1))
    return pd.DataFrame(rows, columns=cols)


def sankey_visualization(df: pd.DataFrame, industry_included: bool = False,
                         usecase_included: bool = False) -> go.Figure:
    """Reference feature_mapper.py:465 — sankey of attribute→feature
    (→industry→usecase) mappings from feature_mapper output."""
    need = ["Input_Attribute_Name", "Matched_Feature_Name"]
    if any(c not in df.columns for c in need):
        raise TypeError("df must be a feature_mapper output")
    levels = ["Input_Attribute_Name", "Matched_Feature_Name"]
    if industry_included:
        levels.append("Industry")
    if usecase_included:
        levels.append("Usecase")
    sub = df[df["Matched_Feature_Name"] != "N/A"]
    labels: List[str] = []
    idx = {}

    def node(v):
        if v not in idx:
            idx[v] = len(labels)
            labels.append(v)
        return idx[v]

    src, dst, val = [], [], []
    for _, r in sub.iterrows():
        for a, b in zip(levels[:-1], levels[1:]):
            s = node(str(r[a]))
            d = node(str(r[b]))
            src.append(s)
            dst.append(d)
            v = r.get("Feature_Similarity_Score", 1)
            val.append(float(v) if v != "N/A" else 0.1)
    fig = go.Figure(go.Sankey(
        node=dict(pad=15, thickness=18, label=labels),
        link=dict(source=src, target=dst, value=val),
    ))
    fig.update_layout(title_text="Attribute → Feature Mapping", font_size=11)
    return fig
