"""Geospatial analyzer — stats, clustering and location charts for the
report (reference parity: ``anovos/data_analyzer/geospatial_analyzer.py``
:64-1234, same output-file contract: CSV stats tables plus plotly JSON
charts under ``master_path``).

MI355X-native: descriptive stats and top-pair tables come from on-device
scatter/unique reductions; K-Means runs as a torch Lloyd loop on the GPU
(cdist + argmin — SURVEY §2.10 K19); DBSCAN fits on a bounded driver
sample via sklearn exactly like the reference did (its clustering was
always driver-side pandas, geospatial_analyzer.py:463-470).
"""

from __future__ import annotations

import os

import numpy as np
import pandas as pd
import torch

import plotly.express as px
import plotly.graph_objects as go

from anovos_amd.core.frame import AnovosFrame
from anovos_amd.data_transformer import geo_utils as gu
from anovos_amd.shared.utils import ends_with

global_theme = px.colors.sequential.Peach
global_plot_bg_color = "rgba(0,0,0,0)"
global_paper_bg_color = "rgba(0,0,0,0)"

_MAPBOX_STYLES = ["open-street-map", "carto-positron", "carto-darkmatter", "stamen-terrain", "stamen-toner", "stamen-watercolor"]



def _rank0_write(fn):
    """Only rank 0 persists report artifacts (stat computation above is
    collective on all ranks; duplicate concurrent writes can tear files)."""
    from anovos_amd.core import dist as _d

    if _d.rank() == 0:
        fn()


def _num(df: AnovosFrame, col: str) -> torch.Tensor:
    from anovos_amd.data_transformer.geospatial import _num as _n

    return _n(df, col)


def _mapbox_style(global_map_box_val: int) -> str:
    try:
        return _MAPBOX_STYLES[int(global_map_box_val) % len(_MAPBOX_STYLES)]
    except Exception:
        return "open-street-map"


# ------------------------------------------------------------- stats gen

def descriptive_stats_gen(df, lat_col, long_col, geohash_col, id_col, master_path, max_val):
    """Reference geospatial_analyzer.py:64 — overall summary + top
    lat-long / geohash tables, written as CSVs under master_path."""
    os.makedirs(master_path, exist_ok=True)
    if lat_col is not None:
        lat, lon = _num(df, lat_col), _num(df, long_col)
        ok = ~torch.isnan(lat) & ~torch.isnan(lon)
        la, lo = lat[ok], lon[ok]
        # top pairs by multiplicity: unique over packed pair; multi-rank
        # merges each rank's top-4x candidates (bounded traffic — exact
        # unless a pair's global count is split below every rank's local
        # top-4x cut, immaterial for a leaderboard table)
        from anovos_amd.core import dist as _dist

        pair = torch.stack([la, lo], dim=1)
        uniq, counts = torch.unique(pair, dim=0, return_counts=True)
        if _dist.is_dist():
            k = int(max_val) * 4
            local_order = torch.argsort(counts, descending=True)[:k]
            av = torch.cat(_dist.all_gather_tensor(uniq[local_order]))
            ac = torch.cat(_dist.all_gather_tensor(counts[local_order]))
            uniq, ginv = torch.unique(av, dim=0, return_inverse=True)
            counts = torch.zeros(uniq.shape[0], dtype=torch.int64, device=av.device)
            counts.index_add_(0, ginv, ac)
        order = torch.argsort(counts, descending=True)
        topn = order[: int(max_val)]
        top_pairs = pd.DataFrame(
            {
                lat_col: uniq[topn, 0].cpu().numpy(),
                long_col: uniq[topn, 1].cpu().numpy(),
                "count": counts[topn].cpu().numpy(),
            }
        )
        from anovos_amd.core import dist as _dist2

        ids = df.count() if id_col is None or id_col not in df.columns else int(_dist2.all_reduce_scalar(int((~df.col(id_col).null_mask()).sum())))
        most = top_pairs.iloc[0] if len(top_pairs) else None
        gen_stats = pd.DataFrame(
            {
                "Stats": [
                    "Total Number of Records",
                    "Distinct {Lat, Long} Pairs",
                    "Most Common {Lat, Long} Pair",
                    "Most Common {Lat, Long} Pair Occurence",
                ],
                "Count": [
                    int(ids),
                    int(uniq.shape[0]),
                    f"{{{float(most[lat_col]):.6f}, {float(most[long_col]):.6f}}}" if most is not None else "NA",
                    int(most["count"]) if most is not None else 0,
                ],
            }
        )
        names = ["Overall_Summary", "Top_" + str(max_val) + "_Lat_Long"]
        for nm, tbl in zip(names, [gen_stats, top_pairs]):
            _rank0_write(lambda: tbl.to_csv(ends_with(master_path) + nm + "_1_" + lat_col + "_" + long_col + ".csv", index=False))
    if geohash_col is not None:
        from anovos_amd.core import dist as _dist3

        c = df.col(geohash_col)
        valid = ~c.null_mask()
        codes = c.data[valid].to(torch.long)
        cnt = torch.bincount(codes, minlength=len(c.dictionary or []))
        _dist3.all_reduce_(cnt, "sum")  # dictionaries are rank-unified at ingest
        order = torch.argsort(cnt, descending=True)
        dist_geohash = int((cnt > 0).sum())
        prec = max((len(s) for s in (c.dictionary or [])), default=0)
        top_idx = order[: int(max_val)].cpu().numpy()
        top_gh = pd.DataFrame(
            {
                geohash_col: [c.dictionary[i] for i in top_idx if int(cnt[i]) > 0],
                "count_records": [int(cnt[i]) for i in top_idx if int(cnt[i]) > 0],
            }
        )
        gen_stats = pd.DataFrame(
            {
                "Stats": [
                    "Total Number of Records",
                    "Distinct Geohashes",
                    "Geohash Precision",
                    "The Most Common Geohash",
                ],
                "Count": [
                    int(_dist3.all_reduce_scalar(int(valid.sum()))),
                    dist_geohash,
                    prec,
                    top_gh[geohash_col].iloc[0] if len(top_gh) else "NA",
                ],
            }
        )
        names = ["Overall_Summary", "Top_" + str(max_val) + "_Geohash_Distribution"]
        _rank0_write(lambda: gen_stats.to_csv(ends_with(master_path) + names[0] + "_2_" + geohash_col + ".csv", index=False))
        _rank0_write(lambda: top_gh.to_csv(ends_with(master_path) + names[1] + "_2_" + geohash_col + ".csv", index=False))


def lat_long_col_stats_gen(df, lat_col, long_col, id_col, master_path, max_val):
    """Reference geospatial_analyzer.py:235."""
    for la, lo in zip(list(lat_col), list(long_col)):
        descriptive_stats_gen(df, la, lo, None, id_col, master_path, max_val)


def geohash_col_stats_gen(df, geohash_col, id_col, master_path, max_val):
    """Reference geospatial_analyzer.py:275."""
    for gh in list(geohash_col):
        descriptive_stats_gen(df, None, None, gh, id_col, master_path, max_val)


def stats_gen_lat_long_geo(df, lat_col, long_col, geohash_col, id_col, master_path, max_val):
    """Reference geospatial_analyzer.py:313."""
    if lat_col:
        lat_long_col_stats_gen(df, lat_col, long_col, id_col, master_path, max_val)
    if geohash_col:
        geohash_col_stats_gen(df, geohash_col, id_col, master_path, max_val)


# ------------------------------------------------------------- clustering

def _kmeans_torch(x: torch.Tensor, k: int, iters: int = 50, seed: int = 0):
    """Lloyd K-Means on device: k-means++ seeding + cdist/argmin loop.
    Returns (labels, inertia). x: [N,2] float64."""
    n = x.shape[0]
    if n == 0:
        return torch.zeros(0, dtype=torch.long, device=x.device), 0.0
    g = torch.Generator(device="cpu").manual_seed(seed)
    centers = x[torch.randint(0, n, (1,), generator=g).item()].unsqueeze(0)
    for _ in range(1, k):
        d2 = torch.cdist(x, centers).min(dim=1).values ** 2
        probs = (d2 / d2.sum().clamp(min=1e-30)).cpu()
        nxt = int(torch.multinomial(probs, 1, generator=g))
        centers = torch.cat([centers, x[nxt].unsqueeze(0)], dim=0)
    labels = torch.zeros(n, dtype=torch.long, device=x.device)
    for _ in range(iters):
        d = torch.cdist(x, centers)
        new_labels = d.argmin(dim=1)
        if bool((new_labels == labels).all()):
            labels = new_labels
            break
        labels = new_labels
        for j in range(k):
            m = labels == j
            if bool(m.any()):
                centers[j] = x[m].mean(dim=0)
    inertia = float((torch.cdist(x, centers).gather(1, labels.unsqueeze(1)) ** 2).sum())
    return labels, inertia


def geo_cluster_analysis(df, lat_col, long_col, max_cluster, eps, min_samples, master_path, col_name, global_map_box_val):
    """Reference geospatial_analyzer.py:390 — K-Means elbow + DBSCAN
    grid, 8 plotly JSON plots + 2 CSVs, same file names."""
    os.makedirs(master_path, exist_ok=True)
    if isinstance(df, AnovosFrame):
        lat = _num(df, lat_col)
        lon = _num(df, long_col)
        ok = ~torch.isnan(lat) & ~torch.isnan(lon)
        x = torch.stack([lat[ok], lon[ok]], dim=1)
        pdf = pd.DataFrame({lat_col: x[:, 0].cpu().numpy(), long_col: x[:, 1].cpu().numpy()})
    else:
        pdf = df[[lat_col, long_col]].dropna().reset_index(drop=True)
        x = torch.tensor(pdf.to_numpy(), dtype=torch.float64)

    style = _mapbox_style(global_map_box_val)
    max_k = max(int(max_cluster), 4)
    distortions = []
    for i in range(2, max_k + 1):
        if x.shape[0] >= i:
            _, inertia = _kmeans_torch(x, i, seed=0)
            distortions.append(inertia)
    if len(distortions) >= 3:
        dd2 = np.diff(distortions, 2)
        k = int(np.argmin(dd2)) + 2
    else:
        k = 2
    f1 = go.Figure(go.Scatter(x=list(range(1, len(distortions) + 1)), y=distortions, mode="lines+markers",
                              line=dict(color=global_theme[2], width=2, dash="dash"), marker=dict(size=10)))
    f1.update_yaxes(title="Distortion")
    f1.update_xaxes(title="Values of K")
    f1.add_vline(x=k, line_width=3, line_dash="dash")
    f1.update_layout(title_text=f"Elbow Curve Showing the Optimal Number of Clusters [K : {k}] <br><sup>Algorithm Used : KMeans</sup>")
    f1.layout.plot_bgcolor = global_plot_bg_color
    f1.layout.paper_bgcolor = global_paper_bg_color
    _rank0_write(lambda: f1.write_json(ends_with(master_path) + "cluster_plot_1_elbow_" + col_name))

    labels, _ = _kmeans_torch(x, max(k, 2), seed=0)
    pdf["cluster"] = labels.cpu().numpy()
    _rank0_write(lambda: pdf.to_csv(ends_with(master_path) + "cluster_output_kmeans_" + col_name + ".csv", index=False))

    cluster_dtls = pdf.groupby("cluster").size().reset_index(name="counts")
    f2 = go.Figure(go.Pie(labels=cluster_dtls["cluster"], values=cluster_dtls["counts"], hole=0.3,
                          marker_colors=px.colors.sequential.Peach))
    f2.update_layout(title_text="Distribution of Clusters <br><sup>Algorithm Used : KMeans</sup>")
    _rank0_write(lambda: f2.write_json(ends_with(master_path) + "cluster_plot_2_kmeans_" + col_name))

    sample = pdf.sample(min(len(pdf), 10000), random_state=0) if len(pdf) else pdf
    f3 = px.scatter_mapbox(sample, lat=lat_col, lon=long_col, color=sample["cluster"].astype(str),
                           color_discrete_sequence=px.colors.qualitative.Safe, zoom=1)
    f3.update_layout(mapbox_style=style, title_text="Cluster-wise Geospatial Datapoints <br><sup>Algorithm Used : KMeans</sup>")
    _rank0_write(lambda: f3.write_json(ends_with(master_path) + "cluster_plot_3_kmeans_" + col_name))

    # ---- DBSCAN on a bounded driver sample (reference drove sklearn on pandas)
    from sklearn.cluster import DBSCAN
    from sklearn.metrics import silhouette_score

    samp = pdf[[lat_col, long_col]].sample(min(len(pdf), 20000), random_state=0).to_numpy()
    eps = list(eps) if isinstance(eps, (list, tuple)) else [float(eps), float(eps) + 0.1, 0.05]
    min_samples = list(min_samples) if isinstance(min_samples, (list, tuple)) else [int(min_samples), int(min_samples) + 10, 5]
    eps_grid = np.arange(eps[0], eps[1], eps[2] if len(eps) > 2 else 0.05)
    ms_grid = np.arange(int(min_samples[0]), int(min_samples[1]), int(min_samples[2]) if len(min_samples) > 2 else 5)
    sil = np.full((len(eps_grid), len(ms_grid)), np.nan)
    best = (None, -2.0)
    for a, e in enumerate(eps_grid):
        for b, ms in enumerate(ms_grid):
            lab = DBSCAN(eps=float(e), min_samples=int(ms)).fit_predict(samp)
            if len(set(lab)) > 1 and len(set(lab)) < len(samp):
                try:
                    s = silhouette_score(samp, lab)
                except ValueError:
                    continue
                sil[a, b] = s
                if s > best[1]:
                    best = ((float(e), int(ms)), s)
    f1_ = go.Figure(go.Heatmap(z=sil, x=[str(m) for m in ms_grid], y=[f"{e:.2f}" for e in eps_grid], colorscale="Peach"))
    f1_.update_layout(title_text="Silhouette Scores across (eps x min_samples) <br><sup>Algorithm Used : DBSCAN</sup>",
                      xaxis_title="min_samples", yaxis_title="eps")
    _rank0_write(lambda: f1_.write_json(ends_with(master_path) + "cluster_plot_1_silhoutte_" + col_name))

    e_best, ms_best = best[0] if best[0] else (float(eps_grid[0]), int(ms_grid[0]))
    db_lab = DBSCAN(eps=e_best, min_samples=ms_best).fit_predict(samp)
    db = pd.DataFrame({lat_col: samp[:, 0], long_col: samp[:, 1], "cluster": db_lab})
    _rank0_write(lambda: db.to_csv(ends_with(master_path) + "cluster_output_dbscan_" + col_name + ".csv", index=False))

    db_dtls = db.groupby("cluster").size().reset_index(name="counts")
    f2_ = go.Figure(go.Pie(labels=db_dtls["cluster"], values=db_dtls["counts"], hole=0.3,
                           marker_colors=px.colors.sequential.Peach))
    f2_.update_layout(title_text="Distribution of Clusters <br><sup>Algorithm Used : DBSCAN</sup>")
    _rank0_write(lambda: f2_.write_json(ends_with(master_path) + "cluster_plot_2_dbscan_" + col_name))

    f3_ = px.scatter_mapbox(db[db["cluster"] >= 0], lat=lat_col, lon=long_col,
                            color=db[db["cluster"] >= 0]["cluster"].astype(str),
                            color_discrete_sequence=px.colors.qualitative.Safe, zoom=1)
    f3_.update_layout(mapbox_style=style, title_text="Cluster-wise Geospatial Datapoints <br><sup>Algorithm Used : DBSCAN</sup>")
    _rank0_write(lambda: f3_.write_json(ends_with(master_path) + "cluster_plot_3_dbscan_" + col_name))

    outliers = db[db["cluster"] == -1]
    f4 = go.Figure(go.Scatter(x=outliers[long_col], y=outliers[lat_col], mode="markers",
                              marker_symbol="x", marker_color=global_theme[4]))
    f4.update_layout(title_text="Outlier Points <br><sup>Algorithm Used : DBSCAN (Euclidean)</sup>",
                     xaxis_title=long_col, yaxis_title=lat_col)
    f4.layout.plot_bgcolor = global_plot_bg_color
    f4.layout.paper_bgcolor = global_paper_bg_color
    _rank0_write(lambda: f4.write_json(ends_with(master_path) + "cluster_plot_4_dbscan_1_" + col_name))

    # haversine-metric DBSCAN outliers
    db_lab_h = DBSCAN(eps=e_best / 60.0, min_samples=ms_best, metric="haversine").fit_predict(np.radians(samp))
    dbh = pd.DataFrame({lat_col: samp[:, 0], long_col: samp[:, 1], "cluster": db_lab_h})
    outliers_h = dbh[dbh["cluster"] == -1]
    f4_ = go.Figure(go.Scatter(x=outliers_h[long_col], y=outliers_h[lat_col], mode="markers",
                               marker_symbol="x", marker_color=global_theme[4]))
    f4_.update_layout(title_text="Outlier Points <br><sup>Algorithm Used : DBSCAN (Haversine)</sup>",
                      xaxis_title=long_col, yaxis_title=lat_col)
    _rank0_write(lambda: f4_.write_json(ends_with(master_path) + "cluster_plot_4_dbscan_2_" + col_name))


def _gathered_pair_pdf(df: AnovosFrame, lat_col: str, long_col: str, max_records: int) -> pd.DataFrame:
    """Bounded global (lat, long) sample as pandas: nan-filter + local
    subsample, one varlen all-gather (≤ max_records × world_size rows),
    then a deterministic re-subsample so every rank holds the SAME
    ≤ max_records dataset-wide sample. Clustering itself is collective-
    free and runs on rank 0 only."""
    from anovos_amd.core import dist as _dist

    lat, lon = _num(df, lat_col), _num(df, long_col)
    ok = ~torch.isnan(lat) & ~torch.isnan(lon)
    x = torch.stack([lat[ok], lon[ok]], dim=1)

    def _sub(t):
        if t.shape[0] > int(max_records):
            g = torch.Generator(device="cpu").manual_seed(7)
            idx = torch.randperm(t.shape[0], generator=g)[: int(max_records)].to(t.device)
            return t[idx]
        return t

    x = _sub(x)
    if _dist.world_size() > 1:
        x = _sub(torch.cat(_dist.all_gather_tensor(x)))
    return pd.DataFrame({lat_col: x[:, 0].cpu().numpy(), long_col: x[:, 1].cpu().numpy()})


def geo_cluster_generator(df, lat_col_list, long_col_list, geo_col_list, max_cluster, eps, min_samples, master_path, global_map_box_val, max_records):
    """Reference geospatial_analyzer.py:734 — cluster analysis for every
    lat-long pair and every geohash column (decoded first). The sample
    is gathered dataset-wide first; the clustering/plot pass (no
    collectives, all writes) runs on rank 0 only."""
    from anovos_amd.core import dist as _dist

    def _run(sub, la, lo, tag):
        if isinstance(sub, AnovosFrame):
            sub = _gathered_pair_pdf(sub, la, lo, int(max_records))
        if _dist.rank() == 0 and len(sub) >= 4:
            geo_cluster_analysis(sub, la, lo, max_cluster, eps, min_samples, master_path, tag, global_map_box_val)
        _dist.barrier()

    if lat_col_list:
        for la, lo in zip(list(lat_col_list), list(long_col_list)):
            _run(df, la, lo, la + "_" + lo)
    if geo_col_list:
        from anovos_amd.data_transformer.geospatial import geo_format_geohash

        for ghc in list(geo_col_list):
            dec = geo_format_geohash(df, [ghc], "dd", output_mode="append") if isinstance(df, AnovosFrame) else df
            _run(dec, f"{ghc}_lat_dd", f"{ghc}_lon_dd", ghc)


# ------------------------------------------------------------- loc charts

def generate_loc_charts_processor(df, lat_col, long_col, geohash_col, max_val, id_col, global_map_box_val, master_path):
    """Reference geospatial_analyzer.py:851 — mapbox scatter of (sampled)
    locations, one JSON per analyzed column."""
    os.makedirs(master_path, exist_ok=True)
    style = _mapbox_style(global_map_box_val)
    if lat_col is not None:
        for la, lo in zip(list(lat_col), list(long_col)):
            lat, lon = _num(df, la), _num(df, lo)
            ok = ~torch.isnan(lat) & ~torch.isnan(lon)
            n = int(ok.sum())
            take = min(n, int(max_val))
            idx = ok.nonzero(as_tuple=True)[0][torch.randperm(n)[:take]]
            pdf = pd.DataFrame({la: lat[idx].cpu().numpy(), lo: lon[idx].cpu().numpy()})
            fig = px.scatter_mapbox(pdf, lat=la, lon=lo, color_discrete_sequence=[global_theme[2]], zoom=1)
            fig.update_layout(mapbox_style=style)
            _rank0_write(lambda: fig.write_json(ends_with(master_path) + "loc_charts_ll_" + la + "_" + lo))
    if geohash_col is not None:
        from anovos_amd.data_transformer.geospatial import geo_format_geohash

        for ghc in list(geohash_col):
            dec = geo_format_geohash(df, [ghc], "dd", output_mode="append")
            lat, lon = _num(dec, f"{ghc}_lat_dd"), _num(dec, f"{ghc}_lon_dd")
            ok = ~torch.isnan(lat) & ~torch.isnan(lon)
            n = int(ok.sum())
            take = min(n, int(max_val))
            idx = ok.nonzero(as_tuple=True)[0][torch.randperm(n)[:take]]
            pdf = pd.DataFrame({"lat": lat[idx].cpu().numpy(), "lon": lon[idx].cpu().numpy()})
            fig = px.scatter_mapbox(pdf, lat="lat", lon="lon", color_discrete_sequence=[global_theme[2]], zoom=1)
            fig.update_layout(mapbox_style=style)
            _rank0_write(lambda: fig.write_json(ends_with(master_path) + "loc_charts_gh_" + ghc))


def generate_loc_charts_controller(df, id_col, lat_col, long_col, geohash_col, max_val, global_map_box_val, master_path):
    """Reference geospatial_analyzer.py:1029."""
    if lat_col:
        generate_loc_charts_processor(df, lat_col, long_col, None, max_val, id_col, global_map_box_val, master_path)
    if geohash_col:
        generate_loc_charts_processor(df, None, None, geohash_col, max_val, id_col, global_map_box_val, master_path)


# ------------------------------------------------------------- orchestrator

def geospatial_autodetection(df, id_col, master_path, max_records, top_geo_records, max_cluster, eps, min_samples, global_map_box_val, run_type="local", auth_key=None):
    """Reference geospatial_analyzer.py:1119 — detect lat/long/geohash
    columns, then stats + clusters + charts; returns the detected lists."""
    from anovos_amd.data_ingest.geo_auto_detection import ll_gh_cols

    os.makedirs(master_path, exist_ok=True)
    lat_cols, long_cols, gh_cols = ll_gh_cols(df, max_records)
    if not lat_cols and not gh_cols:
        return [], [], []
    stats_gen_lat_long_geo(df, lat_cols, long_cols, gh_cols, id_col, master_path, top_geo_records)
    geo_cluster_generator(df, lat_cols, long_cols, gh_cols, max_cluster, eps, min_samples, master_path, global_map_box_val, max_records)
    generate_loc_charts_controller(df, id_col, lat_cols, long_cols, gh_cols, max_records, global_map_box_val, master_path)
    return lat_cols, long_cols, gh_cols
