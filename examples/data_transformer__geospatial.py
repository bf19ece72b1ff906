#!/usr/bin/env python3
"""Geospatial transforms (reference notebook
data_transformer__geospatial.ipynb): format conversions, distances,
centroids, country membership, radius of gyration."""

import numpy as np
import pandas as pd

from _common import AnovosFrame, init_context

from anovos_amd.data_transformer import geospatial as geo

rng = np.random.default_rng(13)
n = 1000
pdf = pd.DataFrame({
    "id": rng.integers(0, 50, n).astype(float),
    "lat": rng.uniform(35, 60, n), "lon": rng.uniform(-10, 25, n),
    "lat2": rng.uniform(35, 60, n), "lon2": rng.uniform(-10, 25, n),
})
ctx = init_context()
idf = AnovosFrame.from_pandas(pdf, device=getattr(ctx, "device", "cpu"))
odf = geo.geo_format_latlon(idf, ["lat"], ["lon"], "dd", "geohash", result_prefix=["g"])
print("geohash sample:", odf.col("g_geohash").dictionary[:3])
d = geo.location_distance(idf, ["lat", "lon"], ["lat2", "lon2"], "dd",
                          result_prefix="pair", distance_type="haversine", unit="km")
print("mean pair distance km:", float(d.col("pair_distance").data.mean()))
print(geo.centroid(idf, "lat", "lon").to_pandas().to_string(index=False))
rog = geo.rog_calculation(idf, "lat", "lon", id_col="id").to_pandas()
print("rog head:\n", rog.head(3).to_string(index=False))
uk = geo.location_in_country(ctx, idf, ["lat"], ["lon"], "france", result_prefix=["fr"])
print("rows in France bbox:", int(uk.col("fr_in_france").data.sum()))
