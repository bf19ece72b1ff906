#!/usr/bin/env python3
"""Geo auto-detection (reference notebook
data_ingest__geo_auto_detection.ipynb): lat/long/geohash column
screening + geohash codecs."""

import numpy as np
import pandas as pd

from _common import AnovosFrame, init_context

from anovos_amd.data_ingest.geo_auto_detection import geo_to_latlong, latlong_to_geo, ll_gh_cols

rng = np.random.default_rng(9)
n = 2000
lat = rng.uniform(25, 49, n)
lon = rng.uniform(-124, -67, n)
pdf = pd.DataFrame({
    "id": np.arange(n, dtype=float),
    "latitude": lat,
    "longitude": lon,
    "gh7": latlong_to_geo(lat, lon, precision=7),
    "noise": rng.normal(1000, 5, n),
})
ctx = init_context()
idf = AnovosFrame.from_pandas(pdf, device=getattr(ctx, "device", "cpu"))
print("detected:", ll_gh_cols(idf))
print("decode gh7[0]:", geo_to_latlong(pdf["gh7"].iloc[0], 0), geo_to_latlong(pdf["gh7"].iloc[0], 1))
