"""Geospatial math — vectorized torch implementations (reference parity:
``anovos/data_transformer/geo_utils.py`` — scalar python run inside Spark
UDFs per row; here every function takes/returns torch tensors and runs as
fused elementwise GPU work).

Formats: 'dd' decimal degrees, 'dms' degrees-minutes-seconds (encoded as
d + m/100 + s/10000 triple tensors), 'radian', 'cartesian' (x,y,z on the
sphere), 'geohash' (base-32 string; bit-interleave en/decode on int64
tensors, dictionary materialized host-side only).
"""

from __future__ import annotations

import math
from typing import List, Sequence, Tuple

import numpy as np
import torch

EARTH_RADIUS = 6371009  # meters (reference geo_utils.py:10)

_UNIT_DIV = {"m": 1.0, "km": 1000.0}

_GH_BASE32 = "0123456789bcdefghjkmnpqrstuvwxyz"
_GH_DECODE = {c: i for i, c in enumerate(_GH_BASE32)}


def _as_tensor(v, device=None):
    if isinstance(v, torch.Tensor):
        return v.to(torch.float64)
    return torch.as_tensor(v, dtype=torch.float64, device=device)


def in_range(lat: torch.Tensor, lon: torch.Tensor) -> torch.Tensor:
    """True where (lat, lon) is a valid decimal-degree location
    (reference geo_utils.py:20)."""
    return (lat >= -90) & (lat <= 90) & (lon >= -180) & (lon <= 180)


# ------------------------------------------------------------ conversions

def dd_to_radian(lat, lon):
    return torch.deg2rad(_as_tensor(lat)), torch.deg2rad(_as_tensor(lon))


def radian_to_dd(lat_r, lon_r):
    return torch.rad2deg(_as_tensor(lat_r)), torch.rad2deg(_as_tensor(lon_r))


def dd_to_dms(dd: torch.Tensor):
    """Decimal degrees → (deg, min, sec) tensors (reference
    geo_utils.py:117 decimal_degrees_to_degrees_minutes_seconds)."""
    dd = _as_tensor(dd)
    sign = torch.sign(dd)
    a = dd.abs()
    d = torch.floor(a)
    m = torch.floor((a - d) * 60)
    s = (a - d - m / 60) * 3600
    return sign * d, m, s


def dms_to_dd(d: torch.Tensor, m: torch.Tensor, s: torch.Tensor) -> torch.Tensor:
    d, m, s = _as_tensor(d), _as_tensor(m), _as_tensor(s)
    sign = torch.where(d < 0, -torch.ones_like(d), torch.ones_like(d))
    return sign * (d.abs() + m / 60 + s / 3600)


def dd_to_cartesian(lat, lon, radius=EARTH_RADIUS):
    """(lat, lon) dd → (x, y, z) on the sphere (reference
    from_latlon_decimal_degrees geo_utils.py:161)."""
    la, lo = dd_to_radian(lat, lon)
    x = radius * torch.cos(la) * torch.cos(lo)
    y = radius * torch.cos(la) * torch.sin(lo)
    z = radius * torch.sin(la)
    return x, y, z


def cartesian_to_dd(x, y, z, radius=EARTH_RADIUS):
    x, y, z = _as_tensor(x), _as_tensor(y), _as_tensor(z)
    lat = torch.rad2deg(torch.asin((z / radius).clamp(-1, 1)))
    lon = torch.rad2deg(torch.atan2(y, x))
    return lat, lon


# ------------------------------------------------------------ geohash

def geohash_encode_int(lat: torch.Tensor, lon: torch.Tensor, precision: int = 8) -> torch.Tensor:
    """Bit-interleaved geohash as int64 (5·precision bits), fully on
    device. The base-32 string materializes host-side only via
    geohash_int_to_str (dictionary path)."""
    lat, lon = _as_tensor(lat), _as_tensor(lon)
    nbits = 5 * precision
    lon_bits = (nbits + 1) // 2
    lat_bits = nbits // 2
    # quantize to integer cells
    lon_q = torch.floor((lon + 180.0) / 360.0 * (1 << lon_bits)).to(torch.int64).clamp(0, (1 << lon_bits) - 1)
    lat_q = torch.floor((lat + 90.0) / 180.0 * (1 << lat_bits)).to(torch.int64).clamp(0, (1 << lat_bits) - 1)
    out = torch.zeros_like(lon_q)
    # interleave: even bit positions (from MSB) are longitude
    for i in range(nbits):
        pos = nbits - 1 - i  # bit position in output (MSB first)
        if i % 2 == 0:  # longitude bit
            src_bit = lon_bits - 1 - (i // 2)
            bit = (lon_q >> src_bit) & 1
        else:
            src_bit = lat_bits - 1 - (i // 2)
            bit = (lat_q >> src_bit) & 1
        out = out | (bit << pos)
    return out


def geohash_int_to_str(gh: torch.Tensor, precision: int = 8) -> List[str]:
    """Packed int64 → base-32 strings, vectorized (high-cardinality
    geohash dictionaries reach ~n entries; the python per-char loop was
    a hotspot in the geospatial config)."""
    v = gh.cpu().numpy().astype(np.int64)
    shifts = 5 * np.arange(precision - 1, -1, -1, dtype=np.int64)
    idx = (v[:, None] >> shifts[None, :]) & 31  # [n, precision]
    chars = np.array(list(_GH_BASE32), dtype="U1")
    mat = np.ascontiguousarray(chars[idx])  # [n, precision] U1
    return mat.view(f"U{precision}").ravel().tolist()


def geohash_str_to_int(ghs: Sequence[str]) -> Tuple[np.ndarray, int]:
    """Decode base-32 strings to packed int64, vectorized over the
    (possibly ~n-sized) dictionary. Short strings are left-aligned and
    zero-padded — identical to the scalar shift semantics. Returns
    (ints, precision)."""
    prec = max((len(g) for g in ghs if g), default=0)
    if prec == 0 or not len(ghs):
        return np.zeros(len(ghs), dtype=np.int64), prec
    arr = np.array(["" if g is None else str(g)[:prec] for g in ghs], dtype=f"U{prec}")
    cp = arr.view(np.uint32).reshape(len(arr), prec)  # UCS4 codepoints, 0-padded
    lut = np.zeros(1024, dtype=np.int64)  # unknown chars decode to 0 (as before)
    for ch, val in _GH_DECODE.items():
        lut[ord(ch)] = val
    vals = lut[np.clip(cp, 0, 1023)]  # [n, prec]; pad cp=0 -> 0
    weights = (np.int64(1) << (5 * np.arange(prec - 1, -1, -1, dtype=np.int64)))
    return (vals * weights[None, :]).sum(axis=1), prec


def geohash_decode_int(gh: torch.Tensor, precision: int) -> Tuple[torch.Tensor, torch.Tensor]:
    """int64 geohash → (lat, lon) cell centers, on device."""
    nbits = 5 * precision
    lon_bits = (nbits + 1) // 2
    lat_bits = nbits // 2
    lon_q = torch.zeros_like(gh)
    lat_q = torch.zeros_like(gh)
    for i in range(nbits):
        pos = nbits - 1 - i
        bit = (gh >> pos) & 1
        if i % 2 == 0:
            lon_q = lon_q | (bit << (lon_bits - 1 - i // 2))
        else:
            lat_q = lat_q | (bit << (lat_bits - 1 - i // 2))
    lon = (lon_q.to(torch.float64) + 0.5) / (1 << lon_bits) * 360.0 - 180.0
    lat = (lat_q.to(torch.float64) + 0.5) / (1 << lat_bits) * 180.0 - 90.0
    return lat, lon


def geohash_is_valid(s: str) -> bool:
    return isinstance(s, str) and 1 <= len(s) <= 12 and all(c in _GH_DECODE for c in s.lower())


# ------------------------------------------------------------ distances

def haversine_distance(lat1, lon1, lat2, lon2, unit="m", radius=EARTH_RADIUS) -> torch.Tensor:
    """Great-circle distance, fused elementwise (reference
    geo_utils.py:228)."""
    la1, lo1 = dd_to_radian(lat1, lon1)
    la2, lo2 = dd_to_radian(lat2, lon2)
    dla, dlo = la2 - la1, lo2 - lo1
    a = torch.sin(dla / 2) ** 2 + torch.cos(la1) * torch.cos(la2) * torch.sin(dlo / 2) ** 2
    d = 2 * radius * torch.asin(torch.sqrt(a.clamp(0, 1)))
    return d / _UNIT_DIV.get(unit, 1.0)


def vincenty_distance(lat1, lon1, lat2, lon2, unit="m", max_iter=200, tol=1e-12) -> torch.Tensor:
    """Vincenty inverse on the WGS-84 ellipsoid, vectorized with a fixed
    iteration loop (reference geo_utils.py:283 delegated to geopy)."""
    a, f = 6378137.0, 1 / 298.257223563
    b = (1 - f) * a
    la1, lo1 = dd_to_radian(lat1, lon1)
    la2, lo2 = dd_to_radian(lat2, lon2)
    U1, U2 = torch.atan((1 - f) * torch.tan(la1)), torch.atan((1 - f) * torch.tan(la2))
    L = lo2 - lo1
    lam = L.clone()
    sinU1, cosU1 = torch.sin(U1), torch.cos(U1)
    sinU2, cosU2 = torch.sin(U2), torch.cos(U2)
    for _ in range(max_iter):
        sinLam, cosLam = torch.sin(lam), torch.cos(lam)
        sin_sigma = torch.sqrt((cosU2 * sinLam) ** 2 + (cosU1 * sinU2 - sinU1 * cosU2 * cosLam) ** 2)
        cos_sigma = sinU1 * sinU2 + cosU1 * cosU2 * cosLam
        sigma = torch.atan2(sin_sigma, cos_sigma)
        sin_alpha = torch.where(sin_sigma == 0, torch.zeros_like(sin_sigma), cosU1 * cosU2 * sinLam / sin_sigma.clamp(min=1e-300))
        cos2_alpha = 1 - sin_alpha ** 2
        cos_2sigma_m = torch.where(cos2_alpha == 0, torch.zeros_like(cos_sigma), cos_sigma - 2 * sinU1 * sinU2 / cos2_alpha.clamp(min=1e-300))
        C = f / 16 * cos2_alpha * (4 + f * (4 - 3 * cos2_alpha))
        lam_new = L + (1 - C) * f * sin_alpha * (sigma + C * sin_sigma * (cos_2sigma_m + C * cos_sigma * (-1 + 2 * cos_2sigma_m ** 2)))
        if bool((lam_new - lam).abs().max() < tol):
            lam = lam_new
            break
        lam = lam_new
    sinLam, cosLam = torch.sin(lam), torch.cos(lam)
    sin_sigma = torch.sqrt((cosU2 * sinLam) ** 2 + (cosU1 * sinU2 - sinU1 * cosU2 * cosLam) ** 2)
    cos_sigma = sinU1 * sinU2 + cosU1 * cosU2 * cosLam
    sigma = torch.atan2(sin_sigma, cos_sigma)
    sin_alpha = torch.where(sin_sigma == 0, torch.zeros_like(sin_sigma), cosU1 * cosU2 * sinLam / sin_sigma.clamp(min=1e-300))
    cos2_alpha = 1 - sin_alpha ** 2
    cos_2sigma_m = torch.where(cos2_alpha == 0, torch.zeros_like(cos_sigma), cos_sigma - 2 * sinU1 * sinU2 / cos2_alpha.clamp(min=1e-300))
    u2 = cos2_alpha * (a ** 2 - b ** 2) / b ** 2
    A = 1 + u2 / 16384 * (4096 + u2 * (-768 + u2 * (320 - 175 * u2)))
    B = u2 / 1024 * (256 + u2 * (-128 + u2 * (74 - 47 * u2)))
    d_sigma = B * sin_sigma * (cos_2sigma_m + B / 4 * (cos_sigma * (-1 + 2 * cos_2sigma_m ** 2)
                                                       - B / 6 * cos_2sigma_m * (-3 + 4 * sin_sigma ** 2) * (-3 + 4 * cos_2sigma_m ** 2)))
    s = b * A * (sigma - d_sigma)
    s = torch.where(sin_sigma == 0, torch.zeros_like(s), s)
    return s / _UNIT_DIV.get(unit, 1.0)


def euclidean_distance(lat1, lon1, lat2, lon2, unit="m") -> torch.Tensor:
    """Chord distance through the sphere (reference geo_utils.py:325)."""
    x1, y1, z1 = dd_to_cartesian(lat1, lon1)
    x2, y2, z2 = dd_to_cartesian(lat2, lon2)
    d = torch.sqrt((x1 - x2) ** 2 + (y1 - y2) ** 2 + (z1 - z2) ** 2)
    return d / _UNIT_DIV.get(unit, 1.0)


# ------------------------------------------------------------ polygons

def point_in_polygon(lat: torch.Tensor, lon: torch.Tensor, polygon) -> torch.Tensor:
    """Vectorized ray casting: all points × all polygon edges at once
    (reference geo_utils.py:368 — per-point python loop in a UDF).
    polygon: sequence of (lon, lat) vertices (GeoJSON order)."""
    poly = torch.as_tensor(polygon, dtype=torch.float64, device=lat.device)
    px, py = poly[:, 0], poly[:, 1]  # lon, lat
    qx, qy = torch.roll(px, -1), torch.roll(py, -1)
    x = _as_tensor(lon).unsqueeze(1)  # [N,1]
    y = _as_tensor(lat).unsqueeze(1)
    cond = (py > y) != (qy > y)  # edge straddles the horizontal ray
    slope_x = px + (y - py) * (qx - px) / torch.where(qy == py, torch.full_like(qy, 1e-300), qy - py)
    crossing = cond & (x < slope_x)
    return crossing.sum(dim=1) % 2 == 1


def point_in_polygons(lat: torch.Tensor, lon: torch.Tensor, polygon_list, south_west_loc=(), north_east_loc=()) -> torch.Tensor:
    """OR over a list of polygons with an optional bounding-box shortcut
    (reference geo_utils.py:453)."""
    inside = torch.zeros(lat.shape, dtype=torch.bool, device=lat.device)
    if len(south_west_loc) == 2 and len(north_east_loc) == 2:
        bb = (lat >= south_west_loc[0]) & (lat <= north_east_loc[0]) & (lon >= south_west_loc[1]) & (lon <= north_east_loc[1])
    else:
        bb = torch.ones_like(inside)
    idx = bb.nonzero(as_tuple=True)[0]
    if idx.numel() == 0:
        return inside
    la, lo = lat[idx], lon[idx]
    hit = torch.zeros_like(la, dtype=torch.bool)
    for poly in polygon_list:
        hit = hit | point_in_polygon(la, lo, poly)
    inside[idx] = hit
    return inside


# very coarse country bounding boxes for "approx" country membership
# (reference geo_utils.py:799 point_in_country_approx ships a bbox table)
COUNTRY_BBOXES = {
    "india": (6.5, 68.1, 35.5, 97.4),
    "united states": (24.5, -125.0, 49.4, -66.9),
    "usa": (24.5, -125.0, 49.4, -66.9),
    "united kingdom": (49.9, -8.6, 60.9, 1.8),
    "uk": (49.9, -8.6, 60.9, 1.8),
    "australia": (-43.6, 113.3, -10.7, 153.6),
    "china": (18.2, 73.5, 53.6, 135.1),
    "brazil": (-33.8, -73.9, 5.3, -34.8),
    "germany": (47.3, 5.9, 55.1, 15.0),
    "france": (41.3, -5.1, 51.1, 9.6),
    "japan": (24.0, 122.9, 45.5, 145.8),
    "canada": (41.7, -141.0, 83.1, -52.6),
    "russia": (41.2, 19.6, 81.9, 180.0),
    "singapore": (1.2, 103.6, 1.5, 104.0),
    "south africa": (-34.8, 16.5, -22.1, 32.9),
}


def point_in_country_approx(lat: torch.Tensor, lon: torch.Tensor, country: str) -> torch.Tensor:
    key = str(country).strip().lower()
    if key not in COUNTRY_BBOXES:
        raise ValueError(f"No approx bounding box for country '{country}'")
    s, w, n, e = COUNTRY_BBOXES[key]
    return (lat >= s) & (lat <= n) & (lon >= w) & (lon <= e)


# ---------------------------------------------------------- parity aliases
# (reference geo_utils.py names — same math as the tensor helpers above)

def decimal_degrees_to_degrees_minutes_seconds(dd):
    """Reference geo_utils.py:117."""
    d, m, s = dd_to_dms(torch.as_tensor(dd, dtype=torch.float64))
    if d.dim() == 0:
        return float(d), float(m), float(s)
    return d, m, s


def to_latlon_decimal_degrees(loc, input_format, radius=EARTH_RADIUS):
    """Reference geo_utils.py:51 — (lat, lon[, z]) in input_format → dd."""
    if input_format == "dd":
        return _as_tensor(loc[0]), _as_tensor(loc[1])
    if input_format == "radian":
        return radian_to_dd(loc[0], loc[1])
    if input_format == "dms":
        lat = dms_to_dd(*[_as_tensor(v) for v in loc[0]]) if isinstance(loc[0], (tuple, list)) else _as_tensor(loc[0])
        lon = dms_to_dd(*[_as_tensor(v) for v in loc[1]]) if isinstance(loc[1], (tuple, list)) else _as_tensor(loc[1])
        return lat, lon
    if input_format == "cartesian":
        return cartesian_to_dd(loc[0], loc[1], loc[2], radius=radius)
    if input_format == "geohash":
        ints, prec = geohash_str_to_int([loc] if isinstance(loc, str) else list(loc))
        return geohash_decode_int(torch.from_numpy(ints), prec)
    raise TypeError(f"Invalid input_format {input_format}")


def from_latlon_decimal_degrees(loc, output_format, radius=EARTH_RADIUS, geohash_precision=8):
    """Reference geo_utils.py:161 — dd (lat, lon) → output_format."""
    lat, lon = _as_tensor(loc[0]), _as_tensor(loc[1])
    if output_format == "dd":
        return lat, lon
    if output_format == "radian":
        return dd_to_radian(lat, lon)
    if output_format == "dms":
        return dd_to_dms(lat), dd_to_dms(lon)
    if output_format == "cartesian":
        return dd_to_cartesian(lat, lon, radius=radius)
    if output_format == "geohash":
        gh = geohash_encode_int(lat.reshape(-1), lon.reshape(-1), precision=geohash_precision)
        out = geohash_int_to_str(gh, precision=geohash_precision)
        return out[0] if lat.dim() == 0 else out
    raise TypeError(f"Invalid output_format {output_format}")


def f_point_in_polygons(polygon_list, south_west_loc=(), north_east_loc=()):
    """Reference geo_utils.py:530 — curried membership test."""
    def f(lat, lon):
        return point_in_polygons(_as_tensor(lat), _as_tensor(lon), polygon_list, south_west_loc, north_east_loc)

    return f
