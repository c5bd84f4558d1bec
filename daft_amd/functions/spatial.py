"""Spatial functions (capability of /root/reference/src/daft-geo/src/
great_circle_distance.rs:10-35 + daft/functions/spatial.py:8).
Vectorized haversine on the device (torch elementwise — fused enough for
a 4-input transcendental op)."""
from __future__ import annotations

import math

import torch

from ..expressions.expressions import Expression, ScalarFn, _to_node
from ..schema import DataType
from ..series import Series

_EARTH_RADIUS_M = 6_371_000.0


def _gcd_series(lat1: Series, lon1: Series, lat2: Series,
                lon2: Series) -> Series:
    ts = [s.data.to(torch.float64) for s in (lat1, lon1, lat2, lon2)]
    n = max(t.numel() for t in ts)
    ts = [t.expand(n) if t.numel() == 1 and n > 1 else t for t in ts]
    a1, o1, a2, o2 = [torch.deg2rad(t) for t in ts]
    dlat = a2 - a1
    dlon = o2 - o1
    s1 = torch.sin(dlat * 0.5)
    s2 = torch.sin(dlon * 0.5)
    h = (s1 * s1 + torch.cos(a1) * torch.cos(a2) * s2 * s2).clamp(0.0, 1.0)
    out = 2.0 * _EARTH_RADIUS_M * torch.asin(torch.sqrt(h))
    # invalid coordinates -> null (matches the reference's validity rule)
    deg = [t.to(torch.float64) for t in ts]
    ok = (torch.isfinite(deg[0]) & torch.isfinite(deg[1]) &
          torch.isfinite(deg[2]) & torch.isfinite(deg[3]) &
          (deg[0].abs() <= 90) & (deg[2].abs() <= 90) &
          (deg[1].abs() <= 180) & (deg[3].abs() <= 180))
    validity = ok
    for s in (lat1, lon1, lat2, lon2):
        if s.validity is not None:
            v = s.validity
            if v.numel() == 1 and n > 1:
                v = v.expand(n)
            validity = validity & v
    if bool(validity.all()):
        validity = None
    return Series(lat1.name, DataType.float64(), data=out,
                  validity=validity)


def great_circle_distance(lat1, lon1, lat2, lon2) -> Expression:
    """Great-circle (haversine) distance in meters between two
    (lat, lon) coordinate pairs in degrees."""
    nodes = [_to_node(x) for x in (lat1, lon1, lat2, lon2)]
    return Expression(ScalarFn("great_circle_distance", _gcd_series,
                               nodes, DataType.float64()))
