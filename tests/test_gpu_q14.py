"""Q14 (part join + CASE-style conditional aggregation), pinned to the public
TPC-H Q14 answer @SF1: promo_revenue = 16.38077862639554 (tolerance 1e-6 rel:
f64 atomic scalar sums)."""
import pytest

import oracle

pytestmark = pytest.mark.gpu


def test_q14_sf1_official_answer():
    import trino_amd
    from trino_amd import tpch_queries
    s = trino_amd.Session(0)
    try:
        got = tpch_queries.q14_gpu(s, 1.0)
        assert abs(got["promo_revenue"] - 16.38077862639554) <= 1e-6 * 16.38
    finally:
        s.close()
