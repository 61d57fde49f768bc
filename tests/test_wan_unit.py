"""Unit tests for the WAN token bucket and charge model."""

import time

import pytest

from geomx_amd.kvstore.wan import TokenBucket, cross_party_bytes


def test_disabled_bucket_never_blocks():
    tb = TokenBucket(0.0)
    t0 = time.perf_counter()
    tb.charge(10 ** 12, sync_device=False)
    assert time.perf_counter() - t0 < 0.05
    assert tb.total_bytes == 0


def test_charge_paces_to_rate():
    tb = TokenBucket(1.0)  # 1 Gbit/s
    nbytes = 2.5e6  # 20 ms at 1 Gbit/s
    t0 = time.perf_counter()
    tb.charge(nbytes, sync_device=False)
    dt = time.perf_counter() - t0
    assert 0.015 < dt < 0.08, dt
    assert tb.total_bytes == nbytes


def test_charge_async_overlaps():
    tb = TokenBucket(1.0)
    t0 = time.perf_counter()
    ready = tb.charge_async(2.5e6)  # reserves 20 ms, returns immediately
    assert time.perf_counter() - t0 < 0.01
    time.sleep(0.025)  # "compute" longer than the transfer
    t1 = time.perf_counter()
    tb.wait_until(ready)
    assert time.perf_counter() - t1 < 0.01  # nothing left to wait


def test_back_to_back_charges_accumulate():
    tb = TokenBucket(1.0)
    r1 = tb.charge_async(1.25e6)  # 10 ms
    r2 = tb.charge_async(1.25e6)  # +10 ms, serialized on the link
    assert r2 - r1 == pytest.approx(0.01, rel=0.2)


def test_cross_party_bytes_model():
    assert cross_party_bytes("all_reduce", 100, 1) == 0
    assert cross_party_bytes("all_reduce", 100, 2) == pytest.approx(100.0)
    assert cross_party_bytes("all_reduce", 100, 4) == pytest.approx(150.0)
    assert cross_party_bytes("reduce", 100, 2) == pytest.approx(50.0)
    assert cross_party_bytes("broadcast", 100, 4) == pytest.approx(300.0)
    assert cross_party_bytes("all_gather", 100, 4) == pytest.approx(300.0)
    assert cross_party_bytes("send", 100, 4) == 100.0
    with pytest.raises(ValueError):
        cross_party_bytes("nope", 1, 2)


def test_token_bucket_rtt():
    """RTT adds per-transfer latency on top of serialization; rtt with
    zero bandwidth still delays (latency-only link)."""
    import time

    from geomx_amd.kvstore.wan import TokenBucket

    tb = TokenBucket(gbps=1.0, rtt_ms=20.0)
    t0 = time.perf_counter()
    tb.charge(125_000, sync_device=False)   # 1 ms serialization + 20 ms RTT
    dt = time.perf_counter() - t0
    assert 0.019 <= dt < 0.08, dt

    lat_only = TokenBucket(gbps=0.0, rtt_ms=10.0)
    assert lat_only.enabled
    t0 = time.perf_counter()
    lat_only.charge(1_000_000, sync_device=False)
    assert 0.009 <= time.perf_counter() - t0 < 0.05

    # async reservation reports completion incl. RTT
    tb2 = TokenBucket(gbps=1.0, rtt_ms=50.0)
    ready = tb2.charge_async(125_000)
    assert ready - time.perf_counter() > 0.045
