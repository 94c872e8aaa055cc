# Robustness of the native thrift/footer parser (csrc/parquet_meta.cpp,
# thrift_compact.h): corrupt or truncated SSTs must surface HX_ERR_FORMAT /
# HX_ERR_IO through hx_open — never a crash, hang, or silent bad catalog.
# CPU-only: hx_open parses footers without touching the GPU.
import os
import shutil

import numpy as np
import pytest

from horaedb_amd import Store, HxError


@pytest.fixture(scope="module")
def valid_sst(tmp_path_factory):
    from tools.gen_ssts import write_sst
    d = tmp_path_factory.mktemp("fuzzsrc")
    path = str(d / "1.sst")
    rng = np.random.default_rng(9)
    n = 20_000
    series = np.sort(rng.integers(0, 100, n).astype(np.uint64))
    ts = np.arange(n, dtype=np.int64) * 3
    write_sst(path, series, ts, rng.random(n), 1)
    return path


def _store_with(tmp_path, data: bytes):
    ddir = tmp_path / "data"
    ddir.mkdir(exist_ok=True)
    (ddir / "1.sst").write_bytes(data)
    return str(tmp_path)


def test_truncations_fail_cleanly(valid_sst, tmp_path):
    blob = open(valid_sst, "rb").read()
    # cut at many points, including inside the footer and the length word
    for cut in [0, 1, 3, 4, 7, 8, len(blob) // 2, len(blob) - 200,
                len(blob) - 12, len(blob) - 8, len(blob) - 5,
                len(blob) - 4, len(blob) - 1]:
        sub = tmp_path / f"cut{cut}"
        sub.mkdir()
        store = _store_with(sub, blob[:cut])
        with pytest.raises(HxError) as ei:
            Store(store)
        assert ei.value.code in (1, 2, 7), f"cut={cut} -> {ei.value}"


def test_footer_bitflips_never_crash(valid_sst, tmp_path):
    # flip bytes across the thrift footer: parser must either still produce
    # a catalog (flip hit a value) or raise, never crash/hang
    blob = bytearray(open(valid_sst, "rb").read())
    flen = int.from_bytes(blob[-8:-4], "little")
    foot_start = len(blob) - 8 - flen
    rng = np.random.default_rng(12)
    opened = 0
    raised = 0
    for _ in range(120):
        b = bytearray(blob)
        pos = int(rng.integers(foot_start, len(b) - 8))
        b[pos] ^= int(rng.integers(1, 256))
        sub = tmp_path / f"flip{opened + raised}"
        sub.mkdir()
        store = _store_with(sub, bytes(b))
        try:
            with Store(store) as st:
                st.catalog()
            opened += 1
        except HxError as e:
            assert e.code in (1, 2, 7)
            raised += 1
        shutil.rmtree(sub)
    assert opened + raised == 120


def test_bad_footer_length_word(valid_sst, tmp_path):
    blob = bytearray(open(valid_sst, "rb").read())
    for val in [0xFFFFFFFF, len(blob), len(blob) * 2, 0]:
        b = bytearray(blob)
        b[-8:-4] = int(val).to_bytes(4, "little")
        sub = tmp_path / f"len{val}"
        sub.mkdir()
        store = _store_with(sub, bytes(b))
        try:
            with Store(store) as st:
                st.catalog()
        except HxError as e:
            assert e.code in (1, 2, 7)


def test_wrong_magic(valid_sst, tmp_path):
    blob = bytearray(open(valid_sst, "rb").read())
    blob[-4:] = b"NOPE"
    store = _store_with(tmp_path, bytes(blob))
    with pytest.raises(HxError) as ei:
        Store(store)
    assert ei.value.code == 2
