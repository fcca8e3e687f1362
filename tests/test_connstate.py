"""ConnStateTable (config 5): id lifecycle, counters, HBM sizing math.

CPU-side semantics tests (the numpy path is the same class the GPU path
runs; tests/test_gpu_engine.py covers device-resident parity).
"""

import numpy as np
import pytest

from gofr_amd.engine.connstate import (CS_BYTES_IN, CS_BYTES_OUT, CS_FLAGS,
                                       CS_LAST_BATCH, CS_PROTO, CS_REQS,
                                       ConnStateTable, PROTO_GRPC,
                                       PROTO_HTTP, per_conn_bytes)


def test_open_close_reuse():
    tab = ConnStateTable(capacity=8)
    a = tab.open(3, PROTO_HTTP)
    assert len(set(a.tolist())) == 3 and tab.n_open == 3
    assert (tab.stats(a)[:, CS_FLAGS] == 1).all()
    tab.close(a[:2])
    assert tab.n_open == 1
    b = tab.open(2, PROTO_GRPC)
    # LIFO free list: the two just-closed ids come back first
    assert set(b.tolist()) == set(a[:2].tolist())
    assert (tab.stats(b)[:, CS_PROTO] == PROTO_GRPC).all()


def test_capacity_enforced():
    tab = ConnStateTable(capacity=4)
    tab.open(4)
    with pytest.raises(RuntimeError, match="full"):
        tab.open(1)


def test_record_batch_counters():
    tab = ConnStateTable(capacity=16)
    ids = tab.open(4)
    bin_ = np.array([100, 200, 300, 400], np.int64)
    bout = np.array([10, 20, 30, 40], np.int64)
    tab.record_batch(ids, bin_, bout, batch_no=7)
    # same conn appearing twice in a batch accumulates
    tab.record_batch(ids[:2], bin_[:2], bout[:2], batch_no=8)
    s = tab.stats(ids)
    assert s[:, CS_REQS].tolist() == [2, 2, 1, 1]
    assert s[:, CS_BYTES_IN].tolist() == [200, 400, 300, 400]
    assert s[:, CS_BYTES_OUT].tolist() == [20, 40, 30, 40]
    assert s[:, CS_LAST_BATCH].tolist() == [8, 8, 7, 7]
    tab.close(ids)
    assert (tab.stats(ids) == 0).all()


def test_hbm_sizing_math():
    # planning numbers against the MI355X's 288 GB HBM3E
    hbm = 288 * 10**9
    cap = ConnStateTable.sized_for_hbm(frac=0.5, hbm_bytes=hbm)
    # >= 8M connections fit in half of one GPU's HBM at 16 KiB windows
    assert cap > 8_000_000
    # 100k conns (the config-5 target) are < 1% of the card
    assert 100_000 * per_conn_bytes() < 0.01 * hbm
    tab = ConnStateTable(capacity=100_000)
    assert tab.hbm_bytes() == 100_000 * per_conn_bytes()
