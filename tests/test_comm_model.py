"""Analytic comm-model tests (reference ``actions/utils.py:42-124``)."""
import math

import numpy as np
import pytest

from ddls_amd.cluster.comm_model import (calc_one_to_one_time,
                                         calc_ramp_all_reduce_time,
                                         effective_trx_per_comm,
                                         get_parallel_add_comp_time)


def test_effective_trx():
    assert effective_trx_per_comm(cg=32, d=1, J=1) == 0
    # spare = min(32//1, 32//(d-1)) - 1
    assert effective_trx_per_comm(cg=32, d=2, J=1) == 1 + (min(32, 32) - 1)
    assert effective_trx_per_comm(cg=4, d=3, J=1) == 1 + (min(4, 2) - 1)


def test_parallel_add_comp_time():
    # devices=4: n_op=2, n_bytes=10, AI=0.2, total_ops=2*(data/4)/2
    data = 1e6
    t = get_parallel_add_comp_time(data, 4, MEM_FRQ=2e12, pi=130e12,
                                   bytes_per_comp=2)
    expected = (2 * (data / 4) / 2) / min(2e12 * 0.2, 130e12)
    assert t == pytest.approx(expected)


def test_all_reduce_hand_computed():
    # 2 servers in 2 comm groups, 1 rack value, same node index:
    # cgs=2, node_ids=1, racks=1 -> subgroups [2, min(2,1)=1, 1, ceil(1/4)=1]
    # only step 0 active
    x, DR, lat, io = 4, 4e11, 50e-9, 100e-9
    msg = 1e6
    t = calc_ramp_all_reduce_time(message_size=msg, node_ids=1, racks=1, cgs=2,
                                  x=x, DATA_RATE=DR, latency=lat, IO_latency=io)
    data_per_tx = DR / x
    eff_bw = effective_trx_per_comm(cg=x, d=2, J=1) * data_per_tx
    m0 = math.ceil(msg / 2)
    comp = get_parallel_add_comp_time(m0 * 2, 2)
    comm = lat + 2 * io + m0 / eff_bw
    assert t == pytest.approx(2 * comm + comp)


def test_one_to_one():
    t = calc_one_to_one_time(1e6, DATA_RATE=4e11, latency=50e-9, IO_latency=100e-9)
    assert t == pytest.approx(50e-9 + 200e-9 + 1e6 / 4e11)


def test_all_reduce_monotone_in_message_size():
    ts = [calc_ramp_all_reduce_time(s, node_ids=2, racks=2, cgs=2, x=4,
                                    DATA_RATE=4e11)
          for s in (1e5, 1e6, 1e7)]
    assert ts[0] < ts[1] < ts[2]
