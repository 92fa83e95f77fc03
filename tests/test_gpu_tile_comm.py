"""RCCL collective layer on a real MI355X (single-GPU boxes: world=1
init + round-trip; the gpurun pool has one GPU per box, so multi-device
RCCL runs under the driver's 8-GPU scaling bench via bench.py --mode
tile)."""

import os

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

hipflux = pytest.importorskip("hipflux")
from hipflux import _native


def require_gpu():
    if hipflux.hip_device_count() == 0:
        pytest.fail("gpu test ran on a host with no HIP device")


def test_rccl_init_and_roundtrip():
    require_gpu()
    uid = _native.TileComm.make_uid()
    assert len(uid) == 128
    comm = _native.TileComm(0, 1, uid, 0)
    payload = bytes(range(256)) * 64          # 16 KB
    for schedule in (0, 1):
        ms = comm.exchange_host(payload, schedule)
        assert ms >= 0
        assert comm.gathered(0, len(payload)) == payload


def test_rccl_boundary_from_recon():
    """boundary_dev packs real recon rows; the exchanged halo equals the
    encoder's own reconstruction boundary."""
    require_gpu()
    rng = np.random.default_rng(3)
    w, h = 320, 192
    pipe = _native.BenchPipeline("gpu", w, h, qp=28, stripe_height=64,
                                 output_mode=2)
    frame = np.ascontiguousarray(
        rng.integers(0, 256, (h, w, 4), dtype=np.uint8))
    pipe.encode(frame, True)
    ptr, nbytes = pipe.boundary_dev(16)
    assert ptr != 0 and nbytes == 2 * 16 * w
    uid = _native.TileComm.make_uid()
    comm = _native.TileComm(0, 1, uid, 0)
    comm.exchange(ptr, nbytes, 1)
    halo = np.frombuffer(comm.gathered(0, nbytes), np.uint8)
    top, bottom = halo[:16 * w].reshape(16, w), halo[16 * w:].reshape(16, w)
    # compare against the pipeline's own recon dump
    _, dump = _native._pipeline_encode("gpu", [frame], w, h, 28, 64, 2, True)
    ry = np.frombuffer(dump["y"], np.uint8).reshape(-1, dump["ypitch"])
    assert np.array_equal(top, ry[:16, :w])
    assert np.array_equal(bottom, ry[h - 16:h, :w])
