"""SoA (MeasurementArray) pipeline unit tests — CPU."""
import numpy as np
import pytest

from dpo_amd.chordal import odometry_initialization
from dpo_amd.measurements import (MeasurementArray,
                                  concat_measurement_arrays,
                                  odometry_initialization_array,
                                  partition_measurement_array)
from dpo_amd.partition import contiguous_partition, partition_measurements
from dpo_amd.quadratic import assemble_connection_laplacian
from dpo_amd.synthetic import grid3d, grid3d_soa


def test_ma_roundtrip():
    meas, n = grid3d(side=3, seed=1)
    ma = MeasurementArray.from_list(meas)
    back = ma.to_list()
    assert len(back) == len(meas)
    for a, b in zip(meas, back):
        assert np.allclose(a.R, b.R) and np.allclose(a.t, b.t)
        assert a.p1 == b.p1 and a.p2 == b.p2


def test_partition_ma_matches_object_path():
    meas, n = grid3d(side=4, seed=2)
    part = contiguous_partition(n, 3)
    odo_o, priv_o, sh_o, pose_map, p2i, counts_o = partition_measurements(
        meas, n, part, 3)
    ma = MeasurementArray.from_list(meas)
    odo_a, priv_a, sh_a, local_idx, global_of, counts_a = \
        partition_measurement_array(ma, n, part, 3)
    assert counts_o == counts_a
    for rb in range(3):
        assert len(odo_o[rb]) == len(odo_a[rb])
        assert len(priv_o[rb]) == len(priv_a[rb])
        assert len(sh_o[rb]) == len(sh_a[rb])
        # identical local indices (same ordering convention)
        assert [m.p1 for m in odo_o[rb]] == list(odo_a[rb].p1)
        assert [m.p2 for m in sh_o[rb]] == list(sh_a[rb].p2)


def test_odometry_prefix_scan_matches_sequential():
    meas, n = grid3d(side=3, seed=3, rot_noise=0.1)
    odo = [m for m in meas if m.p1 + 1 == m.p2]
    T_seq = odometry_initialization(3, n, odo)
    ma = MeasurementArray.from_list(odo)
    T_vec = odometry_initialization_array(3, n, ma)
    assert np.allclose(T_seq, T_vec, atol=1e-10)


def test_grid3d_soa_matches_q_structure():
    # Q assembled from the SoA generator must be a valid PSD Laplacian
    ma, n = grid3d_soa(side=3, seed=0)
    Q = assemble_connection_laplacian(ma, n, 3)
    A = Q.to_scalar_csr().to_dense().numpy()
    assert np.allclose(A, A.T, atol=1e-10)
    w = np.linalg.eigvalsh(A)
    assert w.min() > -1e-8


def test_grid3d_soa_outlier_mask():
    ma, n = grid3d_soa(side=4, seed=1, outlier_prob=0.2)
    frac = ma.outlier_mask.mean()
    n_odo = n - 1
    # outliers only among loop closures, roughly 20%
    assert not ma.outlier_mask[:n_odo].any()
    lc_frac = ma.outlier_mask[n_odo:].mean()
    assert 0.1 < lc_frac < 0.3


def test_chordal_soa_matches_direct():
    """SoA/GPU-native chordal init (normal-equations PCG on the rotation
    connection Laplacian, BSR kernels) must match the reference-style
    direct least-squares solve (DPGO_utils.cpp:273-409)."""
    from dpo_amd.chordal import (chordal_initialization,
                                 chordal_initialization_soa)
    from dpo_amd.synthetic import city2d
    for mk, args in ((grid3d, dict(side=4, seed=0)),
                     (city2d, dict(side=8, seed=1))):
        meas, n = mk(**args)
        ma = MeasurementArray.from_list(meas)
        T1 = chordal_initialization(meas[0].d, n, meas)
        T2 = chordal_initialization_soa(ma, n, tol=1e-11)
        assert np.abs(T1 - T2).max() < 1e-7


def test_soa_l2_driver_uses_chordal_init():
    """SoA L2 driver initializes from the SoA chordal solve (not
    odometry): its round-0 cost must match the object-path driver that
    computes the reference chordal init."""
    from dpo_amd.comm import Comm
    from dpo_amd.dist_driver import DistributedRBCDDriver
    from dpo_amd.synthetic import grid3d, grid3d_soa
    ma, n = grid3d_soa(side=4, seed=5)
    meas = ma.to_list()
    d_soa = DistributedRBCDDriver(ma, n, 2, Comm(), r=5,
                                  partition="contiguous", device="cpu")
    d_obj = DistributedRBCDDriver(meas, n, 2, Comm(), r=5,
                                  partition="contiguous", device="cpu")
    r_soa = d_soa.run(max_iters=3, gradnorm_tol=0.0)
    r_obj = d_obj.run(max_iters=3, gradnorm_tol=0.0)
    c_soa, c_obj = r_soa.trace[0][0], r_obj.trace[0][0]
    assert abs(c_soa - c_obj) < 1e-4 * max(1.0, abs(c_obj))
