"""Tests for coords / sky / coherency / consensus / solutions against
loop oracles transcribed from the reference formulas."""

import math

import numpy as np
import pytest
import torch

from smartcal_amd.radio import coords, sky, coherency, consensus, solutions


def test_coords_roundtrip():
    ra0, dec0 = 1.0, 0.7
    ra, dec = 1.05, 0.76
    l, m, n = coords.radectolm(ra, dec, ra0, dec0)
    # the reference's lmtoradec mirrors l (its simulate.py relies on this:
    # random l is sign-symmetric), so the round trip gives (-l, m)
    ra2, dec2 = coords.lmtoradec(l, m, ra0, dec0)
    l2, m2, _ = coords.radectolm(ra2, dec2, ra0, dec0)
    # (m only approximate: the reference formula is small-field)
    assert abs(float(l2) + l) < 1e-9 and abs(float(m2) - m) < 1e-2
    h, mi, s = coords.rad_to_ra(ra)
    assert abs(float(coords.hms_to_rad(h, mi, s)) - ra) < 1e-9
    d, dm, ds = coords.rad_to_dec(dec)
    assert abs(float(coords.dms_to_rad(d, dm, ds)) - dec) < 1e-9


def _mk_sky(rng, ns=5, gaussian_frac=0.4, f0=1e6):
    names = [("G%d" % i if rng.random() < gaussian_frac else "P%d" % i)
             for i in range(ns)]
    ra = 1.0 + 0.01 * rng.standard_normal(ns)
    dec = 0.7 + 0.01 * rng.standard_normal(ns)
    sI = rng.random(ns) * 2 + 0.1
    sP = rng.standard_normal((ns, 3)) * 0.1
    eX = rng.random(ns) * 0.01
    eY = rng.random(ns) * 0.01
    eP = rng.random(ns)
    return sky.SkyModel.from_arrays(names, ra, dec, sI, sP, f0, eX, eY, eP)


def test_sky_text_roundtrip():
    rng = np.random.default_rng(0)
    sm = _mk_sky(rng)
    sm2 = sky.parse_sky_text(sky.write_sky_text(sm))
    np.testing.assert_allclose(sm2.ra, sm.ra, atol=1e-6)
    np.testing.assert_allclose(sm2.dec, sm.dec, atol=1e-6)
    np.testing.assert_allclose(sm2.sI, sm.sI, rtol=1e-5)
    np.testing.assert_allclose(sm2.sP, sm.sP, rtol=1e-4, atol=1e-6)
    assert list(sm2.gaussian) == list(sm.gaussian)
    cs = sky.ClusterSet([sky.ClusterDef(1, 1, sm.names[:3]),
                         sky.ClusterDef(2, 1, sm.names[3:])])
    cs2 = sky.parse_cluster_text(sky.write_cluster_text(cs))
    assert [c.names for c in cs2] == [c.names for c in cs]
    rs, ra_ = sky.parse_rho_text(sky.write_rho_text([1.5, 2.5], [0.1, 0.2]), 2)
    np.testing.assert_allclose(rs, [1.5, 2.5])
    np.testing.assert_allclose(ra_, [0.1, 0.2])


def _oracle_coherency(sm, cs, uvw, freq, ra0, dec0, smear_bw):
    """Transcription of `skytocoherencies_uvw` (calibration_tools.py:371-464)."""
    c = 2.99792458e8
    uu = uvw[:, 0] * 2 * math.pi / c * freq
    vv = uvw[:, 1] * 2 * math.pi / c * freq
    ww = uvw[:, 2] * 2 * math.pi / c * freq
    T = len(uu)
    K = len(cs)
    C = np.zeros((K, T, 4), dtype=np.complex64)
    fdelta = smear_bw / freq if smear_bw else None
    idx = sm.index
    for ck, cl in enumerate(cs):
        for sname in cl.names:
            i = idx[sname]
            ll, mm, nn = coords.radectolm(sm.ra[i], sm.dec[i], ra0, dec0)
            fr = math.log(freq / sm.f0[i])
            sIo = math.exp(math.log(sm.sI[i]) + sm.sP[i, 0] * fr
                           + sm.sP[i, 1] * fr ** 2 + sm.sP[i, 2] * fr ** 3)
            up = uu * ll + vv * mm + ww * nn
            amp = np.full(T, sIo)
            if fdelta is not None:
                amp = amp * np.abs(np.sinc(up * 0.5 * fdelta / np.pi))
            if sname.startswith("G"):
                phi = -math.acos(nn)
                xi = -math.atan2(-ll, mm)
                cxi, sxi = math.cos(xi), math.sin(xi)
                cphi, sphi = math.cos(phi), math.sin(phi)
                eP = sm.eP[i]
                eX = 2 * sm.eX[i]
                eY = 2 * sm.eY[i]
                uup = uu * cxi - vv * cphi * sxi + ww * sphi * sxi
                vvp = uu * sxi + vv * cphi * cxi - ww * sphi * cxi
                cpa, spa = math.cos(eP), math.sin(eP)
                uut = eX * (cpa * uup - spa * vvp)
                vvt = eY * (spa * uup + cpa * vvp)
                amp = amp * 0.5 * math.pi * np.exp(-(uut ** 2 + vvt ** 2))
            C[ck, :, 0] += ((np.cos(up) + 1j * np.sin(up)) * amp).astype(np.complex64)
        C[ck, :, 3] = C[ck, :, 0]
    return C


@pytest.mark.parametrize("smear", [None, 180e3])
def test_coherency(smear):
    rng = np.random.default_rng(3)
    sm = _mk_sky(rng)
    cs = sky.ClusterSet([sky.ClusterDef(1, 1, sm.names[:2]),
                         sky.ClusterDef(2, 1, sm.names[2:])])
    uvw = rng.standard_normal((40, 3)) * 300.0
    freq, ra0, dec0 = 150e6, 1.0, 0.7
    want = _oracle_coherency(sm, cs, uvw, freq, ra0, dec0, smear)
    got = coherency.predict_coherencies_uvw(
        sm, cs, torch.from_numpy(uvw), freq, ra0, dec0,
        smear_bw=smear).numpy()
    np.testing.assert_allclose(got, want, rtol=2e-4, atol=2e-4)


def test_bpoly_and_consensus():
    from math import comb
    x = np.linspace(0, 1, 7)
    Nn = 3
    y = consensus.bpoly(x, Nn)
    want = np.stack([[comb(Nn, r) * xx ** r * (1 - xx) ** (Nn - r)
                      for r in range(Nn + 1)] for xx in x])
    np.testing.assert_allclose(y, want, rtol=1e-5)

    # consensus_poly sanity: rho=0, alpha=0 → F = I (no constraint)
    freqs = np.linspace(115e6, 185e6, 8)
    F, P = consensus.consensus_poly(3, 4, freqs, 150e6, 2, polytype=1,
                                    rho=0.0, alpha=0.0)
    np.testing.assert_allclose(F, np.eye(8), atol=1e-6)
    # with rho>0 F is symmetric and contractive on the poly subspace
    F, P = consensus.consensus_poly(3, 4, freqs, 150e6, 2, polytype=1,
                                    rho=0.9, alpha=0.0)
    np.testing.assert_allclose(F, F.T, atol=1e-5)
    assert np.linalg.norm(F, 2) <= 1.0 + 1e-4

    H = consensus.hessian_addition(3, 4, freqs, 150e6, 2, 0.9, 0.0)
    assert H.shape == (16, 16)
    np.testing.assert_allclose(H, H.T, atol=1e-4)
    H = consensus.hessian_addition(3, 4, freqs, 150e6, 2, 0.9, 0.5)
    assert H.shape == (16, 16) and np.isfinite(H).all()


def test_solutions_roundtrip():
    rng = np.random.default_rng(5)
    N, Nto, K = 3, 2, 2
    a = rng.standard_normal((8 * N * Nto, K)).astype(np.float32)
    J = solutions.solutions_to_J(a, N, Nto)
    # oracle: the reference's stride loops (calibration_tools.py:111-119)
    Jo = np.zeros((K, 2 * N * Nto, 2), dtype=np.complex64)
    for m in range(K):
        for n in range(N):
            Jo[m, 2 * n:2 * N * Nto:2 * N, 0] = \
                a[8 * n:Nto * 8 * N:N * 8, m] + 1j * a[8 * n + 1:Nto * 8 * N:N * 8, m]
            Jo[m, 2 * n:2 * N * Nto:2 * N, 1] = \
                a[8 * n + 2:Nto * 8 * N:N * 8, m] + 1j * a[8 * n + 3:Nto * 8 * N:N * 8, m]
            Jo[m, 2 * n + 1:2 * N * Nto:2 * N, 0] = \
                a[8 * n + 4:Nto * 8 * N:N * 8, m] + 1j * a[8 * n + 5:Nto * 8 * N:N * 8, m]
            Jo[m, 2 * n + 1:2 * N * Nto:2 * N, 1] = \
                a[8 * n + 6:Nto * 8 * N:N * 8, m] + 1j * a[8 * n + 7:Nto * 8 * N:N * 8, m]
    np.testing.assert_allclose(J, Jo, atol=1e-6)
    np.testing.assert_allclose(solutions.J_to_solutions(J, N), a, atol=1e-6)
    # text round trip: the trailing unit column parses as an extra
    # identity direction (K+1 in the header), as in SAGECal files
    freq, J2 = solutions.parse_solutions_text(
        solutions.format_solutions_text(150e6, N, a))
    assert freq == 150e6
    np.testing.assert_allclose(J2[:K], J, atol=1e-5)
    eye = np.tile(np.eye(2, dtype=np.complex64), (Nto * N, 1))
    np.testing.assert_allclose(J2[K], eye, atol=1e-6)


def test_simulate_systematic_errors():
    rng = np.random.default_rng(7)
    K, N, Ts, Nf = 3, 4, 5, 8
    freqs = np.linspace(115e6, 185e6, Nf)
    gs = solutions.simulate_systematic_errors(K, N, Ts, freqs, 150e6, rng,
                                              lm=rng.standard_normal((K, 2)))
    assert gs.shape == (K, 8 * N * Ts, Nf)
    assert np.isfinite(gs).all()
    # J built from a timeslot must have near-diagonal-dominant structure
    a = gs[:, :, 0].T       # (8N·Ts, K)
    J = solutions.solutions_to_J(a, N, Ts)
    assert J.shape == (K, 2 * N * Ts, 2)


def test_shapelet_roundtrip():
    from smartcal_amd.radio import shapelet
    rng = np.random.default_rng(0)
    text, n0, beta, coeff, ptext = shapelet.generate_random_shapelet_model(
        rng, perturbed=True)
    pos, n0b, betab, coeffb = shapelet.parse_shapelet_model(text)
    assert n0b == n0 and abs(betab - beta) < 1e-12
    np.testing.assert_allclose(coeffb, coeff, rtol=1e-10)
    _, n0p, betap, coeffp = shapelet.parse_shapelet_model(ptext)
    assert betap >= beta
    # perturbation is ~10% in norm
    dn = np.linalg.norm(coeffp - coeff) / np.linalg.norm(coeff)
    assert 0.05 < dn < 0.15
    # basis: zeroth mode is a normalized Gaussian, orthonormal-ish
    l = np.linspace(-0.5, 0.5, 101)
    basis = shapelet.shapelet_basis(3, 0.1, l, np.zeros_like(l))
    assert basis.shape == (9, 101)
    assert np.argmax(basis[0]) == 50


def test_vis_io_roundtrip(tmp_path):
    from smartcal_amd.radio import io as rio
    from smartcal_amd.radio import array as arr, sim
    rng = np.random.default_rng(4)
    layout = arr.lofar_like_layout(N=5, rng=rng)
    skym, cs, *_, ra0, dec0 = sim.make_demixing_sky(rng, n_outliers=1)
    vis = sim.simulate_observation(layout, skym, cs,
                                   np.array([150e6]), ra0, dec0,
                                   Ts=1, Tdelta=2, snr=10, rng=rng,
                                   torch_seed=0)
    text = rio.write_corr_text(vis.uvw, vis.data[0])
    uvw2, v2 = rio.read_corr_text(text)
    np.testing.assert_allclose(uvw2.numpy(), vis.uvw.numpy(), rtol=1e-6)
    np.testing.assert_allclose(v2.numpy(), vis.data[0].numpy(), rtol=1e-5,
                               atol=1e-6)
    p = str(tmp_path / "vis.npz")
    rio.save_visdata(vis, p)
    vis2 = rio.load_visdata(p)
    np.testing.assert_allclose(vis2.data.numpy(), vis.data.numpy())
    assert vis2.N == vis.N and vis2.Tdelta == vis.Tdelta


def test_correct_shapelet_modes():
    from smartcal_amd.radio import shapelet
    rng = np.random.default_rng(1)
    text, n0, beta, coeff = shapelet.generate_random_shapelet_model(rng)
    out = shapelet.correct_shapelet_modes(text)
    _, n0b, betab, coeffb = shapelet.parse_shapelet_model(out)
    assert n0b == n0
    # row block ci scaled by 1/(ci+1)
    want = coeff.reshape(n0, n0) / (np.arange(1, n0 + 1)[:, None])
    np.testing.assert_allclose(coeffb.reshape(n0, n0), want, rtol=1e-6)


def test_convert_model_bbs():
    from smartcal_amd.radio.convert import convert_model, parse_bbs_text
    bbs = """# (Name, Type, Patch, Ra, Dec, I, Q, U, V, ReferenceFrequency, SpectralIndex, MajorAxis, MinorAxis, Orientation) = format
s1, POINT, CasA, 23:23:24.0, 58.48.54.0, 100.0, 0, 0, 0, 150e6, [-0.7], , ,
s2, GAUSSIAN, CasA, 23:23:30.0, 58.49.00.0, 50.0, 0, 0, 0, 150e6, [-0.5], 60, 30, 45
t1, POINT, Target, 12:00:00.0, 45.00.00.0, 2.0, 0, 0, 0, 150e6, [0.1], , ,
"""
    skym, clusters = parse_bbs_text(bbs)
    assert len(skym) == 3
    assert len(clusters) == 2
    assert skym.gaussian.sum() == 1
    assert abs(skym.ra[0] - (23 + 23 / 60 + 24 / 3600) * np.pi / 12) < 1e-9
    sky_t, cl_t, rho_t = convert_model(bbs, start_cluster=2)
    assert "GCasA1" in sky_t
    from smartcal_amd.radio.sky import parse_sky_text, parse_cluster_text
    sky2 = parse_sky_text(sky_t)
    assert len(sky2) == 3
    cl2 = parse_cluster_text(cl_t)
    assert cl2[0].cid == 2


def test_hessian_addition_scalar_matches_matrix():
    freqs = np.linspace(115e6, 185e6, 8)
    for rho, alpha in ((0.9, 0.0), (0.9, 0.5), (12.0, 3.0), (100.0, 5.0)):
        H = consensus.hessian_addition(3, 4, freqs, 150e6, 2, rho, alpha)
        c = consensus.hessian_addition_scalar(3, 4, freqs, 150e6, 2, rho,
                                              alpha)
        np.testing.assert_allclose(H, c * np.eye(16), rtol=2e-4,
                                   atol=2e-4 * max(1, abs(c)))


def test_parser_roundtrips_property():
    """Property-style fuzz of the text round trips (hypothesis-lite)."""
    rng = np.random.default_rng(42)
    for trial in range(10):
        ns = int(rng.integers(1, 12))
        sm = _mk_sky(rng, ns=ns, gaussian_frac=rng.random())
        sm2 = sky.parse_sky_text(sky.write_sky_text(sm))
        np.testing.assert_allclose(sm2.ra, sm.ra, atol=1e-6)
        np.testing.assert_allclose(sm2.sI, sm.sI, rtol=1e-4)
        K = int(rng.integers(1, 5))
        rs = rng.random(K) * 100
        ra_ = rng.random(K)
        rs2, ra2 = sky.parse_rho_text(sky.write_rho_text(rs, ra_), K)
        np.testing.assert_allclose(rs2, rs, rtol=1e-5)
        np.testing.assert_allclose(ra2, ra_, rtol=1e-4, atol=1e-5)
        N, Nto, Kj = (int(rng.integers(2, 6)), int(rng.integers(1, 4)),
                      int(rng.integers(1, 4)))
        a = rng.standard_normal((8 * N * Nto, Kj)).astype(np.float32)
        J = solutions.solutions_to_J(a, N, Nto)
        np.testing.assert_allclose(solutions.J_to_solutions(J, N), a,
                                   atol=1e-6)
