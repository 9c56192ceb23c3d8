"""Direct unit tests for small helper modules (small_complex, streams,
gymapi) — elsewhere only covered transitively."""

import numpy as np
import pytest
import torch

from smartcal_amd.radio import small_complex as sc
from smartcal_amd.utils.streams import StreamFork
from smartcal_amd import gymapi


def _rc(*shape):
    g = torch.Generator().manual_seed(0)
    return (torch.randn(*shape, generator=g)
            + 1j * torch.randn(*shape, generator=g)).to(torch.complex64)


def test_small_complex_vs_matmul():
    A = _rc(5, 7, 2, 2)
    B = _rc(5, 7, 2, 2)
    torch.testing.assert_close(sc.mm2(A, B), A @ B)
    torch.testing.assert_close(sc.mm2H(A, B), A @ B.conj().mT)
    torch.testing.assert_close(sc.Hmm2(A, B), A.conj().mT @ B)
    Am = _rc(4, 6, 2)
    Bm = _rc(4, 3, 2)
    torch.testing.assert_close(sc.abH_k2(Am, Bm), Am @ Bm.conj().mT)
    Bk = _rc(4, 2, 5)
    torch.testing.assert_close(sc.outer_k2(Am, Bk), Am @ Bk)


def test_stream_fork_cpu_passthrough():
    sf = StreamFork(torch.device("cpu"))
    r = sf(lambda: 1, lambda: 2, lambda: 3)
    assert r == (1, 2, 3)


def test_gymapi_box_space():
    box = gymapi.spaces.Box(low=-np.ones(3, np.float32),
                            high=np.ones(3, np.float32),
                            dtype=np.float32)
    s = box.sample()
    assert s.shape == (3,)
    assert (s >= -1).all() and (s <= 1).all()


@pytest.mark.gpu
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
def test_stream_fork_gpu_results():
    sf = StreamFork(torch.device("cuda"))
    a = torch.ones(256, device="cuda")
    r = sf(lambda: a + 1, lambda: a * 3, lambda: a - 1)
    torch.cuda.synchronize()
    assert float(r[0].sum()) == 512 and float(r[1].sum()) == 768
    assert float(r[2].sum()) == 0
