"""MFMA fragment-layout probes (gpu): assumed per-lane layouts vs matmul.
Asymmetric inputs so operand/output transposes cannot pass (guide G9)."""

from __future__ import annotations

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")


@pytest.mark.parametrize(("shape", "m", "n", "k"), [(32, 32, 32, 16), (16, 16, 16, 32)])
def test_mfma_layout(shape, m, n, k):
    from vllm_tgis_adapter_amd import _C

    torch.manual_seed(0)
    # asymmetric, non-square-symmetric values
    a = (torch.arange(m * k, device="cuda").reshape(m, k).float() % 7 - 3)
    b = (torch.arange(k * n, device="cuda").reshape(k, n).float() % 5 - 2) * 0.5
    a[0, 1] = 9.0  # extra asymmetry
    b[1, 0] = -7.0
    d = torch.zeros(m, n, dtype=torch.float32, device="cuda")
    _C.mfma_probe(a.bfloat16().contiguous(), b.bfloat16().contiguous(), d, shape)
    torch.cuda.synchronize()
    ref = a.bfloat16().float() @ b.bfloat16().float()
    assert torch.allclose(d, ref, atol=1e-2), (d - ref).abs().max()
