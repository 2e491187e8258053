"""Size-independent property tests at large scale (the full-size parity
gate of DESIGN.md §parity): linearity of the SpMM operator and
cross-backend consistency at sizes where the oracle cannot run row-by-row."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope='module')
def big_engine():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    import bench as B
    from arrow_matrix_amd.arrow_dec import ArrowDecompositionMPI
    rows, nb, k, band = 4_000_000, 4, 128, 1024
    w = rows // nb
    grids, first, last = B.build_blocks_for_rank(0, 1, w, nb, 1, 'cuda', band)
    arrow = ArrowDecompositionMPI.initialize(
        None, np.array([nb]), [None], [None], w, k, device='gpu')
    arrow.load_data_from_blocks(grids)
    arrow.zero_rhs(w, k)
    return arrow, w, k


def _apply(arrow, X):
    arrow.B.set_features(X)
    arrow.step()
    return arrow.B.result_tile().clone()


def test_linearity_at_scale(big_engine):
    """A(aX + bY) == a·A(X) + b·A(Y) — holds for the true operator at any
    size; catches indexing/accumulation corruption the small-size oracle
    comparisons cannot reach."""
    arrow, w, k = big_engine
    g = torch.Generator(device='cuda')
    g.manual_seed(0)
    n = arrow.B.X_i.shape[0]
    X = torch.rand((n, k), generator=g, device='cuda') * 2 - 1
    Y = torch.rand((n, k), generator=g, device='cuda') * 2 - 1
    a, b = 0.75, -1.25
    AX = _apply(arrow, X)
    AY = _apply(arrow, Y)
    AXY = _apply(arrow, (a * X + b * Y).contiguous())
    ref = a * AX + b * AY
    err = (AXY - ref).abs().max().item()
    scale = ref.abs().max().item()
    assert err <= 1e-4 * max(scale, 1.0), (err, scale)


def test_zero_input_gives_zero(big_engine):
    arrow, w, k = big_engine
    n = arrow.B.X_i.shape[0]
    Z = torch.zeros((n, k), device='cuda')
    AZ = _apply(arrow, Z)
    assert AZ.abs().max().item() == 0.0
