"""End-to-end engine parity on GPU (single process): the product HIP path
(device='gpu') against the oracle golden, through the full public API
(save -> load -> initialize -> step -> allgather). `pytest -m gpu`."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("n_blocks,width,k,seed", [
    ([4], 2000, 16, 0),        # BASELINE config 2 shape (8k rows, w=2000, k=16)
    ([3], 64, 32, 1),
    ([4, 2], 50, 8, 2),
    ([4, 3, 2], 32, 16, 3),
    ([3], 48, 300, 4),         # k > 256: column-tiled dual-operand launches
    ([2], 40, 7, 5),           # odd k (VEC=1 path) through the fused engine
])
def test_engine_gpu_matches_golden(n_blocks, width, k, seed):
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from tests.test_engine_cpu import _run_engine
    from arrow_matrix_amd import synth
    decomp = synth.synth_arrow_decomposition(width, n_blocks, avg_deg=8,
                                             seed=seed, hub_rows=2 if seed == 0 else 0)
    results, goldens = _run_engine(decomp, width, n_blocks, k, iters=1,
                                   device='gpu', seed=seed)
    np.testing.assert_allclose(results[0], goldens[0], rtol=1e-4, atol=1e-4)


def test_engine_gpu_iterated():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from tests.test_engine_cpu import _run_engine
    from arrow_matrix_amd import synth
    n_blocks, width, k = [3, 2], 40, 16
    decomp = synth.synth_arrow_decomposition(width, n_blocks, avg_deg=5, seed=9)
    results, goldens = _run_engine(decomp, width, n_blocks, k, iters=3,
                                   device='gpu', seed=9)
    for C, G in zip(results, goldens):
        np.testing.assert_allclose(C, G, rtol=5e-4, atol=5e-4)


def test_native_library_is_loaded():
    """The GPU path must run through the in-tree libarrowspmm.so — no torch
    eager fallback."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from arrow_matrix_amd import hip
    hip._load()
    maps = open('/proc/self/maps').read()
    assert 'libarrowspmm.so' in maps


@pytest.mark.gpu
def test_bench_spmm_gpu_synthetic():
    """The spmm_arrow CLI flow end-to-end on the GPU device (synthetic
    fallback path, arrow_bench.py:28-41)."""
    import os
    import tempfile
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from arrow_matrix_amd.arrow_bench import bench_spmm
    cwd = os.getcwd()
    with tempfile.TemporaryDirectory() as td:
        try:
            os.chdir(td)
            arrow = bench_spmm(None, 64, 8, 2, True, 'gpu',
                               p_per_side=3, ba_neighbors=5)
            assert arrow is not None
            C = arrow.B.allgather_result()
            assert np.isfinite(C).all() and np.abs(C).max() > 0
        finally:
            os.chdir(cwd)


@pytest.mark.gpu
def test_allreduce_x0_iterated_equivalence():
    """The iterated-loop allreduce-X_0 optimisation (bench/cfg5 semantics)
    gives the same results as the reduce+bcast path."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    import os
    import tempfile
    from arrow_matrix_amd import graphio, synth
    from arrow_matrix_amd.arrow_dec import ArrowDecompositionMPI
    from oracle import compute_spmm

    n_blocks, width, k = [4], 64, 16
    decomp = synth.synth_arrow_decomposition(width, n_blocks, avg_deg=6, seed=31)
    n = n_blocks[0] * width
    rng = np.random.default_rng(31)
    X_orig = (2 * rng.random((n, k)) - 1).astype(np.float32)
    with tempfile.TemporaryDirectory() as td:
        prefix = os.path.join(td, 'g')
        graphio.save_decomposition_new(decomp, prefix, width)
        blocks, nb, tp, tn = ArrowDecompositionMPI.load_decomposition_new(
            None, prefix, width)
        arrow = ArrowDecompositionMPI.initialize(None, nb, tp, tn, width, k,
                                                 device='gpu')
        arrow.load_data_from_blocks(blocks)
        arrow.zero_rhs(width, k)
        for eng in arrow.engines:
            eng.allreduce_x0 = True
        perm0 = decomp[0][1]
        arrow.B.set_features(X_orig[perm0].copy())
        golden_X = X_orig
        for _ in range(3):
            arrow.step()
            C = arrow.B.allgather_result()
            golden = compute_spmm(decomp, golden_X)[perm0]
            np.testing.assert_allclose(C, golden, rtol=5e-4, atol=5e-4)
            golden_X = compute_spmm(decomp, golden_X)
            arrow.B.set_features(arrow.B.result_tile())


@pytest.mark.gpu
def test_engine_gpu_split_col(monkeypatch):
    """ARROW_SPLIT_COL=1 (hub-sorted X_0 structure) parity on GPU."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    monkeypatch.setenv('ARROW_SPLIT_COL', '1')
    from tests.test_engine_cpu import _run_engine
    from arrow_matrix_amd import synth
    decomp = synth.synth_arrow_decomposition(64, [4], avg_deg=8, seed=17,
                                             hub_rows=2)
    results, goldens = _run_engine(decomp, 64, [4], 16, iters=2,
                                   device='gpu', seed=17)
    for C, G in zip(results, goldens):
        np.testing.assert_allclose(C, G, rtol=2e-4, atol=2e-4)


@pytest.mark.gpu
def test_engine_gpu_chunked_row0(monkeypatch):
    """Force the chunked C_0 pipeline (ARROW_ROW0_CHUNKS=4) at world=1 so
    its numerics are validated before the driver's multi-GPU run."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    monkeypatch.setenv('ARROW_ROW0_CHUNKS', '4')
    from tests.test_engine_cpu import _run_engine
    from arrow_matrix_amd import synth
    decomp = synth.synth_arrow_decomposition(64, [4], avg_deg=8, seed=23,
                                             hub_rows=2, hub_deg=100)
    results, goldens = _run_engine(decomp, 64, [4], 16, iters=2,
                                   device='gpu', seed=23)
    for C, G in zip(results, goldens):
        np.testing.assert_allclose(C, G, rtol=2e-4, atol=2e-4)


@pytest.mark.gpu
def test_allreduce_x0_multi_part_invalidation():
    """With allreduce_x0 force-enabled at L=2, the exchange must invalidate
    the cached X_0 (stale-X_0 regression test)."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    import os
    import tempfile
    from arrow_matrix_amd import graphio, synth
    from arrow_matrix_amd.arrow_dec import ArrowDecompositionMPI
    from oracle import compute_spmm

    n_blocks, width, k = [3, 2], 40, 8
    decomp = synth.synth_arrow_decomposition(width, n_blocks, avg_deg=5, seed=41)
    n = n_blocks[0] * width
    rng = np.random.default_rng(41)
    X_orig = (2 * rng.random((n, k)) - 1).astype(np.float32)
    with tempfile.TemporaryDirectory() as td:
        prefix = os.path.join(td, 'g')
        graphio.save_decomposition_new(decomp, prefix, width)
        blocks, nb, tp, tn = ArrowDecompositionMPI.load_decomposition_new(
            None, prefix, width)
        arrow = ArrowDecompositionMPI.initialize(None, nb, tp, tn, width, k,
                                                 device='gpu')
        arrow.load_data_from_blocks(blocks)
        arrow.zero_rhs(width, k)
        for eng in arrow.engines:
            eng.allreduce_x0 = True  # deliberately forced at L>1
        perm0 = decomp[0][1]
        arrow.B.set_features(X_orig[perm0].copy())
        golden_X = X_orig
        for _ in range(3):
            arrow.step()
            C = arrow.B.allgather_result()
            golden = compute_spmm(decomp, golden_X)[perm0]
            np.testing.assert_allclose(C, golden, rtol=5e-4, atol=5e-4)
            golden_X = compute_spmm(decomp, golden_X)
            arrow.B.set_features(arrow.B.result_tile())


@pytest.mark.gpu
def test_engine_gpu_rest_chunked(monkeypatch):
    """Force multi-chunk rest structures (int32-overflow headroom path)."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    monkeypatch.setenv('ARROW_REST_CHUNK_NNZ', '500')
    from tests.test_engine_cpu import _run_engine
    from arrow_matrix_amd import synth
    decomp = synth.synth_arrow_decomposition(64, [4], avg_deg=8, seed=29)
    results, goldens = _run_engine(decomp, 64, [4], 16, iters=2,
                                   device='gpu', seed=29)
    for C, G in zip(results, goldens):
        np.testing.assert_allclose(C, G, rtol=2e-4, atol=2e-4)


@pytest.mark.gpu
def test_engine_gpu_multipart_sequential_exchange(monkeypatch):
    """ARROW_FOLD=0: the sequential permutation-exchange path (device
    permute/scatter kernels) must stay correct at L>1 on GPU now that the
    folded path is the single-process default."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    monkeypatch.setenv('ARROW_FOLD', '0')
    from tests.test_engine_cpu import _run_engine
    from arrow_matrix_amd import synth
    decomp = synth.synth_arrow_decomposition(50, [4, 2], avg_deg=8, seed=2)
    results, goldens = _run_engine(decomp, 50, [4, 2], 8, iters=2,
                                   device='gpu', seed=2)
    for C, G in zip(results, goldens):
        np.testing.assert_allclose(C, G, rtol=1e-4, atol=1e-4)


@pytest.mark.gpu
def test_engine_gpu_folded_matches_sequential(monkeypatch):
    """GPU fold parity: folded vs sequential exchange on the same L=3
    decomposition."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    import os
    import tempfile
    from arrow_matrix_amd import graphio, synth
    from arrow_matrix_amd.arrow_dec import ArrowDecompositionMPI
    width, n_blocks, k = 40, [4, 3, 2], 16
    decomp = synth.synth_arrow_decomposition(width, n_blocks, avg_deg=6, seed=33)
    n = n_blocks[0] * width
    rng = np.random.default_rng(33)
    X = (2 * rng.random((n, k)) - 1).astype(np.float32)

    def run(fold):
        monkeypatch.setenv('ARROW_FOLD', '1' if fold else '0')
        with tempfile.TemporaryDirectory() as td:
            prefix = os.path.join(td, 'g')
            graphio.save_decomposition_new(decomp, prefix, width)
            blocks, nb, tp, tn = ArrowDecompositionMPI.load_decomposition_new(
                None, prefix, width)
            arrow = ArrowDecompositionMPI.initialize(None, nb, tp, tn, width,
                                                     k, device='gpu')
            arrow.load_data_from_blocks(blocks)
            arrow.zero_rhs(width, k)
            assert (arrow._folded is not None) == fold
            arrow.B.set_features(X[decomp[0][1]].copy())
            outs = []
            for _ in range(2):
                arrow.step()
                outs.append(arrow.B.allgather_result().copy())
                arrow.B.set_features(arrow.B.result_tile())
            return outs

    folded, seq = run(True), run(False)
    for F, S in zip(folded, seq):
        np.testing.assert_allclose(F, S, rtol=1e-5, atol=1e-5)


@pytest.mark.gpu
def test_engine_gpu_row_folded_matches_golden(monkeypatch):
    """ROW-fold (ARROW_FOLD=2) on GPU vs oracle golden at L=2/L=3."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    monkeypatch.setenv('ARROW_FOLD', '2')
    from tests.test_engine_cpu import _run_engine
    from arrow_matrix_amd import synth
    for n_blocks, seed in ([4, 2], 2), ([4, 3, 2], 3):
        decomp = synth.synth_arrow_decomposition(50, n_blocks, avg_deg=8,
                                                 seed=seed)
        results, goldens = _run_engine(decomp, 50, n_blocks, 8, iters=2,
                                       device='gpu', seed=seed)
        for C, G in zip(results, goldens):
            np.testing.assert_allclose(C, G, rtol=1e-4, atol=1e-4)
