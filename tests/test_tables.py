"""Table construction parity: product (vectorised) vs oracle (literal
restatement) vs golden vectors from the reference itself."""
import os

import numpy as np
import pytest

from arrow_matrix_amd import tables
from oracle import all_to_all_tables_ref, aggregation_permutation_ref

GOLDEN = os.path.join(os.path.dirname(__file__), 'golden', 'reference_katsets.npz')


@pytest.fixture(scope='module')
def golden():
    return np.load(GOLDEN)


def test_all_to_all_tables_vs_reference_golden(golden):
    n = int(golden['a2a_n_cases'][0])
    assert n >= 10
    for ci in range(n):
        perm = golden[f'a2a_{ci}_in_perm']
        rpr, cols, total, off = (int(x) for x in golden[f'a2a_{ci}_meta'])
        counts, displs, sp, ap = tables.all_to_all_tables(perm, rpr, cols, total, off)
        np.testing.assert_array_equal(counts, golden[f'a2a_{ci}_counts'])
        np.testing.assert_array_equal(displs, golden[f'a2a_{ci}_displs'])
        np.testing.assert_array_equal(sp, golden[f'a2a_{ci}_send_perm'])
        np.testing.assert_array_equal(ap, golden[f'a2a_{ci}_agg_perm'])


def test_oracle_restatement_vs_reference_golden(golden):
    n = int(golden['a2a_n_cases'][0])
    for ci in range(n):
        perm = golden[f'a2a_{ci}_in_perm']
        rpr, cols, total, off = (int(x) for x in golden[f'a2a_{ci}_meta'])
        counts, displs, sp, ap = all_to_all_tables_ref(perm, rpr, cols, total, off)
        np.testing.assert_array_equal(counts, golden[f'a2a_{ci}_counts'])
        np.testing.assert_array_equal(displs, golden[f'a2a_{ci}_displs'])
        np.testing.assert_array_equal(sp, golden[f'a2a_{ci}_send_perm'])
        np.testing.assert_array_equal(ap, golden[f'a2a_{ci}_agg_perm'])


def test_product_vs_oracle_random():
    rng = np.random.default_rng(3)
    for _ in range(50):
        rpr = int(rng.integers(1, 60))
        total = int(rng.integers(1, 10))
        off = int(rng.integers(0, total))
        vals = rng.integers(0, rpr * (total + 2), size=rpr).astype(np.int64)
        cols = int(rng.integers(1, 9))
        a = tables.all_to_all_tables(vals, rpr, cols, total, off)
        b = all_to_all_tables_ref(vals, rpr, cols, total, off)
        for x, y in zip(a, b):
            np.testing.assert_array_equal(np.asarray(x), np.asarray(y))


def test_inverse_permutation_case():
    """The reference's own unit test (test_arrowmpi.py:24-48)."""
    ranks, prev_ranks, rpr, cols = 2, 6, 4, 6
    permutation = np.asarray(list(reversed(range(ranks * rpr))))
    for i in range(ranks):
        sl = permutation[i * rpr:(i + 1) * rpr]
        counts, displs, p, out_p = tables.all_to_all_tables(sl, rpr, cols,
                                                            prev_ranks + ranks, prev_ranks)
        assert counts[ranks + prev_ranks - i - 1] == rpr * cols
        assert int(np.sum(counts)) == rpr * cols
        assert displs[ranks + prev_ranks - i - 1] == 0
        counts, displs, p, out_p = tables.all_to_all_tables(sl, rpr, cols,
                                                            ranks + prev_ranks, 0)
        assert counts[ranks - i - 1] == rpr * cols
        assert displs[ranks - i - 1] == 0


def test_routing_tables_match_reference_layout():
    """At one block per rank, the generalised shared-rank tables reduce to the
    reference's: same send grouping and receive order."""
    rng = np.random.default_rng(5)
    for _ in range(20):
        nb_src = int(rng.integers(1, 6))
        nb_dst = int(rng.integers(1, 6))
        w = int(rng.integers(2, 10))
        P = max(nb_src, nb_dst)
        # a random to_next-style mapping from src rows onto dst rows (+ sentinel)
        n_src, n_dst = nb_src * w, nb_dst * w
        vals = rng.permutation(max(n_src, n_dst * 2))[:n_src].astype(np.int64)
        vals[vals >= n_dst] = 2 * w * max(nb_src, nb_dst)  # sentinel
        own_dst = tables.contiguous_block_owners(nb_dst, P)  # identity here
        assert np.array_equal(own_dst, np.arange(nb_dst))
        for r in range(nb_src):
            sl = vals[r * w:(r + 1) * w]
            # reference tables: total = nb_src + nb_dst, offset nb_src; its
            # counts for dest rank nb_src+d == generalised counts for rank d
            ref_counts, _, ref_sp, _ = all_to_all_tables_ref(sl, w, 1,
                                                             nb_src + nb_dst, nb_src)
            cnt, send_rows = tables.routing_send_tables(sl, w, own_dst, nb_dst, P)
            np.testing.assert_array_equal(cnt[:nb_dst], np.asarray(ref_counts[nb_src:]))
            n_valid = int(cnt.sum())
            np.testing.assert_array_equal(send_rows, ref_sp[:n_valid])
        own_src = tables.contiguous_block_owners(nb_src, P)
        for r in range(nb_dst):
            sl = vals_recv = rng.permutation(max(n_dst, n_src * 2))[:w].astype(np.int64)
            sl = np.where(sl >= n_src, 2 * w * max(nb_src, nb_dst), sl)
            ref_counts, _, _, ref_ap = all_to_all_tables_ref(sl, w, 1,
                                                             nb_src + nb_dst, 0)
            cnt, recv_rows = tables.routing_recv_tables(sl, w, own_src, nb_src, P)
            np.testing.assert_array_equal(cnt[:nb_src], np.asarray(ref_counts[:nb_src]))
            np.testing.assert_array_equal(recv_rows, ref_ap)


def test_number_of_blocks_vs_golden(golden):
    n = int(golden['nb_n_cases'][0])
    for ci in range(n):
        nnz_per_row = golden[f'nb_{ci}_nnz_per_row']
        w = int(golden[f'nb_{ci}_width'][0])
        assert tables.number_of_blocks(nnz_per_row, w) == int(golden[f'nb_{ci}_result'][0])


def test_number_of_blocks_all_zero_raises():
    with pytest.raises(ValueError):
        tables.number_of_blocks(np.zeros(10, dtype=np.int64), 2)


def test_golden_fixtures_are_fresh(tmp_path, monkeypatch):
    """When the reference tree is available (build container), regenerating
    the golden fixtures must reproduce the committed ones — guards against
    fixture drift after generator or synth changes. Skipped on boxes
    without /root/reference."""
    import subprocess
    import sys
    ref = os.environ.get('ARROW_REFERENCE_PATH', '/root/reference')
    if not os.path.isdir(ref):
        pytest.skip("reference tree not available")
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    gen = os.path.join(repo, 'tests', 'golden', 'gen_golden.py')
    out = tmp_path / 'fresh.npz'
    env = dict(os.environ, ARROW_GOLDEN_OUT=str(out))
    r = subprocess.run([sys.executable, gen], capture_output=True, text=True,
                       timeout=300, env=env)
    assert r.returncode == 0, r.stderr[-1500:]
    fresh = np.load(out)
    committed = np.load(GOLDEN)
    assert sorted(fresh.files) == sorted(committed.files)
    for key in committed.files:
        np.testing.assert_array_equal(fresh[key], committed[key], err_msg=key)


def test_fold_maps_match_exchange_semantics():
    """The composed fold maps (arrow_dec._compute_fold_maps) must agree
    with what the forward/backward exchanges actually do at P=1: routing
    X through the exchange chain equals gathering through M_i, and the
    backward cascade's accumulation equals scattering through R_i."""
    import numpy as np
    from arrow_matrix_amd import synth, tables
    from arrow_matrix_amd.arrow_dec import ArrowDecompositionMPI

    rng = np.random.default_rng(77)
    for n_blocks, width, seed in ([4, 2], 5, 1), ([3, 3, 2], 4, 2):
        decomp = synth.synth_arrow_decomposition(width, n_blocks, avg_deg=4,
                                                 seed=seed)
        perms = [np.asarray(p) for _, p in decomp]
        nb = np.asarray(n_blocks)
        _, to_prev, to_next = tables.pad_and_compose_permutations(
            perms, nb, width)
        # single process: each part's slice is its first nb_i*width rows
        # (the loader's slicing, arrow_dec.py::load_decomposition_new)
        tp_s = [None if t is None else t[:n_blocks[i] * width]
                for i, t in enumerate(to_prev)]
        tn_s = [None if t is None else t[:n_blocks[i] * width]
                for i, t in enumerate(to_next)]
        arrow = ArrowDecompositionMPI.initialize(
            None, nb, tp_s, tn_s, width, 3, device='cpu')
        assert arrow._fold_maps is not None
        n0 = n_blocks[0] * width
        X0 = rng.random((n0, 3)).astype(np.float32)
        # forward chain: X_i via successive routing == gather through M_i
        X_prev = X0
        for i in range(1, len(n_blocks)):
            ni = n_blocks[i] * width
            tn = tn_s[i - 1]
            X_i = np.zeros((ni, 3), np.float32)
            valid = tn < ni
            X_i[tn[valid]] = X_prev[valid]
            M_i, R_i = arrow._fold_maps[i]
            X_via_map = np.zeros((ni, 3), np.float32)
            m_ok = M_i >= 0
            X_via_map[m_ok] = X0[M_i[m_ok]]
            np.testing.assert_array_equal(X_i, X_via_map)
            # backward single hop vs R map through the chain: scatter C_i
            # through the cascade == scatter through R_i into part 0
            C_i = rng.random((ni, 3)).astype(np.float32)
            acc = C_i
            for j in range(i, 0, -1):
                tp = tp_s[j]
                nj = n_blocks[j - 1] * width
                nxt = np.zeros((nj, 3), np.float32)
                v = tp[:acc.shape[0]] < nj
                nxt[tp[:acc.shape[0]][v]] = acc[v]
                acc = nxt
            via_R = np.zeros((n0, 3), np.float32)
            r_ok = R_i >= 0
            via_R[R_i[r_ok]] = C_i[r_ok]
            np.testing.assert_array_equal(acc, via_R)
            X_prev = X_i
