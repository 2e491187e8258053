# TEST INFRASTRUCTURE ONLY.
#
# This package is the CPU oracle: a numpy/scipy restatement of the reference's
# arrow-decomposition iterated-SpMM semantics (spcl/arrow-matrix). The
# reference's own CPU arithmetic IS scipy CSR `@` (arrow_slim_mpi.py:109-144),
# so scipy is the arithmetic ground truth; this package restates the
# distributed dataflow around it sequentially.
#
# Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
# import or call anything in here. The product path (arrow_matrix_amd with
# device='gpu') must never route through this package; it fails loudly when
# its HIP extension is missing.
from .arrow_oracle import (  # noqa: F401
    all_to_all_tables_ref,
    aggregation_permutation_ref,
    compute_spmm,
    slim_arrow_spmm,
    decomposition_step,
)
