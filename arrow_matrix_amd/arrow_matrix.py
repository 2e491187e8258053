"""Abstract arrow-matrix interface — the plugin surface the HIP engine slots
behind. Mirrors the reference's `arrow/arrow_matrix.py:12-111` ABC (same
method names and meanings) minus the mpi4py types.
"""
from abc import ABC, abstractmethod


class ArrowMatrix(ABC):

    # The number of block tiles per side of this matrix
    tiles_per_side: int

    @abstractmethod
    def result_tile(self):
        """Returns this rank's result tile (C_i)."""

    @abstractmethod
    def feature_tile(self):
        """Returns this rank's feature tile (X_i)."""

    @abstractmethod
    def spmm(self, device: str = None):
        """Compute one SpMM over this matrix (collective across its ranks)."""

    @abstractmethod
    def set_features(self, X):
        """Sets this rank's feature slice. Stores a REFERENCE, no copy
        (reference arrow_slim_mpi.py:285-293)."""

    @abstractmethod
    def load_sparse_matrix_from_blocks(self, blocks):
        """Loads this rank's sparse blocks from a block grid."""

    @abstractmethod
    def is_column_rank(self) -> bool:
        """True if this rank holds a column (feature-carrying) tile."""

    @abstractmethod
    def zero_rhs(self, number_of_rows_per_rank: int, number_of_columns: int,
                 dtype=None):
        """Allocate/clear X and C buffers; must be called before the first
        SpMM iteration (reference arrow_matrix.py:64-72)."""

    @abstractmethod
    def allgather_result(self, C):
        """All-gathers the result into C (numpy) and returns it."""

    def set_features_slice_from_features(self, X):
        """Deprecated in the reference (arrow_matrix.py:84-92): set this
        rank's slice from the full feature matrix."""
        raise NotImplementedError

    @staticmethod
    def column_subgroup(tiles_per_side, group):
        return group

    @staticmethod
    def row_subgroup(tiles_per_side, group):
        return group
