/* arrow_spmm — C ABI of the MI355X-native arrow-SpMM compute library.
 *
 * This is the drop-in boundary of the rebuilt hot path of spcl/arrow-matrix:
 * the reference does its per-block compute through cupy's cuSPARSE CSRMM
 * binding (`A @ X` on cp.sparse.csr_matrix, arrow_slim_mpi.py:190,211,231)
 * and its permutation routing through host fancy-indexing
 * (arrow_dec_mpi.py:421,437,526,544). Each exported entry point below
 * replaces one of those interfaces:
 *
 *   arrow_csr_create / arrow_csr_destroy
 *       replaces arrow/common/sp2cp.py:6-16 (_sp2cp host->device CSR
 *       conversion, re-done every call in the reference,
 *       arrow_slim_mpi.py:184,210,226) — here the block is uploaded ONCE
 *       and stays resident in HBM.
 *   arrow_spmm
 *       replaces the cupy `A @ X` CSRMM (arrow_slim_mpi.py:190,211,231):
 *       C (+)= A @ X for one resident CSR block, X/C row-major fp32 device
 *       buffers, X (cols,k), C (rows,k).
 *   arrow_gather_rows_f32
 *       replaces the host gathers sendbuf = T[perm]
 *       (arrow_dec_mpi.py:421,526) — payload stays in HBM.
 *   arrow_scatter_rows_f32 / arrow_scatter_add_rows_f32
 *       replace C_i[perm] = recvbuf (arrow_dec_mpi.py:544) and
 *       C_i[perm] += recvbuf (arrow_dec_mpi.py:437).
 *
 * Conventions: device pointers are HIP device memory owned by the caller
 * (e.g. torch tensors' data_ptr); host pointers are read during the call
 * only. `stream` is a hipStream_t (or NULL for the default stream). All
 * functions return 0 on success, negative on error (arrow_last_error()
 * gives the message). One process per GPU; not thread-safe.
 */
#ifndef ARROW_SPMM_H
#define ARROW_SPMM_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* Library/device introspection */
int arrow_device_count(void);
int arrow_set_device(int device);
const char *arrow_last_error(void);
int arrow_synchronize(void);
/* ABI version (major*1000 + minor) */
int arrow_abi_version(void);

/* Upload a CSR block (host arrays) once; returns a handle >= 0, < 0 on
 * error. indptr has rows+1 int64 entries; indices int32; data fp32.
 * Builds the row-segment work list for the SpMM kernel at upload time. */
int64_t arrow_csr_create(int64_t rows, int64_t cols, int64_t nnz,
                         const int64_t *indptr, const int32_t *indices,
                         const float *data);
/* As arrow_csr_create_rows plus option flags. flags bit 0: order work
 * items by first column instead of row — for hub-heavy structures whose
 * long rows split into column-contiguous items, a queue segment then
 * covers a COLUMN window of X (locality for the per-XCD scheduler).
 * row_ids may be NULL. */
int64_t arrow_csr_create_opts(int64_t rows, int64_t cols, int64_t nnz,
                              const int64_t *indptr, const int32_t *indices,
                              const float *data, const int64_t *row_ids,
                              int flags);
/* As arrow_csr_create, but structure row r writes output row row_ids[r]
 * (reordered layouts, e.g. hub-sorted first-block-column entries). */
int64_t arrow_csr_create_rows(int64_t rows, int64_t cols, int64_t nnz,
                              const int64_t *indptr, const int32_t *indices,
                              const float *data, const int64_t *row_ids);
int arrow_csr_destroy(int64_t handle);
/* nnz of a resident block (for flop accounting) */
int64_t arrow_csr_nnz(int64_t handle);
/* Enable the XCD-contiguous work remap for this structure (a performance
 * hint for uniform banded rows; off by default). */
int arrow_csr_set_xcd_remap(int64_t handle, int enable);
/* Per-XCD queue scheduler for this structure: workgroups drain contiguous
 * nnz-balanced row segments via per-XCD atomic chunk counters (keeps the
 * consumers of each X row in ONE XCD's L2). mode: 1 on, 0 off,
 * -1 follow the ARROW_QUEUE env default. */
int arrow_csr_set_queue(int64_t handle, int mode);
/* Per-structure launch-grid override for the queue scheduler (workgroup
 * count; 0 = the ARROW_Q_BLOCKS env default). Used to co-schedule two
 * concurrent structure launches on separate HIP streams so neither grid
 * fills the whole chip. */
int arrow_csr_set_qblocks(int64_t handle, int blocks);

/* C (+)= A @ X.  X: (cols, k) fp32 row-major device;  C: (rows, k).
 * beta = 0: C = A@X (rows not touched by A are zeroed);  beta = 1: C += A@X. */
int arrow_spmm(int64_t handle, const float *X_dev, float *C_dev, int64_t k,
               int beta, void *stream);

/* Dual-operand SpMM for the FUSED diagonal + first-block-column layout
 * (replaces the reference's back-to-back C_i = A_ii@X_i; C_i += A_i0@X_0,
 * arrow_slim_mpi.py:121-144, writing C once instead of read-modify-write):
 * a column index c >= 0 reads X0[c]; c < 0 reads X1[-c-1]. The negative
 * encoding is fixed at arrow_csr_create time by the caller's indices. */
int arrow_spmm_dual(int64_t handle, const float *X0_dev, const float *X1_dev,
                    float *C_dev, int64_t k, int beta, void *stream);

/* dst[i, :] = src[idx[i], :]   (n rows of width k, fp32, device) */
int arrow_gather_rows_f32(const float *src_dev, float *dst_dev,
                          const int64_t *idx_dev, int64_t n, int64_t k,
                          void *stream);
/* dst[idx[i], :] = src[i, :] */
int arrow_scatter_rows_f32(float *dst_dev, const float *src_dev,
                           const int64_t *idx_dev, int64_t n, int64_t k,
                           void *stream);
/* dst[idx[i], :] += src[i, :] */
int arrow_scatter_add_rows_f32(float *dst_dev, const float *src_dev,
                               const int64_t *idx_dev, int64_t n, int64_t k,
                               void *stream);

/* Fused gather+scatter for the rank-local share of the permutation routing
 * (arrow_dec_mpi.py:526+544 / 421+437 collapsed into one pass):
 * dst[dst_idx[i], :] (+)= src[src_idx[i], :] */
int arrow_permute_rows_f32(float *dst_dev, const float *src_dev,
                           const int64_t *dst_idx_dev, const int64_t *src_idx_dev,
                           int64_t n, int64_t k, void *stream);
int arrow_permute_add_rows_f32(float *dst_dev, const float *src_dev,
                               const int64_t *dst_idx_dev, const int64_t *src_idx_dev,
                               int64_t n, int64_t k, void *stream);

#ifdef __cplusplus
}
#endif

#endif /* ARROW_SPMM_H */
