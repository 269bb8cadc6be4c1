// torch.ops bindings for the sparse HIP (gfx950) kernels.
//
// Loaded via torch.ops.load_library from sparse/kernels/_build/sparse_hip.so;
// Python wrappers live in sparse/kernels/__init__.py.
#include <torch/library.h>

#include <ATen/ATen.h>

// launchers defined in the .hip translation units
void spmv_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor,
              int64_t, double);
void spmv_dot_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                  at::Tensor, at::Tensor, int64_t);
void csr_row_spmv_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                      at::Tensor, int64_t);
void axpby_norm2_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor, bool,
                     bool, at::Tensor);
void build_ell_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                   int64_t, int64_t);
void build_dia_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                   int64_t, int64_t);
void ell_spmv_plain_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                        at::Tensor, at::Tensor, int64_t, int64_t, int64_t,
                        int64_t, int64_t);
void ell_spmv_dot_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                      at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                      int64_t, int64_t, int64_t, int64_t, int64_t);
void ell_jacobi_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                    at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                    at::Tensor, int64_t, int64_t, int64_t, double);
void cg_xr_norm2_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                     at::Tensor, at::Tensor, at::Tensor);
void dia_spmv_bpdot_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                        at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                        at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                        at::Tensor, int64_t, int64_t, int64_t, int64_t,
                        int64_t);
void dia_spmv_plain_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                        at::Tensor, at::Tensor, int64_t, int64_t, int64_t,
                        int64_t, int64_t, int64_t, int64_t);
void dia_spmv_dot_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                      at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                      int64_t, int64_t, int64_t, int64_t, int64_t, int64_t,
                      int64_t);
void dia_residual_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                      at::Tensor, at::Tensor, at::Tensor, int64_t, int64_t,
                      int64_t, int64_t, int64_t, int64_t, int64_t);
void dia_jacobi_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                    at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                    at::Tensor, int64_t, int64_t, int64_t, int64_t, int64_t,
                    double, int64_t, int64_t);
void add_nnz_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor);
void add_compute_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                     at::Tensor, at::Tensor, at::Tensor, at::Tensor, double,
                     double);
void mult_nnz_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor);
void mult_compute_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                      at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                      at::Tensor);
void mult_dense_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor);
void axpby_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor, bool, bool);
void spmm_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor,
              int64_t);
void rspmm_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor);
void bsr_spmm_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                  int64_t);
void sddmm_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor,
               at::Tensor, int64_t);
void csr_to_dense_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor);
void coo_to_csr_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                    at::Tensor, at::Tensor, at::Tensor, at::Tensor);
void dense_to_csr_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor, bool);
void csr_diagonal_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor, int64_t);
void csc_spmv_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                  int64_t);
void csc_spmm_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                  int64_t);
void tropical_spmv_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor, int64_t);
void rk_calc_dy_hip(at::Tensor, at::Tensor, double, at::Tensor);
void cdist_hip(at::Tensor, at::Tensor, at::Tensor);
void spgemm_nnz_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                    at::Tensor, int64_t, int64_t);
void spgemm_compute_hip(at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                        at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                        at::Tensor, at::Tensor, int64_t, int64_t);

TORCH_LIBRARY(sparse_hip, m) {
  m.def("spmv(Tensor indptr, Tensor indices, Tensor values, Tensor x, "
        "Tensor(a!) y, int col_lo, float beta) -> ()");
  m.def("add_nnz(Tensor aip, Tensor aix, Tensor bip, Tensor bix, "
        "Tensor(a!) out) -> ()");
  m.def("add_compute(Tensor aip, Tensor aix, Tensor av, Tensor bip, "
        "Tensor bix, Tensor bv, Tensor cip, Tensor(a!) cix, Tensor(b!) cv, "
        "float alpha, float beta) -> ()");
  m.def("mult_nnz(Tensor aip, Tensor aix, Tensor bip, Tensor bix, "
        "Tensor(a!) out) -> ()");
  m.def("mult_compute(Tensor aip, Tensor aix, Tensor av, Tensor bip, "
        "Tensor bix, Tensor bv, Tensor cip, Tensor(a!) cix, Tensor(b!) cv) -> ()");
  m.def("mult_dense(Tensor indptr, Tensor indices, Tensor vals, Tensor D, "
        "Tensor(a!) out) -> ()");
  m.def("axpby(Tensor(a!) y, Tensor x, Tensor a, Tensor b, bool isalpha, "
        "bool negate) -> ()");
  m.def("spmv_dot(Tensor indptr, Tensor indices, Tensor values, Tensor x, "
        "Tensor(a!) y, Tensor pvec, Tensor(b!) dot_out, int col_lo) -> ()");
  m.def("csr_row_spmv(Tensor indptr, Tensor indices, Tensor values, Tensor x, "
        "Tensor(a!) y, int col_lo) -> ()");
  m.def("axpby_norm2(Tensor(a!) y, Tensor x, Tensor a, Tensor b, bool isalpha, "
        "bool negate, Tensor(b!) dot_out) -> ()");
  m.def("build_ell(Tensor indptr, Tensor indices, Tensor values, "
        "Tensor(a!) eidx, Tensor(b!) evals, int W, int pad_idx) -> ()");
  m.def("build_dia(Tensor indptr, Tensor indices, Tensor values, "
        "Tensor offs, Tensor(a!) dvals, int W, int row0) -> ()");
  m.def("ell_spmv(Tensor eidx, Tensor evals, Tensor hlo, Tensor own, "
        "Tensor hhi, Tensor(a!) y, int W, int m, int col_lo, "
        "int rbase, int rhi) -> ()");
  m.def("ell_spmv_dot(Tensor eidx, Tensor evals, Tensor hlo, Tensor own, "
        "Tensor hhi, Tensor(a!) y, Tensor pvec, Tensor(b!) dot_partial, "
        "int W, int m, int col_lo, int rbase, int rhi) -> ()");
  m.def("ell_jacobi(Tensor eidx, Tensor evals, Tensor hlo, Tensor own, "
        "Tensor hhi, Tensor xloc, Tensor b, Tensor dinv, Tensor(a!) xout, "
        "int W, int m, int col_lo, float omega) -> ()");
  m.def("cg_xr_norm2(Tensor(a!) x, Tensor p, Tensor(b!) r, Tensor q, "
        "Tensor a, Tensor b, Tensor(c!) dot_out) -> ()");
  m.def("dia_spmv_bpdot(Tensor dvals, Tensor offs, Tensor r_hlo, Tensor r_own, "
        "Tensor r_hhi, Tensor p_hlo, Tensor p_own, Tensor p_hhi, "
        "Tensor(a!) pnew, Tensor(b!) q, Tensor beta_num, Tensor beta_den, "
        "Tensor(c!) dot_partial, int W, int m, int col_lo, int row0, "
        "int wsize) -> ()");
  m.def("dia_spmv(Tensor dvals, Tensor offs, Tensor hlo, Tensor own, "
        "Tensor hhi, Tensor(a!) y, int W, int m, int col_lo, int row0, "
        "int wsize, int rbase, int rhi) -> ()");
  m.def("dia_spmv_dot(Tensor dvals, Tensor offs, Tensor hlo, Tensor own, "
        "Tensor hhi, Tensor(a!) y, Tensor pvec, Tensor(b!) dot_partial, "
        "int W, int m, int col_lo, int row0, int wsize, int rbase, "
        "int rhi) -> ()");
  m.def("dia_residual(Tensor dvals, Tensor offs, Tensor hlo, Tensor own, "
        "Tensor hhi, Tensor b, Tensor(a!) y, int W, int m, int col_lo, "
        "int row0, int wsize, int rbase, int rhi) -> ()");
  m.def("dia_jacobi(Tensor dvals, Tensor offs, Tensor hlo, Tensor own, "
        "Tensor hhi, Tensor xloc, Tensor b, Tensor dinv, Tensor(a!) xout, "
        "int W, int m, int col_lo, int row0, int wsize, float omega, "
        "int rbase, int rhi) -> ()");
  m.def("spmm(Tensor indptr, Tensor indices, Tensor vals, Tensor B, "
        "Tensor(a!) C, int col_lo) -> ()");
  m.def("rspmm(Tensor indptr, Tensor indices, Tensor vals, Tensor A, "
        "Tensor(a!) C) -> ()");
  m.def("bsr_spmm(Tensor bptr, Tensor bcol, Tensor bvals, Tensor B, "
        "Tensor(a!) C, int col_lo) -> ()");
  m.def("sddmm(Tensor indptr, Tensor indices, Tensor vals, Tensor C, "
        "Tensor D, Tensor(a!) out, int col_lo) -> ()");
  m.def("csr_to_dense(Tensor indptr, Tensor indices, Tensor vals, "
        "Tensor(a!) out) -> ()");
  m.def("coo_to_csr(Tensor rows, Tensor cols, Tensor vals, Tensor(a!) cursor, "
        "Tensor indptr, Tensor(b!) out_idx, Tensor(c!) out_vals, "
        "Tensor(d!) flags) -> ()");
  m.def("dense_to_csr(Tensor D, Tensor(a!) indptr_or_counts, "
        "Tensor(b!) indices, Tensor(c!) vals, bool fill) -> ()");
  m.def("csr_diagonal(Tensor indptr, Tensor indices, Tensor vals, "
        "Tensor(a!) out, int row_offset) -> ()");
  m.def("csc_spmv(Tensor colptr, Tensor rowidx, Tensor vals, Tensor x, "
        "Tensor(a!) y, int rlo) -> ()");
  m.def("csc_spmm(Tensor colptr, Tensor rowidx, Tensor vals, Tensor B, "
        "Tensor(a!) C, int rlo) -> ()");
  m.def("tropical_spmv(Tensor indptr, Tensor indices, Tensor x, "
        "Tensor(a!) y, int col_lo) -> ()");
  m.def("rk_calc_dy(Tensor K, Tensor avec, float h, Tensor(a!) dy) -> ()");
  m.def("cdist(Tensor XA, Tensor XB, Tensor(a!) out) -> ()");
  m.def("spgemm_nnz(Tensor aip, Tensor aix, Tensor bip, Tensor bix, "
        "Tensor rowlist, Tensor(a!) nnz_out, int a_col_lo, int hash_size) -> ()");
  m.def("spgemm_compute(Tensor aip, Tensor aix, Tensor av, Tensor bip, "
        "Tensor bix, Tensor bv, Tensor rowlist, Tensor cip, Tensor(a!) cix, "
        "Tensor(b!) cv, int a_col_lo, int hash_size) -> ()");
}

TORCH_LIBRARY_IMPL(sparse_hip, CUDA, m) {
  m.impl("spmv", spmv_hip);
  m.impl("add_nnz", add_nnz_hip);
  m.impl("add_compute", add_compute_hip);
  m.impl("mult_nnz", mult_nnz_hip);
  m.impl("mult_compute", mult_compute_hip);
  m.impl("mult_dense", mult_dense_hip);
  m.impl("axpby", axpby_hip);
  m.impl("spmv_dot", spmv_dot_hip);
  m.impl("csr_row_spmv", csr_row_spmv_hip);
  m.impl("axpby_norm2", axpby_norm2_hip);
  m.impl("build_ell", build_ell_hip);
  m.impl("build_dia", build_dia_hip);
  m.impl("ell_spmv", ell_spmv_plain_hip);
  m.impl("ell_spmv_dot", ell_spmv_dot_hip);
  m.impl("ell_jacobi", ell_jacobi_hip);
  m.impl("cg_xr_norm2", cg_xr_norm2_hip);
  m.impl("dia_spmv_bpdot", dia_spmv_bpdot_hip);
  m.impl("dia_spmv", dia_spmv_plain_hip);
  m.impl("dia_spmv_dot", dia_spmv_dot_hip);
  m.impl("dia_residual", dia_residual_hip);
  m.impl("dia_jacobi", dia_jacobi_hip);
  m.impl("spmm", spmm_hip);
  m.impl("rspmm", rspmm_hip);
  m.impl("bsr_spmm", bsr_spmm_hip);
  m.impl("sddmm", sddmm_hip);
  m.impl("csr_to_dense", csr_to_dense_hip);
  m.impl("coo_to_csr", coo_to_csr_hip);
  m.impl("dense_to_csr", dense_to_csr_hip);
  m.impl("csr_diagonal", csr_diagonal_hip);
  m.impl("csc_spmv", csc_spmv_hip);
  m.impl("csc_spmm", csc_spmm_hip);
  m.impl("tropical_spmv", tropical_spmv_hip);
  m.impl("rk_calc_dy", rk_calc_dy_hip);
  m.impl("cdist", cdist_hip);
  m.impl("spgemm_nnz", spgemm_nnz_hip);
  m.impl("spgemm_compute", spgemm_compute_hip);
}
