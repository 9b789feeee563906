// Native graph machinery (CPU, OpenMP): CSR transpose, row sorting,
// degree counting. Replaces the numpy paths for the 10^8-edge graphs
// (counting sort is O(E) vs argsort's O(E log E)); the reference's
// equivalents are the C++ loader/partition code in `gnn.cc:751-872`.
#include <torch/extension.h>

#include <atomic>
#include <cstring>
#include <vector>

#ifdef _OPENMP
#include <omp.h>
#endif

// transpose of a (possibly rectangular) CSR: rows -> num_cols rows.
std::vector<torch::Tensor> csr_transpose(int64_t num_cols,
                                         torch::Tensor rowptr,
                                         torch::Tensor colidx) {
  TORCH_CHECK(rowptr.device().is_cpu() && colidx.device().is_cpu());
  TORCH_CHECK(rowptr.scalar_type() == torch::kInt64);
  TORCH_CHECK(colidx.scalar_type() == torch::kInt32);
  rowptr = rowptr.contiguous();
  colidx = colidx.contiguous();
  const int64_t nr = rowptr.numel() - 1;
  const int64_t ne = colidx.numel();
  const int64_t* rp = rowptr.data_ptr<int64_t>();
  const int* ci = colidx.data_ptr<int>();

  auto t_rowptr = torch::zeros({num_cols + 1}, torch::kInt64);
  auto t_colidx = torch::empty({ne}, torch::kInt32);
  int64_t* trp = t_rowptr.data_ptr<int64_t>();
  int* tci = t_colidx.data_ptr<int>();

  // 1) count in-degree of each column
  {
    std::vector<std::atomic<int64_t>> cnt(num_cols);
    for (int64_t i = 0; i < num_cols; ++i)
      cnt[i].store(0, std::memory_order_relaxed);
#pragma omp parallel for schedule(static)
    for (int64_t e = 0; e < ne; ++e)
      cnt[ci[e]].fetch_add(1, std::memory_order_relaxed);
    int64_t run = 0;
    for (int64_t c = 0; c < num_cols; ++c) {
      trp[c] = run;
      run += cnt[c].load(std::memory_order_relaxed);
    }
    trp[num_cols] = run;
  }
  // 2) scatter destinations. Serial pass: keeps the result DETERMINISTIC
  //    (stable by source row within each column), which keeps reduction
  //    order — and therefore training — reproducible across runs.
  {
    std::vector<int64_t> cursor(num_cols);
    std::memcpy(cursor.data(), trp, num_cols * sizeof(int64_t));
    for (int64_t v = 0; v < nr; ++v) {
      for (int64_t e = rp[v]; e < rp[v + 1]; ++e)
        tci[cursor[ci[e]]++] = (int)v;
    }
  }
  return {t_rowptr, t_colidx};
}

// sort column ids within each row (in place)
void csr_sort_rows(torch::Tensor rowptr, torch::Tensor colidx) {
  TORCH_CHECK(rowptr.device().is_cpu() && colidx.device().is_cpu());
  const int64_t nr = rowptr.numel() - 1;
  const int64_t* rp = rowptr.data_ptr<int64_t>();
  int* ci = colidx.data_ptr<int>();
#pragma omp parallel for schedule(dynamic, 1024)
  for (int64_t v = 0; v < nr; ++v) std::sort(ci + rp[v], ci + rp[v + 1]);
}

void register_graph_cpu(pybind11::module_& m) {
  m.def("csr_transpose", &csr_transpose,
        "counting-sort CSR transpose (CPU, OpenMP)");
  m.def("csr_sort_rows", &csr_sort_rows, "sort column ids within rows");
}
