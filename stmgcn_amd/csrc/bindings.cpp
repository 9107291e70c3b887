// Python bindings + host-side sequencing for the stmgcn_amd HIP kernels.
// Device code lives in the .hip translation units (pure HIP, no torch);
// this file owns at::Tensor plumbing, stream lookup and launch ordering.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

extern "C" void stmgcn_spmm_step(
    void* stream, int dtype, const int* rowptr, const int* colidx,
    const float* vals, const void* xin, const void* p1, const void* p2,
    void* out, int B, int N, int C,
    long sx, long s1, long s2, long so,
    long bx, long b1, long b2, long bo,
    float alpha, float beta, float gamma);

namespace {

int dtype_code(const at::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kFloat: return 0;
    case at::kBFloat16: return 1;
    case at::kHalf: return 2;
    default: TORCH_CHECK(false, "unsupported dtype for stmgcn kernels");
  }
}

void* stream() { return (void*)at::cuda::getCurrentCUDAStream().stream(); }

struct Slice {
  const void* ptr;
  long srow, sbatch;
};

// one recurrence step: out_slice = alpha * G @ x_slice + beta*p1 + gamma*p2
void step(int dt, const at::Tensor& rowptr, const at::Tensor& colidx,
          const at::Tensor& vals, Slice xin, Slice p1, Slice p2,
          void* out, long so, long bo, int B, int N, int C,
          float alpha, float beta, float gamma) {
  stmgcn_spmm_step(stream(), dt, rowptr.data_ptr<int>(), colidx.data_ptr<int>(),
                   vals.data_ptr<float>(), xin.ptr, p1.ptr, p2.ptr, out,
                   B, N, C, xin.srow, p1.srow, p2.srow, so,
                   xin.sbatch, p1.sbatch, p2.sbatch, bo, alpha, beta, gamma);
}

}  // namespace

// x: (B, N, C) -> S: (B, N, K_s, C); slice k of S is T_k(G) @ x.
// kind single (localpool): K_s == 1, S[...,0,:] = G @ x.
at::Tensor cheb_apply(at::Tensor x, at::Tensor rowptr, at::Tensor colidx,
                      at::Tensor vals, int64_t K_s, bool single) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 3 && x.is_contiguous());
  TORCH_CHECK(rowptr.scalar_type() == at::kInt && colidx.scalar_type() == at::kInt);
  TORCH_CHECK(vals.scalar_type() == at::kFloat);
  const int B = x.size(0), N = x.size(1), C = x.size(2);
  const int dt = dtype_code(x);
  auto S = at::empty({B, N, K_s, C}, x.options());
  const long es = x.element_size();
  const char* xp = (const char*)x.data_ptr();
  char* Sp = (char*)S.data_ptr();
  const long sx = C, bx = (long)N * C;              // x row/batch strides (elems)
  const long so = K_s * (long)C, bo = (long)N * K_s * C;  // S row/batch strides
  auto sl = [&](int k) -> Slice { return {Sp + (long)k * C * es, so, bo}; };
  const Slice none{nullptr, 0, 0};
  const Slice xs{xp, sx, bx};

  if (single) {
    TORCH_CHECK(K_s == 1);
    step(dt, rowptr, colidx, vals, xs, none, none, Sp, so, bo, B, N, C, 1.f, 0.f, 0.f);
    return S;
  }
  // T_0 x = x (copy into slice 0)
  step(dt, rowptr, colidx, vals, none, none, xs, Sp, so, bo, B, N, C, 0.f, 0.f, 1.f);
  if (K_s > 1)  // T_1 x = G x
    step(dt, rowptr, colidx, vals, xs, none, none, (char*)sl(1).ptr, so, bo,
         B, N, C, 1.f, 0.f, 0.f);
  for (int k = 2; k < K_s; ++k)  // T_k x = 2 G (T_{k-1} x) - T_{k-2} x
    step(dt, rowptr, colidx, vals, sl(k - 1), sl(k - 2), none,
         (char*)sl(k).ptr, so, bo, B, N, C, 2.f, -1.f, 0.f);
  return S;
}

// U: (B, N, K_s, C) -> Z = sum_k T_k(G) @ U_k : (B, N, C), via the Clenshaw
// reverse recurrence (b_j = U_j + 2 G b_{j+1} - b_{j+2}; Z = U_0 + G b_1 - b_2).
// Pass the CSR of G^T here to get the gradient sum_k T_k(G)^T U_k.
at::Tensor cheb_combine(at::Tensor U, at::Tensor rowptr, at::Tensor colidx,
                        at::Tensor vals, bool single) {
  TORCH_CHECK(U.is_cuda() && U.dim() == 4 && U.is_contiguous());
  const int B = U.size(0), N = U.size(1), K_s = U.size(2), C = U.size(3);
  const int dt = dtype_code(U);
  auto Z = at::empty({B, N, C}, U.options());
  const long es = U.element_size();
  const char* Up = (const char*)U.data_ptr();
  const long su = (long)K_s * C, bu = (long)N * K_s * C;
  const long sz = C, bz = (long)N * C;
  auto ul = [&](int k) -> Slice { return {Up + (long)k * C * es, su, bu}; };
  const Slice none{nullptr, 0, 0};

  if (single || K_s == 1) {
    // Z = G^T U_0 (single) or T_0-only cheby (Z = U_0 — alpha 0 copy)
    if (single)
      step(dt, rowptr, colidx, vals, ul(0), none, none, Z.data_ptr(), sz, bz,
           B, N, C, 1.f, 0.f, 0.f);
    else
      step(dt, rowptr, colidx, vals, none, none, ul(0), Z.data_ptr(), sz, bz,
           B, N, C, 0.f, 0.f, 1.f);
    return Z;
  }

  const int K = K_s - 1;
  if (K == 1) {  // Z = U_0 + G U_1
    step(dt, rowptr, colidx, vals, ul(1), none, ul(0), Z.data_ptr(), sz, bz,
         B, N, C, 1.f, 0.f, 1.f);
    return Z;
  }
  auto bk1 = at::empty({B, N, C}, U.options());   // b_{j+1}
  auto bk2 = at::zeros({B, N, C}, U.options());   // b_{j+2}
  // b_K = U_K  (copy)
  step(dt, rowptr, colidx, vals, none, none, ul(K), bk1.data_ptr(), sz, bz,
       B, N, C, 0.f, 0.f, 1.f);
  Slice sb1{bk1.data_ptr(), sz, bz}, sb2{bk2.data_ptr(), sz, bz};
  for (int j = K - 1; j >= 1; --j) {
    auto bnew = at::empty({B, N, C}, U.options());
    // b_j = 2 G b_{j+1} - b_{j+2} + U_j   (p1 = b_{j+2}, p2 = U_j)
    step(dt, rowptr, colidx, vals, sb1, sb2, ul(j), bnew.data_ptr(), sz, bz,
         B, N, C, 2.f, -1.f, 1.f);
    bk2 = bk1; bk1 = bnew;
    sb1 = {bk1.data_ptr(), sz, bz}; sb2 = {bk2.data_ptr(), sz, bz};
  }
  // Z = G b_1 - b_2 + U_0
  step(dt, rowptr, colidx, vals, sb1, sb2, ul(0), Z.data_ptr(), sz, bz,
       B, N, C, 1.f, -1.f, 1.f);
  return Z;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("cheb_apply", &cheb_apply,
        "Support stack S[b,n,k,c] = (T_k(G) x)[b,n,c] via in-kernel recurrence");
  m.def("cheb_combine", &cheb_combine,
        "Z = sum_k T_k(G) U_k via Clenshaw (pass G^T CSR for gradients)");
}
