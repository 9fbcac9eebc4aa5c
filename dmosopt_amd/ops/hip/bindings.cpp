// PyTorch bindings for the dmosopt_amd gfx950 kernels.

#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>
#include <torch/extension.h>

#include <vector>

#define CHECK_GPU(x) \
  TORCH_CHECK(x.is_cuda(), #x " must be a GPU tensor"); \
  TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")

extern "C" {
void launch_matern_assemble_affine(const float*, const float*, const float*, float*, int, int, int, int, int, float, int, int, int, const float*, const float*, hipStream_t);
void launch_gp_mean_gemv(const float*, const float*, const float*, const float*, float*, int, int, int, hipStream_t);
void launch_matern_assemble(const float*, const float*, const float*, float*,
                            int, int, int, int, int, float, int, int, int,
                            hipStream_t);
void launch_cholesky_batched(float*, float*, int*, int, int, hipStream_t);
int launch_cholesky_fused_solve(float*, float*, int*, float*, int*, int,
                                int, hipStream_t);
void launch_forward_solve_batched(const float*, float*, int, int, int,
                                  hipStream_t);
void launch_backward_solve_batched(const float*, float*, int, int, int,
                                   hipStream_t);
void launch_dominance_matrix(const float*, int*, int, int, hipStream_t);
void launch_peel_front(const int*, const unsigned char*, unsigned char*, int*,
                       int, int, hipStream_t);
void launch_commit_front(const unsigned char*, unsigned char*, int*, int, int,
                         hipStream_t);
void launch_crowding(const float*, float*, int, int, hipStream_t);
void launch_sbx_batch(const float*, const int*, const int*, const float*,
                      const float*, const float*, float*, int, int,
                      unsigned long long, hipStream_t);
void launch_mutation_batch(const float*, const int*, const float*,
                           const float*, const float*, float*, int, int, float,
                           unsigned long long, hipStream_t);
void launch_hv_mc_uniform(const float*, const float*, const float*,
                          unsigned long long*, long long, int, int,
                          unsigned long long, hipStream_t);
void launch_hv_fpras(const float*, const float*, const float*,
                     unsigned long long*, long long, int, int,
                     unsigned long long, hipStream_t);
void launch_cmaes_update(float*, float*, float*, const float*, const float*,
                         int, int, float, float, float, hipStream_t);
int launch_peel_from_y(const float*, int*, int, int, hipStream_t);
void launch_nmll_reduce(const float*, const float*, const int*, float*, int, int, float, hipStream_t);
void launch_sceua_propose(const float*, const int*, const float*, const float*, float*, int, int, int, int, int, unsigned long long, hipStream_t);
int launch_sceua_accept(float*, float*, const float*, const float*, const int*, const int*, int*, int, int, int, int, int, hipStream_t);
int launch_tournament(const float*, const long long*, float*, long long*, int, int, int, float, unsigned long long, hipStream_t);
void launch_tournament_keys(float*, int, float, unsigned long long, hipStream_t);
int launch_survivor_count(const long long*, const long long*, int, int, int, long long*, long long*, hipStream_t);
void launch_pack_rank_crowd(const long long*, const float*, long long*, int, hipStream_t);
int launch_rank_crowd_sort(const long long*, const float*, long long*, int,
                           int, hipStream_t);
void launch_variation_slots(const float*, const long long*, const long long*, const long long*, const long long*, const float*, const float*, const float*, const float*, float*, int, int, int, float, unsigned long long, unsigned long long, hipStream_t);
void launch_variation_events(const float*, const long long*, const long long*, const long long*, const long long*, const long long*, const float*, const float*, const float*, const float*, float*, int, int, int, float, unsigned long long, unsigned long long, hipStream_t);
void launch_gather3(const float*, const float*, const long long*, const long long*, float*, float*, long long*, int, int, int, hipStream_t);
int launch_peel_bits(const float*, unsigned int*, int*, int, int, hipStream_t);
int launch_coop_peel(const float*, unsigned int*, unsigned int*, int*, int*,
                     int*, int, int, int, hipStream_t);
void launch_peel_single_block(const int*, int*, int, int, hipStream_t);
void launch_hv2d(const double*, const double*, double*, int, hipStream_t);
void launch_hv3d_slices(const double*, const double*, const double*,
                        const double*, double*, int, hipStream_t);
void launch_ehvi(const double*, const double*, const double*, const double*,
                 double*, int, int, int, hipStream_t);
void launch_lacour_flags(const double*, const long long*, const double*,
                         const double*, unsigned char*, unsigned char*, int,
                         int, hipStream_t);
void launch_lacour_scatter(const double*, const long long*, const double*,
                           const unsigned char*, const unsigned char*,
                           const long long*, const long long*,
                           const long long*, long long, long long, double*,
                           long long*, int, int, hipStream_t);
void launch_lacour_volumes(const double*, const long long*, const double*,
                           const double*, double*, int, int, hipStream_t);
void launch_mfma_bf16_probe(const float*, const float*, float*, hipStream_t);
void launch_matern_cross_bf16(const float*, const float*, const float*,
                              float*, int, int, int, int, int, int, int,
                              const float*, const float*, hipStream_t);
void launch_cholesky_multik_bf16(float*, float*, int*, int, int, hipStream_t);
void launch_agemoea_survival(const float*, const unsigned char*, float*, int,
                             hipStream_t);
void launch_minkowski_norm_matrix(const float*, float*, int, int, float,
                                  hipStream_t);
void launch_cat_rows(const float*, const float*, float*, long long,
                     long long, hipStream_t);
void launch_colsum(const float*, float*, int, int, hipStream_t);
void launch_smpso_velocity(const float*, const float*, const float*,
                           const float*, const float*, const float*, float*,
                           int, int, float, float, float, float, hipStream_t);
}

static hipStream_t cur_stream() {
  return (hipStream_t)c10::hip::getCurrentHIPStream().stream();
}

static int nu_code(double nu) {
  if (nu == 0.5) return 1;
  if (nu == 1.5) return 3;
  if (nu == 2.5) return 5;
  return 0;  // RBF / inf
}

torch::Tensor matern_train(torch::Tensor X, torch::Tensor theta, double nu,
                           bool aniso, double jitter) {
  CHECK_GPU(X);
  CHECK_GPU(theta);
  const int N = X.size(0), D = X.size(1), B = theta.size(0);
  auto K = torch::empty({B, N, N}, X.options());
  launch_matern_assemble(X.data_ptr<float>(), X.data_ptr<float>(),
                         theta.data_ptr<float>(), K.data_ptr<float>(), B, N, N,
                         D, theta.size(1), (float)jitter, nu_code(nu),
                         aniso ? 1 : 0, 1, cur_stream());
  return K;
}

torch::Tensor matern_cross(torch::Tensor Xq, torch::Tensor X,
                           torch::Tensor theta, double nu, bool aniso) {
  CHECK_GPU(Xq);
  CHECK_GPU(X);
  CHECK_GPU(theta);
  const int P = Xq.size(0), N = X.size(0), D = X.size(1), B = theta.size(0);
  auto K = torch::empty({B, P, N}, X.options());
  launch_matern_assemble(Xq.data_ptr<float>(), X.data_ptr<float>(),
                         theta.data_ptr<float>(), K.data_ptr<float>(), B, P, N,
                         D, theta.size(1), 0.f, nu_code(nu), aniso ? 1 : 0, 0,
                         cur_stream());
  return K;
}

// Fused GP negative log marginal likelihood for a batch of theta: kernel
// assembly -> batched Cholesky -> forward solve -> reduction, all queued
// from ONE python call (the per-stage torch dispatch chain was ~40% of the
// SCE-UA fit's host time). y: (N,) shared or (B, N) per-candidate.
torch::Tensor gp_nmll(torch::Tensor X, torch::Tensor theta, torch::Tensor y,
                      double nu, bool aniso, double jitter) {
  CHECK_GPU(X);
  CHECK_GPU(theta);
  CHECK_GPU(y);
  const int N = X.size(0), D = X.size(1), B = theta.size(0);
  auto K = torch::empty({B, N, N}, X.options());
  launch_matern_assemble(X.data_ptr<float>(), X.data_ptr<float>(),
                         theta.data_ptr<float>(), K.data_ptr<float>(), B, N, N,
                         D, theta.size(1), (float)jitter, nu_code(nu),
                         aniso ? 1 : 0, 1, cur_stream());
  auto logdet = torch::empty({B}, X.options());
  auto info = torch::zeros({B}, X.options().dtype(torch::kInt32));
  torch::Tensor Z = (y.dim() == 1)
                        ? y.unsqueeze(0).expand({B, N}).contiguous()
                        : y.contiguous().clone();
  Z = Z.view({B, N, 1});
  // fused factor+solve: the rhs rides the multik launch chain (panel solves
  // its 32-entry segment, the SYRK's diagonal tiles apply the trailing
  // update) — no separate ~61 us serial TRSV kernel per NMLL
  auto ws = torch::empty({2 * B}, X.options().dtype(torch::kInt32));
  if (launch_cholesky_fused_solve(K.data_ptr<float>(),
                                  logdet.data_ptr<float>(),
                                  info.data_ptr<int>(), Z.data_ptr<float>(),
                                  ws.data_ptr<int>(), B, N,
                                  cur_stream()) != 0) {
    launch_cholesky_batched(K.data_ptr<float>(), logdet.data_ptr<float>(),
                            info.data_ptr<int>(), B, N, cur_stream());
    launch_forward_solve_batched(K.data_ptr<float>(), Z.data_ptr<float>(), B,
                                 N, 1, cur_stream());
  }
  auto out = torch::empty({B}, X.options());
  const float c = 0.5f * (float)N * 1.8378770664093453f;  // log(2*pi)
  launch_nmll_reduce(Z.data_ptr<float>(), logdet.data_ptr<float>(),
                     info.data_ptr<int>(), out.data_ptr<float>(), B, N, c,
                     cur_stream());
  return out;
}

// Fused posterior-mean predict for the inner-MOEA surrogate evaluate:
// cross-kernel assembly + bmm(alpha) + de-standardization queued from one
// python call (the torch op chain costs ~0.1 ms of dispatch per
// generation). Xq must already be normalized to the unit box.
torch::Tensor gp_predict_mean(torch::Tensor Xq, torch::Tensor X,
                              torch::Tensor theta, torch::Tensor alpha,
                              torch::Tensor y_mean, torch::Tensor y_std,
                              double nu, bool aniso,
                              c10::optional<torch::Tensor> q_lb = c10::nullopt,
                              c10::optional<torch::Tensor> q_invrg = c10::nullopt) {
  CHECK_GPU(Xq);
  CHECK_GPU(X);
  const int P = Xq.size(0), N = X.size(0), D = X.size(1), B = theta.size(0);
  auto Ks = torch::empty({B, P, N}, X.options());
  // optional per-dim affine normalizes RAW queries inside the kernel load
  launch_matern_assemble_affine(
      Xq.data_ptr<float>(), X.data_ptr<float>(), theta.data_ptr<float>(),
      Ks.data_ptr<float>(), B, P, N, D, theta.size(1), 0.0f, nu_code(nu),
      aniso ? 1 : 0, 0, q_lb ? q_lb->data_ptr<float>() : nullptr,
      q_invrg ? q_invrg->data_ptr<float>() : nullptr, cur_stream());
  // fused matvec + de-standardization into the contiguous (P, B) result:
  // one kernel instead of bmm + addcmul + transpose-contiguous
  auto out = torch::empty({P, (long)B}, X.options());
  launch_gp_mean_gemv(Ks.data_ptr<float>(), alpha.data_ptr<float>(),
                      y_mean.data_ptr<float>(), y_std.data_ptr<float>(),
                      out.data_ptr<float>(), B, P, N, cur_stream());
  return out;
}

// Fused SCE-UA CCE stage: candidate proposal and acceptance+resort, one
// extension call each instead of ~30 torch kernels per stage.
// Fused tournament selection: stable rank sort + Gumbel top-k weighted
// sampling without replacement + pool gather in ONE launch. Returns
// (pool, pool_idx); empty tensors when N exceeds the LDS sort bound.
// Multi-block tournament (N > the one-workgroup bitonic's sweet spot):
// Gumbel keys from the SAME Philox stream as tournament_kernel stage 2,
// sorted with the radix argsort. rank_sorted=true skips the stage-1
// stable rank sort (nsga2_select emits rank-sorted populations).
static std::vector<torch::Tensor> tournament_torch(torch::Tensor population,
                                                   torch::Tensor rank,
                                                   int64_t poolsize,
                                                   float log1mp, int64_t seed,
                                                   bool rank_sorted) {
  const int N = population.size(0);
  auto gk = torch::empty({N}, population.options());
  launch_tournament_keys(gk.data_ptr<float>(), N, log1mp,
                         (unsigned long long)seed, cur_stream());
  auto top = torch::argsort(gk, /*dim=*/-1, /*descending=*/true)
                 .slice(0, 0, poolsize)
                 .contiguous();
  torch::Tensor sel = top;
  if (!rank_sorted) {
    auto order = torch::argsort(rank, /*stable=*/true, /*dim=*/-1,
                                /*descending=*/false);
    sel = order.index_select(0, top).contiguous();
  }
  return {population.index_select(0, sel), sel};
}

std::vector<torch::Tensor> tournament_pool(torch::Tensor population,
                                           torch::Tensor rank,
                                           int64_t poolsize, double p_sel,
                                           int64_t seed) {
  CHECK_GPU(population);
  TORCH_CHECK(population.dtype() == torch::kFloat32 &&
                  rank.dtype() == torch::kLong,
              "tournament_pool: (float32 population, int64 rank) required");
  TORCH_CHECK(rank.size(0) == population.size(0) &&
                  poolsize <= population.size(0),
              "tournament_pool: shape mismatch");
  const int N = population.size(0), d = population.size(1);
  auto pool = torch::empty({poolsize, d}, population.options());
  auto pool_idx =
      torch::empty({poolsize}, population.options().dtype(torch::kLong));
  const float log1mp = logf(1.0f - (float)p_sel);
  if (launch_tournament(population.data_ptr<float>(),
                        (long long*)rank.data_ptr<int64_t>(), pool.data_ptr<float>(),
                        (long long*)pool_idx.data_ptr<int64_t>(), N, d, (int)poolsize,
                        log1mp, (unsigned long long)seed, cur_stream()) != 0)
    return tournament_torch(population, rank, poolsize, log1mp, seed, false);
  return {pool, pool_idx};
}

// Device-side operator-success accounting (no host sync, one launch).
bool survivor_count(torch::Tensor perm, torch::Tensor c_idx,
                    int64_t n_children, torch::Tensor succ_cross,
                    torch::Tensor succ_mut) {
  CHECK_GPU(perm);
  TORCH_CHECK(perm.dtype() == torch::kLong && c_idx.dtype() == torch::kLong &&
                  succ_cross.dtype() == torch::kLong &&
                  succ_mut.dtype() == torch::kLong,
              "survivor_count: int64 tensors required");
  return launch_survivor_count(
             (long long*)perm.data_ptr<int64_t>(), (long long*)c_idx.data_ptr<int64_t>(),
             perm.size(0), c_idx.size(0), (int)n_children,
             (long long*)succ_cross.data_ptr<int64_t>(), (long long*)succ_mut.data_ptr<int64_t>(),
             cur_stream()) == 0;
}

// Whole-generation variation in one launch (see variation.hip).
torch::Tensor variation_slots(torch::Tensor pool, torch::Tensor src_rows,
                              torch::Tensor p1, torch::Tensor p2,
                              torch::Tensor im, torch::Tensor di_c,
                              torch::Tensor di_m, torch::Tensor lo,
                              torch::Tensor hi, double mutation_rate,
                              int64_t C, int64_t seed_sbx, int64_t seed_mut) {
  CHECK_GPU(pool);
  TORCH_CHECK(pool.dtype() == torch::kFloat32 &&
                  src_rows.dtype() == torch::kLong &&
                  p1.size(0) == C && p2.size(0) == C,
              "variation_slots: f32 pool, int64 indices, |p1|=|p2|=C");
  const int total = src_rows.size(0), d = pool.size(1);
  auto out = torch::empty({total, d}, pool.options());
  launch_variation_slots(
      pool.data_ptr<float>(), (long long*)src_rows.data_ptr<int64_t>(),
      (long long*)p1.data_ptr<int64_t>(), (long long*)p2.data_ptr<int64_t>(),
      (long long*)im.data_ptr<int64_t>(), di_c.data_ptr<float>(),
      di_m.data_ptr<float>(), lo.data_ptr<float>(), hi.data_ptr<float>(),
      out.data_ptr<float>(), total, (int)C, d, (float)mutation_rate,
      (unsigned long long)seed_sbx, (unsigned long long)seed_mut,
      cur_stream());
  return out;
}

// Event-decoded whole-generation variation (variation_events_kernel):
// same child bits as variation_slots, but addressed by the per-event slot
// lists so the host never materializes the src_rows inverse map.
torch::Tensor variation_events(torch::Tensor pool, torch::Tensor ci,
                               torch::Tensor mi, torch::Tensor p1,
                               torch::Tensor p2, torch::Tensor im,
                               torch::Tensor di_c, torch::Tensor di_m,
                               torch::Tensor lo, torch::Tensor hi,
                               double mutation_rate, int64_t seed_sbx,
                               int64_t seed_mut) {
  CHECK_GPU(pool);
  const int C = p1.size(0), M = im.size(0), d = pool.size(1);
  TORCH_CHECK(pool.dtype() == torch::kFloat32 && ci.dtype() == torch::kLong &&
                  mi.dtype() == torch::kLong && p2.size(0) == C &&
                  ci.size(0) == 2 * (int64_t)C && mi.size(0) == M,
              "variation_events: f32 pool, int64 slot/parent indices");
  auto out = torch::empty({(long)(2 * C + M), d}, pool.options());
  launch_variation_events(
      pool.data_ptr<float>(),
      C ? (long long*)ci.data_ptr<int64_t>() : nullptr,
      M ? (long long*)mi.data_ptr<int64_t>() : nullptr,
      C ? (long long*)p1.data_ptr<int64_t>() : nullptr,
      C ? (long long*)p2.data_ptr<int64_t>() : nullptr,
      M ? (long long*)im.data_ptr<int64_t>() : nullptr,
      di_c.data_ptr<float>(), di_m.data_ptr<float>(), lo.data_ptr<float>(),
      hi.data_ptr<float>(), out.data_ptr<float>(), C, M, d,
      (float)mutation_rate, (unsigned long long)seed_sbx,
      (unsigned long long)seed_mut, cur_stream());
  return out;
}

torch::Tensor sceua_propose(torch::Tensor cx, torch::Tensor lcs,
                            torch::Tensor bl, torch::Tensor bu, int64_t nps,
                            int64_t seed) {
  CHECK_GPU(cx);
  TORCH_CHECK(cx.dtype() == torch::kFloat32 && cx.dim() == 4 &&
                  lcs.dtype() == torch::kInt32,
              "sceua_propose: cx (S,G,npg,nopt) f32 + int32 lcs required");
  const int S = cx.size(0), G = cx.size(1), npg = cx.size(2),
            nopt = cx.size(3);
  auto cand = torch::empty({3, (long)S * G, nopt}, cx.options());
  launch_sceua_propose(cx.data_ptr<float>(), lcs.data_ptr<int>(),
                       bl.data_ptr<float>(), bu.data_ptr<float>(),
                       cand.data_ptr<float>(), S, G, npg, nopt, (int)nps,
                       (unsigned long long)seed, cur_stream());
  return cand;
}

bool sceua_accept(torch::Tensor cx, torch::Tensor cf, torch::Tensor cand,
                  torch::Tensor fall, torch::Tensor lcs, torch::Tensor act,
                  torch::Tensor icall, int64_t nps) {
  CHECK_GPU(cx);
  const int S = cx.size(0), G = cx.size(1), npg = cx.size(2),
            nopt = cx.size(3);
  return launch_sceua_accept(cx.data_ptr<float>(), cf.data_ptr<float>(),
                             cand.data_ptr<float>(), fall.data_ptr<float>(),
                             lcs.data_ptr<int>(), act.data_ptr<int>(),
                             icall.data_ptr<int>(), S, G, npg, nopt, (int)nps,
                             cur_stream()) == 0;
}

std::vector<torch::Tensor> cholesky_batched_(torch::Tensor A) {
  CHECK_GPU(A);
  const int B = A.size(0), N = A.size(1);
  auto logdet = torch::empty({B}, A.options());
  auto info = torch::zeros({B}, A.options().dtype(torch::kInt32));
  launch_cholesky_batched(A.data_ptr<float>(), logdet.data_ptr<float>(),
                          info.data_ptr<int>(), B, N, cur_stream());
  return {logdet, info};
}

void forward_solve_(torch::Tensor L, torch::Tensor Y) {
  CHECK_GPU(L);
  CHECK_GPU(Y);
  launch_forward_solve_batched(L.data_ptr<float>(), Y.data_ptr<float>(),
                               L.size(0), L.size(1), Y.size(2), cur_stream());
}

void backward_solve_(torch::Tensor L, torch::Tensor Y) {
  CHECK_GPU(L);
  CHECK_GPU(Y);
  launch_backward_solve_batched(L.data_ptr<float>(), Y.data_ptr<float>(),
                                L.size(0), L.size(1), Y.size(2), cur_stream());
}

torch::Tensor dominance_degree_matrix(torch::Tensor Y) {
  CHECK_GPU(Y);
  const int N = Y.size(0), m = Y.size(1);
  auto D = torch::empty({N, N}, Y.options().dtype(torch::kInt32));
  launch_dominance_matrix(Y.data_ptr<float>(), D.data_ptr<int>(), N, m,
                          cur_stream());
  return D;
}

static int COOP_MIN_N = []() {
  const char* e = getenv("DMOSOPT_COOP_MIN_N");
  return e ? atoi(e) : 1024;
}();

torch::Tensor pareto_rank(torch::Tensor Y, int64_t stop = -1) {
  // stop > 0: truncation-selection mode — fronts are peeled only until
  // `stop` points are ranked (the straddling front always completes), the
  // rest receive a sentinel rank larger than every true one. Exact for
  // nsga2_select, which discards everything beyond the kept `pop`; the
  // public op default (-1) ranks everything. Only the cooperative path
  // exploits it — the small-N one-workgroup peels are already ~us-scale.
  CHECK_GPU(Y);
  const int N = Y.size(0), m = Y.size(1);
  // N > 2048: grid-wide COOPERATIVE peel — sync-free from the host (the
  // chased matvec path's readback every 16 fronts stalls pipelined
  // generation loops) and parallel across all CUs (the one-workgroup
  // peels serialize on one CU: 13.8 ms at N=8192 vs ~0.1-1 ms here)
  if (N > COOP_MIN_N) {
    const int W = (N + 31) / 32;
    auto Yc = Y.contiguous().to(torch::kFloat32);
    auto opts_i = Y.options().dtype(torch::kInt32);
    auto Dbits = torch::empty({N, W}, opts_i);
    auto fmask = torch::empty({3 * W}, opts_i);  // triple-buffered front mask
    auto n_dom = torch::empty({N}, opts_i);
    auto ctrl = torch::empty({2}, opts_i);
    // empty, not zeros: the peel kernels write every rank entry (the fill
    // kernel cost ~24 us/call at N=3200)
    auto rank = torch::empty({N}, opts_i);
    if (launch_coop_peel(Yc.data_ptr<float>(),
                         (unsigned int*)Dbits.data_ptr<int>(),
                         (unsigned int*)fmask.data_ptr<int>(),
                         n_dom.data_ptr<int>(), ctrl.data_ptr<int>(),
                         rank.data_ptr<int>(), N, m, (int)stop,
                         cur_stream()) == 0)
      return rank.to(torch::kLong);
  }
  if (N <= 2048) {
    auto Yc = Y.contiguous().to(torch::kFloat32);
    auto rank = torch::zeros({N}, Y.options().dtype(torch::kInt32));
    // bit-matrix path: grid-wide packed dominator build + popcount peel
    const int W = (N + 31) / 32;
    const size_t bits_lds = ((size_t)N * W + W) * sizeof(unsigned int) +
                            (N + 2) * sizeof(int);
    if (bits_lds <= 144 * 1024) {
      auto Dbits = torch::empty({N, W}, Y.options().dtype(torch::kInt32));
      if (launch_peel_bits(Yc.data_ptr<float>(),
                           (unsigned int*)Dbits.data_ptr<int>(),
                           rank.data_ptr<int>(), N, m, cur_stream()) == 0)
        return rank.to(torch::kLong);
    }
    // fallback: one-launch peel recomputing dominance from Y in LDS
    if (launch_peel_from_y(Yc.data_ptr<float>(), rank.data_ptr<int>(), N, m,
                           cur_stream()) == 0)
      return rank.to(torch::kLong);
    auto D = dominance_degree_matrix(Y);
    launch_peel_single_block(D.data_ptr<int>(), rank.data_ptr<int>(), N, m,
                             cur_stream());
    return rank.to(torch::kLong);
  }
  auto D = dominance_degree_matrix(Y);
  auto alive = torch::ones({N}, Y.options().dtype(torch::kUInt8));
  auto front = torch::empty({N}, Y.options().dtype(torch::kUInt8));
  auto rank = torch::zeros({N}, Y.options().dtype(torch::kInt32));
  auto n_front = torch::zeros({1}, Y.options().dtype(torch::kInt32));
  // Peel in chases of CHASE fronts between host syncs: front k+1 depends
  // only on device state from commit k, so the launches chain on-stream and
  // one n_front readback per chase amortizes the sync. Extra launches after
  // exhaustion are no-ops (alive all zero).
  constexpr int CHASE = 16;
  int remaining = N;
  int k = 0;
  while (remaining > 0 && k < N + CHASE) {
    n_front.zero_();
    for (int c = 0; c < CHASE; ++c) {
      launch_peel_front(D.data_ptr<int>(), alive.data_ptr<unsigned char>(),
                        front.data_ptr<unsigned char>(),
                        n_front.data_ptr<int>(), N, m, cur_stream());
      launch_commit_front(front.data_ptr<unsigned char>(),
                          alive.data_ptr<unsigned char>(),
                          rank.data_ptr<int>(), k + c, N, cur_stream());
    }
    const int nf = n_front.item<int>();  // total over the chase; syncs
    if (nf == 0) break;                  // safety against stalls
    remaining -= nf;
    k += CHASE;
  }
  return rank.to(torch::kLong);
}

torch::Tensor crowding_distance(torch::Tensor Y) {
  CHECK_GPU(Y);
  const int N = Y.size(0), m = Y.size(1);
  if (N == 1) return torch::ones({1}, Y.options());
  static int torch_min_n = []() {
    // default DISABLED: the 10-dispatch torch chain measured 35 ms/epoch
    // SLOWER end-to-end at pop=1600 than the 208-us single-CU kernel it
    // replaced (same box) — dispatch cost beats kernel time in the
    // pipelined loop. Kept selectable for bigger-N re-measurement.
    const char* e = getenv("DMOSOPT_CROWD_TORCH_MIN_N");
    return e ? atoi(e) : (1 << 30);
  }();
  if (N > torch_min_n) {
    // the single-workgroup-per-dim LDS bitonic degrades ~N log^2 N on one
    // CU (208 us at N=3200, m=2); above the gate the multi-block radix
    // route wins (same arithmetic as ops/torch_ref.crowding_distance)
    auto lb = std::get<0>(Y.min(0, /*keepdim=*/true));
    auto ub = std::get<0>(Y.max(0, /*keepdim=*/true));
    auto span = (ub - lb).clamp_min(0.0);
    span = torch::where(span == 0, torch::ones_like(span), span);
    auto U = (Y - lb) / span;
    auto idx = torch::argsort(U, /*dim=*/0, /*descending=*/false);
    auto US = torch::gather(U, 0, idx);
    auto DS = torch::empty_like(US);
    DS.slice(0, 0, 1).fill_(1.0);
    DS.slice(0, N - 1, N).fill_(1.0);
    if (N > 2)
      DS.slice(0, 1, N - 1) = US.slice(0, 2, N) - US.slice(0, 0, N - 2);
    // per-column scatter (each column of idx is a permutation) + fixed-
    // order row sum — scatter_ADD on CUDA floats resolves atomics in
    // arbitrary order and would break run-to-run bit determinism
    auto Dm = torch::zeros({N, m}, Y.options());
    Dm.scatter_(0, idx, DS);
    return torch::nan_to_num(Dm.sum(1), 0.0);
  }
  TORCH_CHECK(N <= 16384, "crowding_distance HIP kernel supports N <= 16384");
  auto per_dim = torch::empty({m, N}, Y.options());
  launch_crowding(Y.data_ptr<float>(), per_dim.data_ptr<float>(), N, m,
                  cur_stream());
  // fixed-order column sum (deterministic); a raw kernel instead of
  // at::sum saves ~8 us of host dispatch per generation
  auto out = torch::empty({N}, Y.options());
  launch_colsum(per_dim.data_ptr<float>(), out.data_ptr<float>(), m, N,
                cur_stream());
  return out;
}

// Fused NSGA2 survivor selection: concatenate children+parents, pareto-rank,
// crowding, packed-key stable sort ((rank << 32) | ~float_bits(crowding)),
// truncate to pop and gather — one python call instead of ~15 dispatched
// ops per generation. Mirrors ops.remove_worst with the crowding metric.
std::vector<torch::Tensor> nsga2_select(torch::Tensor x_gen,
                                        torch::Tensor y_gen,
                                        torch::Tensor pop_parm,
                                        torch::Tensor pop_obj, int64_t pop) {
  CHECK_GPU(x_gen);
  TORCH_CHECK(x_gen.dtype() == torch::kFloat32 &&
                  pop_parm.dtype() == torch::kFloat32,
              "nsga2_select: float32 inputs required");
  TORCH_CHECK(x_gen.size(1) == pop_parm.size(1) &&
                  y_gen.size(1) == pop_obj.size(1) &&
                  x_gen.size(0) == y_gen.size(0) &&
                  pop_parm.size(0) == pop_obj.size(0),
              "nsga2_select: shape mismatch");
  const long long ng = x_gen.size(0), np_ = pop_parm.size(0);
  const long long dP = x_gen.size(1), dO = y_gen.size(1);
  auto parm = torch::empty({ng + np_, dP}, x_gen.options());
  auto obj = torch::empty({ng + np_, dO}, y_gen.options());
  launch_cat_rows(x_gen.data_ptr<float>(), pop_parm.data_ptr<float>(),
                  parm.data_ptr<float>(), ng * dP, (ng + np_) * dP,
                  cur_stream());
  launch_cat_rows(y_gen.data_ptr<float>(), pop_obj.data_ptr<float>(),
                  obj.data_ptr<float>(), ng * dO, (ng + np_) * dO,
                  cur_stream());
  const int64_t keep =
      std::min<int64_t>(pop, x_gen.size(0) + pop_parm.size(0));
  // truncation selection only consumes fronts up to the one straddling
  // `keep`; the early-stop peel skips the (typically ~half) deeper rounds
  auto rank = pareto_rank(obj, keep);              // (N,) long
  auto crowd = crowding_distance(obj).to(torch::kFloat32);  // (N,)
  const int N = obj.size(0), d = parm.size(1), m = obj.size(1);
  const int P0 = (int)std::min<int64_t>(pop, N);
  static int rcs_on = []() {
    const char* e = getenv("DMOSOPT_RCS");
    return e && e[0] == '0' ? 0 : 1;
  }();
  torch::Tensor perm;
  // single-block bitonic (key, idx) sort replaces pack + radix argsort
  // (3-4 launches -> 1); comparator (key asc, idx asc) == stable argsort.
  // Gated by N: the one-workgroup bitonic degrades ~N log^2 N on one CU
  // (122 us at N=3200) while the multi-block radix path stays ~40 us.
  static int rcs_max_n = []() {
    const char* e = getenv("DMOSOPT_RCS_MAX_N");
    return e ? atoi(e) : (1 << 30);
  }();
  auto perm_f = torch::empty({P0}, rank.options());
  if (rcs_on && N <= rcs_max_n &&
      launch_rank_crowd_sort((long long*)rank.data_ptr<int64_t>(),
                             crowd.data_ptr<float>(),
                             (long long*)perm_f.data_ptr<int64_t>(), N, P0,
                             cur_stream()) == 0) {
    perm = perm_f;
  } else {
    auto key = torch::empty({N}, rank.options());
    launch_pack_rank_crowd((long long*)rank.data_ptr<int64_t>(),
                           crowd.data_ptr<float>(),
                           (long long*)key.data_ptr<int64_t>(), N,
                           cur_stream());
    perm = torch::argsort(key, /*stable=*/true, /*dim=*/-1,
                          /*descending=*/false)
               .slice(0, 0, pop)
               .contiguous();
  }
  const int P = perm.size(0);
  auto parm_o = torch::empty({P, d}, parm.options());
  auto obj_o = torch::empty({P, m}, obj.options());
  auto rank_o = torch::empty({P}, rank.options());
  launch_gather3(parm.data_ptr<float>(), obj.data_ptr<float>(),
                 (long long*)rank.data_ptr<int64_t>(), (long long*)perm.data_ptr<int64_t>(),
                 parm_o.data_ptr<float>(), obj_o.data_ptr<float>(),
                 (long long*)rank_o.data_ptr<int64_t>(), P, d, m, cur_stream());
  return {parm_o, obj_o, rank_o, perm};
}

// nsga2_select + device-side operator-success accounting in ONE python
// call (the generation loop is host-dispatch-bound; every binding round
// trip costs ~10 us of host time)
std::vector<torch::Tensor> nsga2_select_acc(
    torch::Tensor x_gen, torch::Tensor y_gen, torch::Tensor pop_parm,
    torch::Tensor pop_obj, int64_t pop, torch::Tensor c_idx,
    torch::Tensor succ_cross, torch::Tensor succ_mut) {
  auto r = nsga2_select(x_gen, y_gen, pop_parm, pop_obj, pop);
  auto& perm = r[3];
  launch_survivor_count(
      (long long*)perm.data_ptr<int64_t>(),
      (long long*)c_idx.data_ptr<int64_t>(), perm.size(0), c_idx.size(0),
      (int)x_gen.size(0), (long long*)succ_cross.data_ptr<int64_t>(),
      (long long*)succ_mut.data_ptr<int64_t>(), cur_stream());
  return r;
}

// Tournament pool + event-decoded variation chained inside one binding
// call: the pool tensor never surfaces to python.
torch::Tensor generation_spawn(
    torch::Tensor population, torch::Tensor rank, int64_t poolsize,
    double p_sel, int64_t seed_t, torch::Tensor ci, torch::Tensor mi,
    torch::Tensor p1, torch::Tensor p2, torch::Tensor im, torch::Tensor di_c,
    torch::Tensor di_m, torch::Tensor lo, torch::Tensor hi,
    double mutation_rate, int64_t seed_sbx, int64_t seed_mut,
    bool rank_sorted) {
  CHECK_GPU(population);
  TORCH_CHECK(population.dtype() == torch::kFloat32 &&
                  rank.dtype() == torch::kLong &&
                  poolsize <= population.size(0),
              "generation_spawn: f32 population, int64 rank");
  const int N = population.size(0), d = population.size(1);
  auto pool = torch::empty({poolsize, d}, population.options());
  auto pool_idx =
      torch::empty({poolsize}, population.options().dtype(torch::kLong));
  const float log1mp = logf(1.0f - (float)p_sel);
  static int tour_torch_min_n = []() {
    const char* e = getenv("DMOSOPT_TOUR_TORCH_MIN_N");
    return e ? atoi(e) : (1 << 30);
  }();
  if (N > tour_torch_min_n) {
    auto r = tournament_torch(population, rank, poolsize, log1mp, seed_t,
                              rank_sorted);
    return variation_events(r[0].contiguous(), ci, mi, p1, p2, im, di_c,
                            di_m, lo, hi, mutation_rate, seed_sbx, seed_mut);
  }
  if (launch_tournament(population.data_ptr<float>(),
                        (long long*)rank.data_ptr<int64_t>(),
                        pool.data_ptr<float>(),
                        (long long*)pool_idx.data_ptr<int64_t>(), N, d,
                        (int)poolsize, log1mp, (unsigned long long)seed_t,
                        cur_stream()) != 0)
    return torch::Tensor();  // caller falls back to the split path
  return variation_events(pool, ci, mi, p1, p2, im, di_c, di_m, lo, hi,
                          mutation_rate, seed_sbx, seed_mut);
}

std::vector<torch::Tensor> sbx_batch(torch::Tensor pool, torch::Tensor p1,
                                     torch::Tensor p2, torch::Tensor di,
                                     torch::Tensor lo, torch::Tensor hi,
                                     int64_t seed) {
  CHECK_GPU(pool);
  const int C = p1.size(0), d = pool.size(1);
  auto out = torch::empty({2 * C, d}, pool.options());
  launch_sbx_batch(pool.data_ptr<float>(), p1.data_ptr<int>(),
                   p2.data_ptr<int>(), di.data_ptr<float>(),
                   lo.data_ptr<float>(), hi.data_ptr<float>(),
                   out.data_ptr<float>(), C, d, (unsigned long long)seed,
                   cur_stream());
  return {out.narrow(0, 0, C), out.narrow(0, C, C)};
}

torch::Tensor mutation_batch(torch::Tensor pool, torch::Tensor parents,
                             torch::Tensor di, torch::Tensor lo,
                             torch::Tensor hi, double mutation_rate,
                             int64_t seed) {
  CHECK_GPU(pool);
  const int M = parents.size(0), d = pool.size(1);
  auto out = torch::empty({M, d}, pool.options());
  launch_mutation_batch(pool.data_ptr<float>(), parents.data_ptr<int>(),
                        di.data_ptr<float>(), lo.data_ptr<float>(),
                        hi.data_ptr<float>(), out.data_ptr<float>(), M, d,
                        (float)mutation_rate, (unsigned long long)seed,
                        cur_stream());
  return out;
}

int64_t hv_mc_uniform_hits(torch::Tensor P, torch::Tensor ideal,
                           torch::Tensor ref, int64_t n_samples,
                           int64_t seed) {
  CHECK_GPU(P);
  TORCH_CHECK(P.size(1) <= 16, "hv mc kernel supports d <= 16");
  auto hits = torch::zeros({1}, P.options().dtype(torch::kInt64));
  launch_hv_mc_uniform(P.data_ptr<float>(), ideal.data_ptr<float>(),
                       ref.data_ptr<float>(),
                       (unsigned long long*)hits.data_ptr<int64_t>(),
                       n_samples, P.size(0), P.size(1),
                       (unsigned long long)seed, cur_stream());
  return hits.item<int64_t>();
}

int64_t hv_fpras_hits(torch::Tensor P, torch::Tensor ref, torch::Tensor cdf,
                      int64_t n_samples, int64_t seed) {
  CHECK_GPU(P);
  TORCH_CHECK(P.size(1) <= 16, "hv fpras kernel supports d <= 16");
  auto hits = torch::zeros({1}, P.options().dtype(torch::kInt64));
  launch_hv_fpras(P.data_ptr<float>(), ref.data_ptr<float>(),
                  cdf.data_ptr<float>(),
                  (unsigned long long*)hits.data_ptr<int64_t>(), n_samples,
                  P.size(0), P.size(1), (unsigned long long)seed,
                  cur_stream());
  return hits.item<int64_t>();
}

torch::Tensor get_duplicates(torch::Tensor X, double eps) {
  CHECK_GPU(X);
  // distance-based duplicate mask on device via torch primitives;
  // semantics of MOEA.py:426-436. compute_mode 1 = direct differences:
  // the GEMM (x^2-2xy+y^2) formulation leaves exact duplicates at
  // norm-vs-dot rounding residuals that dwarf the eps=1e-16 threshold
  // in fp32 — the difference path is exact at zero distance.
  const int n = X.size(0);
  auto D = torch::cdist(X, X, 2.0, 1);
  auto iu = torch::triu_indices(n, n, 0, torch::TensorOptions()
                                              .dtype(torch::kLong)
                                              .device(X.device()));
  D.index_put_({iu[0], iu[1]},
               torch::full({iu.size(1)}, INFINITY, X.options()));
  D = torch::nan_to_num(D, INFINITY);
  return std::get<0>((D <= eps).max(1)).to(torch::kBool);
}

void cmaes_update_(torch::Tensor A, torch::Tensor Ainv, torch::Tensor pc,
                   torch::Tensor z, torch::Tensor psucc, double cc,
                   double ccov, double pthresh) {
  CHECK_GPU(A);
  CHECK_GPU(Ainv);
  CHECK_GPU(pc);
  const int K = A.size(0), d = A.size(1);
  TORCH_CHECK(d <= 256, "cmaes_update supports d <= 256");
  launch_cmaes_update(A.data_ptr<float>(), Ainv.data_ptr<float>(),
                      pc.data_ptr<float>(), z.data_ptr<float>(),
                      psucc.data_ptr<float>(), K, d, (float)cc, (float)ccov,
                      (float)pthresh, cur_stream());
}

// ------------------------------------------------------- exact hypervolume
double hv2d(torch::Tensor P, torch::Tensor ref) {
  CHECK_GPU(P);
  CHECK_GPU(ref);
  TORCH_CHECK(P.scalar_type() == torch::kFloat64, "hv2d expects float64");
  const int n = P.size(0);
  if (n == 0) return 0.0;
  TORCH_CHECK(n <= 8192, "hv2d kernel supports n <= 8192");
  auto out = torch::zeros({1}, P.options());
  launch_hv2d(P.data_ptr<double>(), ref.data_ptr<double>(),
              out.data_ptr<double>(), n, cur_stream());
  return out.item<double>();
}

torch::Tensor hv3d_slices(torch::Tensor Px, torch::Tensor z_thr,
                          torch::Tensor dz, torch::Tensor ref) {
  CHECK_GPU(Px);
  CHECK_GPU(z_thr);
  CHECK_GPU(dz);
  CHECK_GPU(ref);
  TORCH_CHECK(Px.scalar_type() == torch::kFloat64, "hv3d expects float64");
  const int n = Px.size(0);
  auto out = torch::zeros({std::max(n, 1)}, Px.options());
  if (n > 0)
    launch_hv3d_slices(Px.data_ptr<double>(), z_thr.data_ptr<double>(),
                       dz.data_ptr<double>(), ref.data_ptr<double>(),
                       out.data_ptr<double>(), n, cur_stream());
  return out;
}

torch::Tensor ehvi_batch(torch::Tensor L, torch::Tensor U, torch::Tensor mu,
                         torch::Tensor var) {
  CHECK_GPU(L);
  CHECK_GPU(U);
  CHECK_GPU(mu);
  CHECK_GPU(var);
  TORCH_CHECK(mu.scalar_type() == torch::kFloat64, "ehvi expects float64");
  const int B = mu.size(0), nb = L.size(0), d = mu.size(1);
  TORCH_CHECK(d <= 16, "ehvi kernel supports d <= 16");
  auto out = torch::zeros({B}, mu.options());
  if (B > 0 && nb > 0)
    launch_ehvi(L.data_ptr<double>(), U.data_ptr<double>(),
                mu.data_ptr<double>(), var.data_ptr<double>(),
                out.data_ptr<double>(), B, nb, d, cur_stream());
  return out;
}

std::vector<torch::Tensor> lacour_flags(torch::Tensor coords,
                                        torch::Tensor defs,
                                        torch::Tensor pts_aug,
                                        torch::Tensor z) {
  CHECK_GPU(coords);
  CHECK_GPU(defs);
  CHECK_GPU(pts_aug);
  CHECK_GPU(z);
  const int U = coords.size(0), d = coords.size(1);
  auto opts_u8 = coords.options().dtype(torch::kUInt8);
  auto dominated = torch::zeros({U}, opts_u8);
  auto okj = torch::zeros({U, d}, opts_u8);
  launch_lacour_flags(coords.data_ptr<double>(),
                      (const long long*)defs.data_ptr<int64_t>(),
                      pts_aug.data_ptr<double>(), z.data_ptr<double>(),
                      dominated.data_ptr<uint8_t>(), okj.data_ptr<uint8_t>(),
                      U, d, cur_stream());
  return {dominated, okj};
}

void lacour_scatter(torch::Tensor coords, torch::Tensor defs, torch::Tensor z,
                    torch::Tensor dominated, torch::Tensor okj,
                    torch::Tensor slotA, torch::Tensor slotBj,
                    torch::Tensor baseBj, int64_t baseK, int64_t point_idx,
                    torch::Tensor out_coords, torch::Tensor out_defs) {
  CHECK_GPU(coords);
  CHECK_GPU(out_coords);
  const int U = coords.size(0), d = coords.size(1);
  launch_lacour_scatter(
      coords.data_ptr<double>(), (const long long*)defs.data_ptr<int64_t>(),
      z.data_ptr<double>(), dominated.data_ptr<uint8_t>(),
      okj.data_ptr<uint8_t>(), (const long long*)slotA.data_ptr<int64_t>(),
      (const long long*)slotBj.data_ptr<int64_t>(),
      (const long long*)baseBj.data_ptr<int64_t>(), baseK, point_idx,
      out_coords.data_ptr<double>(), (long long*)out_defs.data_ptr<int64_t>(),
      U, d, cur_stream());
}

torch::Tensor lacour_volumes(torch::Tensor coords, torch::Tensor defs,
                             torch::Tensor pts_aug, torch::Tensor ref) {
  CHECK_GPU(coords);
  const int U = coords.size(0), d = coords.size(1);
  auto vol = torch::zeros({std::max(U, 1)}, coords.options());
  if (U > 0)
    launch_lacour_volumes(coords.data_ptr<double>(),
                          (const long long*)defs.data_ptr<int64_t>(),
                          pts_aug.data_ptr<double>(), ref.data_ptr<double>(),
                          vol.data_ptr<double>(), U, d, cur_stream());
  return vol.narrow(0, 0, U);
}

torch::Tensor smpso_velocity(torch::Tensor position, torch::Tensor velocity,
                             torch::Tensor leader1, torch::Tensor leader2,
                             torch::Tensor xlb, torch::Tensor xub, double w,
                             double a1, double a2, double chi) {
  CHECK_GPU(position);
  CHECK_GPU(velocity);
  const int n = position.size(0), d = position.size(1);
  auto out = torch::empty_like(position);
  launch_smpso_velocity(position.data_ptr<float>(), velocity.data_ptr<float>(),
                        leader1.data_ptr<float>(), leader2.data_ptr<float>(),
                        xlb.data_ptr<float>(), xub.data_ptr<float>(),
                        out.data_ptr<float>(), n, d, (float)w, (float)a1,
                        (float)a2, (float)chi, cur_stream());
  return out;
}

torch::Tensor minkowski_norm_matrix(torch::Tensor Y, double p) {
  CHECK_GPU(Y);
  const int m = Y.size(0), d = Y.size(1);
  TORCH_CHECK(d <= 16, "minkowski_norm_matrix supports d <= 16");
  auto D = torch::empty({m, m}, Y.options());
  launch_minkowski_norm_matrix(Y.data_ptr<float>(), D.data_ptr<float>(), m, d,
                               (float)p, cur_stream());
  return D;
}

torch::Tensor agemoea_survival(torch::Tensor D, torch::Tensor preselected) {
  CHECK_GPU(D);
  CHECK_GPU(preselected);
  const int m = D.size(0);
  TORCH_CHECK(m <= 8192, "agemoea_survival supports m <= 8192 (LDS)");
  TORCH_CHECK(D.scalar_type() == torch::kFloat32);
  auto crowd = torch::zeros({m}, D.options());
  launch_agemoea_survival(D.data_ptr<float>(),
                          preselected.data_ptr<uint8_t>(),
                          crowd.data_ptr<float>(), m, cur_stream());
  return crowd;
}

// --------------------------------------------------------------- bf16 path
torch::Tensor mfma_bf16_probe(torch::Tensor A, torch::Tensor B) {
  CHECK_GPU(A);
  CHECK_GPU(B);
  TORCH_CHECK(A.sizes() == torch::IntArrayRef({16, 32}) &&
              B.sizes() == torch::IntArrayRef({32, 16}));
  auto D = torch::zeros({16, 16}, A.options());
  launch_mfma_bf16_probe(A.data_ptr<float>(), B.data_ptr<float>(),
                         D.data_ptr<float>(), cur_stream());
  return D;
}

torch::Tensor matern_cross_bf16(torch::Tensor Xq, torch::Tensor X,
                                torch::Tensor theta, double nu, bool aniso,
                                c10::optional<torch::Tensor> q_lb = c10::nullopt,
                                c10::optional<torch::Tensor> q_invrg = c10::nullopt) {
  CHECK_GPU(Xq);
  CHECK_GPU(X);
  CHECK_GPU(theta);
  const int P = Xq.size(0), N = X.size(0), D = X.size(1), B = theta.size(0);
  auto K = torch::empty({B, P, N}, X.options());
  launch_matern_cross_bf16(
      Xq.data_ptr<float>(), X.data_ptr<float>(), theta.data_ptr<float>(),
      K.data_ptr<float>(), B, P, N, D, theta.size(1), nu_code(nu),
      aniso ? 1 : 0, q_lb ? q_lb->data_ptr<float>() : nullptr,
      q_invrg ? q_invrg->data_ptr<float>() : nullptr, cur_stream());
  return K;
}

std::vector<torch::Tensor> cholesky_batched_bf16_(torch::Tensor A) {
  CHECK_GPU(A);
  const int B = A.size(0), N = A.size(1);
  auto logdet = torch::empty({B}, A.options());
  auto info = torch::zeros({B}, A.options().dtype(torch::kInt32));
  launch_cholesky_multik_bf16(A.data_ptr<float>(), logdet.data_ptr<float>(),
                              info.data_ptr<int>(), B, N, cur_stream());
  return {logdet, info};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("matern_train", &matern_train, "Batched Matern train-kernel assembly");
  m.def("matern_cross", &matern_cross, "Batched Matern cross-kernel assembly");
  m.def("variation_slots", &variation_slots);
  m.def("variation_events", &variation_events);
  m.def("tournament_pool", &tournament_pool);
  m.def("survivor_count", &survivor_count);
  m.def("sceua_propose", &sceua_propose);
  m.def("sceua_accept", &sceua_accept);
  m.def("nsga2_select", &nsga2_select, "Fused survivor selection (cat+rank+crowding+sort+gather)");
  m.def("nsga2_select_acc", &nsga2_select_acc);
  m.def("generation_spawn", &generation_spawn, py::arg("population"), py::arg("rank"), py::arg("poolsize"), py::arg("p_sel"), py::arg("seed_t"), py::arg("ci"), py::arg("mi"), py::arg("p1"), py::arg("p2"), py::arg("im"), py::arg("di_c"), py::arg("di_m"), py::arg("lo"), py::arg("hi"), py::arg("mutation_rate"), py::arg("seed_sbx"), py::arg("seed_mut"), py::arg("rank_sorted") = false);
  m.def("gp_predict_mean", &gp_predict_mean, "Fused cross-kernel + posterior mean",
        py::arg("Xq"), py::arg("X"), py::arg("theta"), py::arg("alpha"),
        py::arg("y_mean"), py::arg("y_std"), py::arg("nu"), py::arg("aniso"),
        py::arg("q_lb") = py::none(), py::arg("q_invrg") = py::none());
  m.def("gp_nmll", &gp_nmll, "Fused batched GP NMLL (assemble+chol+solve+reduce)");
  m.def("cholesky_batched_", &cholesky_batched_,
        "In-place batched Cholesky; returns (logdet, info)");
  m.def("forward_solve_", &forward_solve_, "In-place batched L z = y solve");
  m.def("backward_solve_", &backward_solve_, "In-place batched L^T x = z solve");
  m.def("dominance_degree_matrix", &dominance_degree_matrix);
  m.def("pareto_rank", &pareto_rank, py::arg("Y"), py::arg("stop") = -1);
  m.def("crowding_distance", &crowding_distance);
  m.def("sbx_batch", &sbx_batch);
  m.def("mutation_batch", &mutation_batch);
  m.def("hv_mc_uniform_hits", &hv_mc_uniform_hits);
  m.def("hv_fpras_hits", &hv_fpras_hits);
  m.def("get_duplicates", &get_duplicates);
  m.def("smpso_velocity", &smpso_velocity,
        "Fused SMPSO constriction velocity + clamp");
  m.def("minkowski_norm_matrix", &minkowski_norm_matrix,
        "Row-normalized Minkowski-p distance matrix in one pass");
  m.def("agemoea_survival", &agemoea_survival,
        "Greedy 2-NN AGE-MOEA survival scores (single-workgroup loop)");
  m.def("mfma_bf16_probe", &mfma_bf16_probe,
        "Single-tile bf16 MFMA fragment-layout probe");
  m.def("matern_cross_bf16", &matern_cross_bf16,
        "bf16-MFMA Matern cross-kernel assembly", py::arg("Xq"), py::arg("X"),
        py::arg("theta"), py::arg("nu"), py::arg("aniso"),
        py::arg("q_lb") = c10::nullopt, py::arg("q_invrg") = c10::nullopt);
  m.def("cholesky_batched_bf16_", &cholesky_batched_bf16_,
        "Batched Cholesky with bf16-MFMA trailing updates");
  m.def("hv2d", &hv2d, "Exact 2D hypervolume (LDS bitonic staircase sweep)");
  m.def("hv3d_slices", &hv3d_slices,
        "Per-slice 2D sweep terms of the 3D hypervolume");
  m.def("ehvi_batch", &ehvi_batch,
        "Batched EHVI over the dominated-space box decomposition");
  m.def("lacour_flags", &lacour_flags);
  m.def("lacour_scatter", &lacour_scatter);
  m.def("lacour_volumes", &lacour_volumes);
  m.def("cmaes_update_", &cmaes_update_,
        "In-place batched MO-CMA-ES rank-1 Cholesky update");
}
