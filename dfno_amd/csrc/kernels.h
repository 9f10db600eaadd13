// Host-visible declarations of the gfx950 HIP kernels (implemented in
// pointwise.hip / spectral.hip, bound in bindings.cpp).
#pragma once

#include <torch/extension.h>
// Surface a refused/failed kernel launch loudly instead of silently
// returning stale/zero output (several kernels carry large static LDS or
// max-VGPR configurations where a bad launch is conceivable).
#define DFNO_CHECK_LAUNCH(what)                                            \
  do {                                                                     \
    hipError_t _e = hipGetLastError();                                     \
    TORCH_CHECK(_e == hipSuccess, what, " launch failed: ",                \
                hipGetErrorString(_e));                                    \
  } while (0)

#include <vector>

// fused channel-contraction linear: y[b,o,s] = act(sum_i W[o,i] x[b,i,s] + b[o])
// returns {y, z} where z is the pre-activation (z == y when act == false).
std::vector<at::Tensor> channel_mix_fwd(const at::Tensor& x, const at::Tensor& W,
                                        const at::Tensor& b, bool act);

// transposed contraction for grad-x: gx[b,i,s] = sum_o W[o,i] gz[b,o,s]
at::Tensor channel_mix_fwd_t(const at::Tensor& gz, const at::Tensor& W);

// fused residual linear epilogue: y = gelu(W @ x + res); returns {y, z}
std::vector<at::Tensor> linear_res_gelu_fwd(const at::Tensor& x, const at::Tensor& W,
                                            const at::Tensor& res);

at::Tensor gelu_fwd(const at::Tensor& x);
at::Tensor gelu_bwd(const at::Tensor& gy, const at::Tensor& z);
std::vector<at::Tensor> add_gelu_fwd(const at::Tensor& a, const at::Tensor& b);

// fused projection head: out = W4 @ gelu(W3 @ x + b3) + b4 (no intermediates)
at::Tensor proj_head_fwd(const at::Tensor& x, const at::Tensor& W3,
                         const at::Tensor& b3, const at::Tensor& W4,
                         const at::Tensor& b4);
// returns {gz3, gb3, gW4, gb4}; grad-x = W3^T gz3 (channel_mix_fwd_t) and
// grad-W3 = gz3 @ x^T (channel_mix_bwd_w)
std::vector<at::Tensor> proj_head_bwd(const at::Tensor& gy, const at::Tensor& x,
                                      const at::Tensor& W3, const at::Tensor& b3,
                                      const at::Tensor& W4);
// fully-fused flagship backward (I=20, M=128, O2<=2, fp32): returns
// {gx, gW3, gb3, gW4, gb4} without materializing gz3
std::vector<at::Tensor> proj_head_bwd_fused(const at::Tensor& gy,
                                            const at::Tensor& x,
                                            const at::Tensor& W3,
                                            const at::Tensor& b3,
                                            const at::Tensor& W4);

// split-s outer-product reduction: gW[o,i] = sum_{b,s} gz[b,o,s] x[b,i,s]
// (+ gb[o] = sum gz when want_bias); requires I <= 32.
std::vector<at::Tensor> channel_mix_bwd_w(const at::Tensor& gz, const at::Tensor& x,
                                          bool want_bias);

// fused trunk 20x20 mix backward (mix_bwd.hip): gz = gy * gelu'(z) lives
// only as an LDS tile; returns {gx, gW, gb, gz} (gb empty unless
// want_bias, gz empty unless want_gz — the linear_res_gelu residual grad)
std::vector<at::Tensor> channel_mix_bwd_fused(const at::Tensor& gy,
                                              const at::Tensor& z,
                                              const at::Tensor& x,
                                              const at::Tensor& W,
                                              bool want_bias, bool want_gz);

// fused lift head (T_in == 1): y = gelu(W2 @ gelu(W1 *t x + b1) + b2)
at::Tensor lift_head_fwd(const at::Tensor& x, const at::Tensor& W1,
                         const at::Tensor& b1, const at::Tensor& W2,
                         const at::Tensor& b2);
std::vector<at::Tensor> lift_head_bwd(const at::Tensor& gy, const at::Tensor& x,
                                      const at::Tensor& W1, const at::Tensor& b1,
                                      const at::Tensor& W2, const at::Tensor& b2);

// fused Adam step on flat real views
void adam_step_(at::Tensor& p, const at::Tensor& g, at::Tensor& m, at::Tensor& v,
                double lr, double beta1, double beta2, double eps,
                double weight_decay, int64_t step);
void adam_step_batch_(std::vector<at::Tensor> ps, std::vector<at::Tensor> gs,
                      std::vector<at::Tensor> ms, std::vector<at::Tensor> vs,
                      double lr, double beta1, double beta2, double eps,
                      double weight_decay, std::vector<int64_t> steps);
// fp8 quantize-in-Adam (delayed scaling): qs/am_ins/am_outs parallel to ps
// (undefined/empty tensors for non-fp8 params); returns the indices whose
// e4m3 copies were refreshed by the update kernel itself.
std::vector<int64_t> adam_step_batch_fp8_(
    std::vector<at::Tensor> ps, std::vector<at::Tensor> gs,
    std::vector<at::Tensor> ms, std::vector<at::Tensor> vs,
    double lr, double beta1, double beta2, double eps, double weight_decay,
    std::vector<int64_t> steps, std::vector<at::Tensor> qs,
    std::vector<at::Tensor> am_ins, std::vector<at::Tensor> am_outs);

// fused truncated-spectrum DFTs (see ops/fft.py):
at::Tensor dft_c2c(const at::Tensor& x, int64_t dim, int64_t n,
                   int64_t m_lo, int64_t m_hi, bool analysis, double scale);
at::Tensor dft_rfft_trunc(const at::Tensor& x, int64_t dim, int64_t m);
at::Tensor dft_rfft_trunc_adj(const at::Tensor& gy, int64_t dim, int64_t n);
at::Tensor dft_rfft_trunc_adj_acc(const at::Tensor& gy, int64_t dim, int64_t n,
                                  const at::Tensor& accum);
at::Tensor dft_pad_irfft(const at::Tensor& y, int64_t dim, int64_t n_out, int64_t m);
at::Tensor dft_pad_irfft_adj(const at::Tensor& gx, int64_t dim, int64_t m);
// fused (z,t) boundary 2-D transforms (dft2d.hip): plane-resident in LDS,
// replaces rfft_trunc(t)+fft_trunc(z) (and the inverse pair) without the
// [L, Z, mt] intermediate
at::Tensor dft_zt_fwd(const at::Tensor& x, int64_t mz_lo, int64_t mz_hi,
                      int64_t mt, double scale, bool factors);
at::Tensor dft_zt_inv(const at::Tensor& y, int64_t Z, int64_t T,
                      int64_t mz_lo, int64_t mz_hi, double scale, bool factors,
                      bool out_bf16, const at::Tensor& accum);

// bf16-IO variants (bf16 real-side storage, fp32 compute, c64 spectrum):
at::Tensor dft_pad_irfft_bf16(const at::Tensor& y, int64_t dim, int64_t n_out,
                              int64_t m);
at::Tensor dft_rfft_trunc_adj_bf16(const at::Tensor& gy, int64_t dim,
                                   int64_t n, const at::Tensor& accum);

// bf16-storage pointwise kernels (bf16.hip; fp32 arithmetic):
std::vector<at::Tensor> bf16_channel_mix(const at::Tensor& x, const at::Tensor& W,
                                         const at::Tensor& b, bool act, bool wt,
                                         bool write_z, const at::Tensor& res);
std::vector<at::Tensor> bf16_channel_mix_bwd_w(const at::Tensor& gz,
                                               const at::Tensor& x,
                                               bool want_bias);
at::Tensor bf16_gelu_fwd(const at::Tensor& x);
at::Tensor bf16_gelu_bwd(const at::Tensor& gy, const at::Tensor& z);
std::vector<at::Tensor> bf16_add_gelu(const at::Tensor& a, const at::Tensor& b);

// repartition pack/unpack (pack.hip): gather/scatter block-intersection
// boxes between a contiguous tensor (word view) and one flat staging buffer
void pack_boxes(const at::Tensor& src, at::Tensor& flat,
                const at::Tensor& desc, int64_t max_numel);
void unpack_boxes(const at::Tensor& flat, at::Tensor& dst,
                  const at::Tensor& desc, int64_t max_numel);

// corner-block spectral contraction on the truncated complex spectrum:
//   y[b,o,f] += sum_i x[b,i,f] * w[i,o,f_box]  for f in the corner box
void spectral_corner_fwd(const at::Tensor& x, const at::Tensor& w, at::Tensor& y,
                         std::vector<int64_t> starts);
// gx[b,i,f] += sum_o conj(w[i,o,f_box]) * gy[b,o,f]
void spectral_corner_bwd_x(const at::Tensor& gy, const at::Tensor& w, at::Tensor& gx,
                           std::vector<int64_t> starts);

// single-launch variants covering all corner boxes of one spectral conv
void spectral_corners_fwd(const at::Tensor& x, std::vector<at::Tensor> ws,
                          at::Tensor& y, std::vector<std::vector<int64_t>> starts);
void spectral_corners_bwd_x(const at::Tensor& gy, std::vector<at::Tensor> ws,
                            at::Tensor& gx, std::vector<std::vector<int64_t>> starts);
void spectral_corners_bwd_w(const at::Tensor& x, const at::Tensor& gy,
                            std::vector<at::Tensor> gws,
                            std::vector<std::vector<int64_t>> starts);

// fp8 (e4m3 packed-pair) spectral-weight variants; device amax per corner
// (scale = amax/448 computed on device, no host sync)
void spectral_corners_fwd_fp8(const at::Tensor& x, std::vector<at::Tensor> w16s,
                              std::vector<at::Tensor> amaxes, at::Tensor& y,
                              std::vector<std::vector<int64_t>> starts);
void spectral_corners_bwd_x_fp8(const at::Tensor& gy, std::vector<at::Tensor> w16s,
                                std::vector<at::Tensor> amaxes, at::Tensor& gx,
                                std::vector<std::vector<int64_t>> starts);
void fp8_quant_corners(std::vector<at::Tensor> ws, std::vector<at::Tensor> w16s,
                       std::vector<at::Tensor> amaxes);
