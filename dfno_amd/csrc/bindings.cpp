// Python bindings for the dfno_amd gfx950 HIP kernels.
#include <torch/extension.h>

#include "kernels.h"

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "dfno_amd MI355X (gfx950) native kernels";
  m.def("channel_mix_fwd", &channel_mix_fwd,
        "fused channel linear + bias + gelu: returns (y, z)");
  m.def("channel_mix_fwd_t", &channel_mix_fwd_t,
        "transposed channel contraction (grad-x)");
  m.def("channel_mix_bwd_w", &channel_mix_bwd_w,
        "split-s grad-W (+grad-bias) reduction");
  m.def("linear_res_gelu_fwd", &linear_res_gelu_fwd,
        "fused y = gelu(W @ x + res): returns (y, z)");
  m.def("gelu_fwd", &gelu_fwd, "exact gelu");
  m.def("gelu_bwd", &gelu_bwd, "gelu backward (gy, z) -> gz");
  m.def("add_gelu_fwd", &add_gelu_fwd, "fused residual add + gelu: returns (y, z)");
  m.def("proj_head_fwd", &proj_head_fwd, "fused linear->gelu->linear head");
  m.def("proj_head_bwd", &proj_head_bwd,
        "fused head backward: returns (gz3, gb3, gW4, gb4)");
  m.def("channel_mix_bwd_fused", &channel_mix_bwd_fused,
        "fused trunk mix backward: (gx, gW, gb, gz)");
  m.def("proj_head_bwd_fused", &proj_head_bwd_fused,
        "fully-fused flagship head backward: (gx, gW3, gb3, gW4, gb4)");
  m.def("lift_head_fwd", &lift_head_fwd, "fused time-lift + channel-lift + gelus");
  m.def("lift_head_bwd", &lift_head_bwd,
        "lift head backward: (gx, gW1, gb1, gW2, gb2)");
  m.def("adam_step_", &adam_step_, "fused Adam step (in-place)");
  m.def("adam_step_batch_", &adam_step_batch_, "fused Adam step over a tensor list");
  m.def("adam_step_batch_fp8_", &adam_step_batch_fp8_,
        "fused Adam step with in-kernel e4m3 requantization (delayed scaling)");
  m.def("dft_c2c", &dft_c2c, "truncated/padded complex DFT along a dim");
  m.def("dft_rfft_trunc", &dft_rfft_trunc, "real->kept-low-modes DFT (last dim)");
  m.def("dft_rfft_trunc_adj", &dft_rfft_trunc_adj, "adjoint of dft_rfft_trunc");
  m.def("dft_rfft_trunc_adj_acc", &dft_rfft_trunc_adj_acc,
        "adjoint of dft_rfft_trunc with fused accumulate addend");
  m.def("dft_pad_irfft", &dft_pad_irfft, "kept modes -> real inverse (last dim)");
  m.def("dft_pad_irfft_adj", &dft_pad_irfft_adj, "adjoint of dft_pad_irfft");
  m.def("dft_zt_fwd", &dft_zt_fwd,
        "fused truncated (z,t) 2-D analysis (real -> kept-mode c64)");
  m.def("dft_zt_inv", &dft_zt_inv,
        "fused padded (z,t) 2-D synthesis (kept-mode c64 -> real)");
  m.def("dft_pad_irfft_bf16", &dft_pad_irfft_bf16,
        "pad_irfft with bf16 output (bf16 compute config)");
  m.def("dft_rfft_trunc_adj_bf16", &dft_rfft_trunc_adj_bf16,
        "rfft_trunc adjoint with bf16 output");
  m.def("spectral_corner_fwd", &spectral_corner_fwd,
        "corner-block complex spectral contraction (accumulate into y box)");
  m.def("spectral_corner_bwd_x", &spectral_corner_bwd_x,
        "corner-block spectral contraction adjoint wrt x");
  m.def("spectral_corners_fwd", &spectral_corners_fwd,
        "all corner boxes in one launch");
  m.def("spectral_corners_bwd_x", &spectral_corners_bwd_x,
        "all corner boxes adjoint wrt x in one launch");
  m.def("spectral_corners_bwd_w", &spectral_corners_bwd_w,
        "multi-corner spectral grad-W (gw[i,o,e] = sum_b x conj(gy))");
  m.def("spectral_corners_fwd_fp8", &spectral_corners_fwd_fp8,
        "fp8-weight spectral contraction (e4m3 packed pairs + per-corner scale)");
  m.def("spectral_corners_bwd_x_fp8", &spectral_corners_bwd_x_fp8,
        "fp8-weight spectral contraction adjoint wrt x");
  m.def("fp8_quant_corners", &fp8_quant_corners,
        "device-side e4m3 requantization of the corner masters");
  m.def("bf16_channel_mix", &bf16_channel_mix,
        "bf16-storage fused channel linear (+bias/res/gelu): returns (y, z)");
  m.def("bf16_channel_mix_bwd_w", &bf16_channel_mix_bwd_w,
        "bf16 grad-W/grad-bias reduction (fp32 accumulation)");
  m.def("bf16_gelu_fwd", &bf16_gelu_fwd, "bf16 exact gelu");
  m.def("bf16_gelu_bwd", &bf16_gelu_bwd, "bf16 gelu backward");
  m.def("bf16_add_gelu", &bf16_add_gelu, "bf16 fused add+gelu: returns (y, z)");
  m.def("pack_boxes", &pack_boxes,
        "gather repartition boxes into one flat staging buffer");
  m.def("unpack_boxes", &unpack_boxes,
        "scatter one flat staging buffer into repartition boxes");
}
