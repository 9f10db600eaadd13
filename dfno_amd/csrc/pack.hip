// Repartition pack/unpack kernels ("pack/unpack via HIP
// kernels").  Reference counterpart: DistDL's Repartition moves each
// block-intersection as an individual tensor slice through MPI (used at
// /root/reference/dfno/dfno.py:51-60); here every box is gathered into
// ONE flat staging buffer per peer (and scattered back) so the RCCL
// exchange sends a single contiguous message per rank pair.  A repartition plan's send side packs every block-intersection
// box of the (contiguous) local tensor into ONE flat staging buffer (per-peer
// contiguous ranges are sliced off for grouped ncclSend), and the recv side
// scatters one flat buffer into the destination block — one kernel launch per
// direction instead of one aten slice-copy launch per piece (the python-side
// slicing VERDICT.md round-1 flagged costs ~3-8 us per piece per launch,
// x pieces x 4 repartitions x 2 directions per block at 8 ranks).
//
// Descriptors are precomputed host-side per (plan, dtype, device) and cached
// on the plan: per piece 2 + 2 + 8 + 8 = 20 longs
//   [0] flat_off   (words into the flat buffer)
//   [1] tens_off   (words into the tensor)
//   [2] numel      (words in the box)
//   [3] ndim       (effective dims after host-side contiguous-run merging)
//   [4..11]  dims    (outer -> inner; innermost varies fastest)
//   [12..19] strides (words)
// Boxes are expressed in WORDS (fp32/fp64 units; complex = 2 words, the
// trailing word dim merged into the innermost contiguous run), so one kernel
// serves fp32/c64 and fp64/c128.  Consecutive threads walk the innermost
// (stride-1) run: reads and writes are coalesced on both sides.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

#include "kernels.h"

namespace {

constexpr int kRec = 20;   // longs per piece record
constexpr int kTPB = 256;

template <typename T, bool PACK>
__global__ __launch_bounds__(kTPB) void copy_boxes_kernel(
    const T* __restrict__ tens_in, T* __restrict__ tens_out,
    const T* __restrict__ flat_in, T* __restrict__ flat_out,
    const long* __restrict__ desc) {
  const long* d = desc + (long)blockIdx.y * kRec;
  const long flat_off = d[0];
  const long tens_off = d[1];
  const long numel = d[2];
  const int nd = (int)d[3];

  long t0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long gstride = (long)gridDim.x * blockDim.x;
  for (long e = t0; e < numel; e += gstride) {
    long rem = e;
    long off = tens_off;
#pragma unroll
    for (int k = 7; k >= 1; --k) {
      if (k < nd) {
        const long dim = d[4 + k];
        const long idx = rem % dim;
        rem /= dim;
        off += idx * d[12 + k];
      }
    }
    off += rem * d[12];            // outermost (k == 0)
    if constexpr (PACK) {
      flat_out[flat_off + e] = tens_in[off];
    } else {
      tens_out[off] = flat_in[flat_off + e];
    }
  }
}

int grid_x_for(long max_numel) {
  long g = (max_numel + kTPB - 1) / kTPB;
  if (g > 2048) g = 2048;
  if (g < 1) g = 1;
  return (int)g;
}

}  // namespace

void pack_boxes(const at::Tensor& src, at::Tensor& flat,
                const at::Tensor& desc, int64_t max_numel) {
  TORCH_CHECK(src.is_cuda() && flat.is_cuda() && desc.is_cuda(), "pack_boxes: GPU only");
  TORCH_CHECK(desc.scalar_type() == at::kLong && desc.is_contiguous(), "pack_boxes: bad desc");
  const int npieces = (int)(desc.numel() / kRec);
  if (npieces == 0) return;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 grid(grid_x_for(max_numel), npieces);
  if (src.scalar_type() == at::kDouble) {
    hipLaunchKernelGGL((copy_boxes_kernel<double, true>), grid, dim3(kTPB), 0, stream,
                       src.data_ptr<double>(), nullptr, nullptr,
                       flat.data_ptr<double>(), desc.data_ptr<long>());
  } else if (src.scalar_type() == at::kBFloat16) {
    // bf16 exchanges: 2-byte words (the descriptors count elements)
    hipLaunchKernelGGL((copy_boxes_kernel<unsigned short, true>), grid,
                       dim3(kTPB), 0, stream,
                       reinterpret_cast<const unsigned short*>(src.data_ptr()),
                       nullptr, nullptr,
                       reinterpret_cast<unsigned short*>(flat.data_ptr()),
                       desc.data_ptr<long>());
  } else {
    TORCH_CHECK(src.scalar_type() == at::kFloat,
                "pack_boxes: fp32/fp64/bf16 words only");
    hipLaunchKernelGGL((copy_boxes_kernel<float, true>), grid, dim3(kTPB), 0, stream,
                       src.data_ptr<float>(), nullptr, nullptr,
                       flat.data_ptr<float>(), desc.data_ptr<long>());
  }
  DFNO_CHECK_LAUNCH("pack_boxes");
}

void unpack_boxes(const at::Tensor& flat, at::Tensor& dst,
                  const at::Tensor& desc, int64_t max_numel) {
  TORCH_CHECK(flat.is_cuda() && dst.is_cuda() && desc.is_cuda(), "unpack_boxes: GPU only");
  TORCH_CHECK(desc.scalar_type() == at::kLong && desc.is_contiguous(), "unpack_boxes: bad desc");
  const int npieces = (int)(desc.numel() / kRec);
  if (npieces == 0) return;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  dim3 grid(grid_x_for(max_numel), npieces);
  if (dst.scalar_type() == at::kDouble) {
    hipLaunchKernelGGL((copy_boxes_kernel<double, false>), grid, dim3(kTPB), 0, stream,
                       nullptr, dst.data_ptr<double>(),
                       flat.data_ptr<double>(), nullptr, desc.data_ptr<long>());
  } else if (dst.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL((copy_boxes_kernel<unsigned short, false>), grid,
                       dim3(kTPB), 0, stream, nullptr,
                       reinterpret_cast<unsigned short*>(dst.data_ptr()),
                       reinterpret_cast<const unsigned short*>(flat.data_ptr()),
                       nullptr, desc.data_ptr<long>());
  } else {
    TORCH_CHECK(dst.scalar_type() == at::kFloat,
                "unpack_boxes: fp32/fp64/bf16 words only");
    hipLaunchKernelGGL((copy_boxes_kernel<float, false>), grid, dim3(kTPB), 0, stream,
                       nullptr, dst.data_ptr<float>(),
                       flat.data_ptr<float>(), nullptr, desc.data_ptr<long>());
  }
  DFNO_CHECK_LAUNCH("unpack_boxes");
}
