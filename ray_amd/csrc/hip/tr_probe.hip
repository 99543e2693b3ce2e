// Probe: empirical semantics of ds_read_tr16_b64 on gfx950 (which LDS
// element lands in which lane/slot). Pattern: lds[i] = i.
#include <hip/hip_runtime.h>
#include <torch/extension.h>
typedef __attribute__((ext_vector_type(4))) short short4v;

extern "C" __global__ void tr_probe_kernel(short* out, const int* addrs) {
  __shared__ short lds[2048];
  for (int i = threadIdx.x; i < 2048; i += blockDim.x) lds[i] = (short)i;
  __syncthreads();
  int l = threadIdx.x;
  auto p = (__attribute__((address_space(3))) short4v*)&lds[addrs[l]];
  short4v r = __builtin_amdgcn_ds_read_tr16_b64_v4i16(p);
  for (int i = 0; i < 4; ++i) out[l * 4 + i] = r[i];
}

at::Tensor tr_probe(at::Tensor addrs) {
  auto out = at::zeros({64 * 4}, addrs.options().dtype(at::kShort));
  hipLaunchKernelGGL(tr_probe_kernel, dim3(1), dim3(64), 0, 0,
                     (short*)out.data_ptr(), addrs.data_ptr<int>());
  return out;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) { m.def("tr_probe", &tr_probe); }
