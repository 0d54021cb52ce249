/* tnc_hip — C ABI of the MI355X-native tensor-network contraction executor.
 *
 * This is the drop-in boundary. The reference delegates every
 * pairwise contraction to tblis::tensor_mult via
 *   TensorView::new(labels, shape, strides, ptr)  +
 *   tensor_mult(out_labels, out_shape, a, b) -> Vec<Complex64>
 * at tnc/src/tensornetwork/contraction.rs:111-113. tn_einsum_c128 exports the
 * same operation over complex128 buffers; tn_einsum_c128_dev is the
 * device-resident variant so the contraction walk never round-trips to host.
 * The tn_net_* entry points replace contract_tensor_network
 * (contraction.rs:35-68) for a flat replace-left plan.
 *
 * Conventions (matching the reference call site):
 *  - complex128, row-major unless strides say otherwise; strides are in
 *    ELEMENTS (ndarray-style, may describe non-contiguous views).
 *  - contracted labels = labels present in both A and B (never in out);
 *    out label set = symmetric difference, order chosen by the caller;
 *    K may be 1 (outer product); out may be rank 0 (scalar).
 *  - caller owns all buffers; out is caller-allocated with
 *    prod(out_shape) elements.
 *  - returns 0 on success, nonzero on error (tn_last_error() has the text).
 *    The reference unwraps/panics on error; callers should abort.
 *  - all compute requires an AMD GPU; there is no CPU fallback. Calls fail
 *    with TN_ERR_NO_GPU when no device is present.
 */

#ifndef TNC_HIP_H
#define TNC_HIP_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

enum {
  TN_OK = 0,
  TN_ERR_INVALID = 1,
  TN_ERR_NO_GPU = 2,
  TN_ERR_HIP = 3,
  TN_ERR_OOM = 4,
};

/* Last error message for this thread (valid until the next failing call). */
const char* tn_last_error(void);

/* Number of visible HIP devices (0 if the runtime reports none). */
int tn_device_count(void);

/* Synchronous einsum over HOST buffers (uploads, contracts on device 0 or
 * the device selected with tn_set_device, downloads). */
int tn_einsum_c128(const uint64_t* out_labels, const uint64_t* out_shape,
                   size_t out_ndim, const uint64_t* a_labels,
                   const uint64_t* a_shape, const int64_t* a_strides,
                   const void* a_data, size_t a_ndim, const uint64_t* b_labels,
                   const uint64_t* b_shape, const int64_t* b_strides,
                   const void* b_data, size_t b_ndim, void* out_data);

/* Einsum over DEVICE buffers, asynchronous on `stream` (a hipStream_t; NULL =
 * default stream). All pointers are device pointers on the current device.
 * Workspace (packing buffers) is drawn from the library's device pool. */
int tn_einsum_c128_dev(const uint64_t* out_labels, const uint64_t* out_shape,
                       size_t out_ndim, const uint64_t* a_labels,
                       const uint64_t* a_shape, const int64_t* a_strides,
                       const void* a_dev, size_t a_ndim,
                       const uint64_t* b_labels, const uint64_t* b_shape,
                       const int64_t* b_strides, const void* b_dev,
                       size_t b_ndim, void* out_dev, void* stream);

/* complex64 variants (the config-5 precision/throughput trade: f32 MFMA
 * runs at ~2x the f64 matrix rate). Same contract, 8-byte elements. */
int tn_einsum_c64(const uint64_t* out_labels, const uint64_t* out_shape,
                  size_t out_ndim, const uint64_t* a_labels,
                  const uint64_t* a_shape, const int64_t* a_strides,
                  const void* a_data, size_t a_ndim, const uint64_t* b_labels,
                  const uint64_t* b_shape, const int64_t* b_strides,
                  const void* b_data, size_t b_ndim, void* out_data);

int tn_einsum_c64_dev(const uint64_t* out_labels, const uint64_t* out_shape,
                      size_t out_ndim, const uint64_t* a_labels,
                      const uint64_t* a_shape, const int64_t* a_strides,
                      const void* a_dev, size_t a_ndim,
                      const uint64_t* b_labels, const uint64_t* b_shape,
                      const int64_t* b_strides, const void* b_dev,
                      size_t b_ndim, void* out_dev, void* stream);

/* Select the HIP device used by subsequently created nets / einsum calls. */
int tn_set_device(int device);

/* ------------ network executor (contract_tensor_network) ------------- */

typedef struct tn_net tn_net;

/* Create an executor bound to `device` (complex128 tensors). */
tn_net* tn_net_create(int device);

/* Like tn_net_create with an element type: dtype 0 = complex128,
 * 1 = complex64 (leaf/result buffers are then 8-byte elements). */
tn_net* tn_net_create2(int device, int dtype);

/* Reserve a device arena of `bytes` for intermediates and packing
 * workspaces (one hipMalloc; first-fit, stream-ordered reuse). Without a
 * reservation the executor falls back to stream-ordered hipMallocAsync,
 * which costs seconds per contraction at tens-of-GB intermediates. */
int tn_net_reserve(tn_net* net, uint64_t bytes);

/* Register leaf `index = return value` with labels/dims and host data
 * (complex128, row-major, contiguous). Data is uploaded immediately and
 * persists across contract calls. Returns a negative value on error. */
int64_t tn_net_add_leaf(tn_net* net, const uint64_t* labels,
                        const uint64_t* dims, size_t ndim,
                        const void* host_data);

/* Contract along a flat replace-left path: pairs = [i0,j0, i1,j1, ...].
 * Equivalent to the reference's recursive walk after flattening
 * (contraction.rs:35-68). Intermediates stay on device; leaves are not
 * consumed, so the call is repeatable. Synchronizes before returning;
 * elapsed_ms (optional) gets the device-side wall time of the walk. */
int tn_net_contract(tn_net* net, const uint64_t* pairs, size_t nsteps,
                    double* elapsed_ms);

/* Like tn_net_contract but also fills per-step kernel timings:
 * step_ms[nsteps] = HIP-event time of step s (all kernels of the step),
 * gemm_ms[nsteps] = time of the GEMM kernel alone (0 for non-GEMM steps),
 * kind[nsteps] = kernel class (0=smallk stream, 1=dot, 2=gemm+packs,
 * 3 = gemm with unpack). */
int tn_net_contract_profiled(tn_net* net, const uint64_t* pairs, size_t nsteps,
                             double* step_ms, double* gemm_ms, int32_t* kind,
                             double* elapsed_ms);

/* Synchronous device-to-device copy on the current device (for exchanging
 * buffers with an external allocator, e.g. RCCL-communicated tensors). */
int tn_memcpy_dtod(void* dst, const void* src, uint64_t bytes);

/* Synchronous device-to-host copy (fetch an exchanged intermediate). */
int tn_memcpy_dtoh(void* host_dst, const void* dev_src, uint64_t bytes);

/* Metadata of the final tensor after the last contract call. labels/dims
 * must have room for 64 entries. */
int tn_net_result_meta(tn_net* net, uint64_t* labels, uint64_t* dims,
                       size_t* ndim);

/* Copy the final tensor to host (caller-allocated, prod(dims) c128). */
int tn_net_result_data(tn_net* net, void* host_out);

/* Device pointer of the final tensor (valid until the next contract call). */
void* tn_net_result_dev(tn_net* net);

/* Import an external DEVICE tensor (e.g. an RCCL-received intermediate) as a
 * new leaf without copying; the buffer must outlive the net or be detached
 * before reuse. */
int64_t tn_net_add_leaf_dev(tn_net* net, const uint64_t* labels,
                            const uint64_t* dims, size_t ndim, void* dev_data);

/* Bytes currently held by the net's device pool (watermark telemetry). */
int tn_net_pool_bytes(tn_net* net, uint64_t* in_use, uint64_t* cached);

void tn_net_destroy(tn_net* net);

#ifdef __cplusplus
}
#endif

#endif /* TNC_HIP_H */
