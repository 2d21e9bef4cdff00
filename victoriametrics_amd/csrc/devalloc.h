/* Stream-ordered device allocation (hipMallocAsync on the library stream,
 * release threshold = keep-everything): repeated create/destroy of
 * multi-GB batches costs microseconds instead of a fresh VA carve per
 * query — the memory-pool pattern a production allocator would use. */
#ifndef VMGPU_DEVALLOC_H
#define VMGPU_DEVALLOC_H
#include <hip/hip_runtime.h>

hipError_t vm_dev_malloc_raw(void** p, size_t n);
hipError_t vm_dev_free_raw(void* p);
/* the library context stream the pool allocations are ordered on — any
 * standalone entry point doing its own copies/launches must use THIS
 * stream (stream-ordered memory is undefined on other streams without
 * explicit synchronization) */
hipStream_t vm_ctx_stream(void);

template <class T>
static inline hipError_t vm_dev_malloc(T** p, size_t n) {
  return vm_dev_malloc_raw((void**)p, n);
}
static inline hipError_t vm_dev_free(void* p) { return vm_dev_free_raw(p); }
#endif
