/* copr_comm.cpp — RCCL merge steps behind the C-ABI (include/copr_gpu.h).
 *
 * The engine's ONLY collective surface: the final partial-aggregate merge of
 * Region-sharded execution. One process per
 * GPU, RCCL over xGMI; the reference has no collectives at all on this path
 * (its parallelism is per-Region request fan-out, endpoint.rs:238-248), so
 * these mirror what the per-rank partial results need:
 *   count      -> ncclAllReduce sum on u64
 *   f64 sum    -> ncclAllReduce sum on double (1-ULP class, north_star budget)
 *   i128 sum   -> allgather 16-byte limbs + exact host fold with carries
 *   CRC64 XOR  -> allgather u64 + host XOR fold (RCCL has no XOR reduce;
 *                 checksum.rs:78-87 order-independence makes any fold valid)
 */
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>
#include "../../include/copr_gpu.h"
#include "copr_internal.h"

#include <cstring>
#include <string>
#include <vector>

namespace copr {

#define SET_ERR(st, msg) (copr::tls_err() = (msg), (st))
#define HIP_TRY(expr, what)                                                  \
  do {                                                                       \
    hipError_t _e = (expr);                                                  \
    if (_e != hipSuccess) {                                                  \
      copr::tls_err() = std::string(what) + ": " + hipGetErrorString(_e);    \
      return COPR_ERR_INTERNAL;                                              \
    }                                                                        \
  } while (0)
#define NCCL_TRY(expr, what)                                                 \
  do {                                                                       \
    ncclResult_t _r = (expr);                                                \
    if (_r != ncclSuccess) {                                                 \
      copr::tls_err() = std::string(what) + ": " + ncclGetErrorString(_r);   \
      return COPR_ERR_INTERNAL;                                              \
    }                                                                        \
  } while (0)

struct CommState {
  ncclComm_t comm = nullptr;
  int n_ranks = 0, rank = -1;
  /* persistent device scratch: 16 B send + 16 B * n_ranks recv */
  uint8_t *d_send = nullptr;
  uint8_t *d_recv = nullptr;
};

void comm_free(copr_engine *eng) {
  if (!eng || !eng->comm_state) return;
  CommState *cs = (CommState *)eng->comm_state;
  if (cs->comm) ncclCommDestroy(cs->comm);
  if (cs->d_send) hipFree(cs->d_send);
  if (cs->d_recv) hipFree(cs->d_recv);
  delete cs;
  eng->comm_state = nullptr;
}

static CommState *get_cs(copr_engine *eng) {
  return eng ? (CommState *)eng->comm_state : nullptr;
}

}  // namespace copr

using namespace copr;

static_assert(sizeof(ncclUniqueId) == COPR_COMM_ID_BYTES,
              "ncclUniqueId size is the ABI's 128-byte id");

extern "C" copr_status copr_comm_id(uint8_t out[COPR_COMM_ID_BYTES]) {
  ncclUniqueId id;
  NCCL_TRY(ncclGetUniqueId(&id), "ncclGetUniqueId");
  memcpy(out, &id, sizeof(id));
  return COPR_OK;
}

extern "C" copr_status copr_comm_create(copr_engine *eng,
                                        const uint8_t id[COPR_COMM_ID_BYTES],
                                        int n_ranks, int rank) {
  if (!eng) return SET_ERR(COPR_ERR_INVALID_REQUEST, "null engine");
  if (eng->comm_state)
    return SET_ERR(COPR_ERR_INVALID_REQUEST, "communicator already created");
  if (n_ranks < 1 || rank < 0 || rank >= n_ranks)
    return SET_ERR(COPR_ERR_INVALID_REQUEST, "bad rank/n_ranks");
  HIP_TRY(hipSetDevice(eng->device), "hipSetDevice");
  CommState *cs = new CommState();
  cs->n_ranks = n_ranks;
  cs->rank = rank;
  hipError_t e = hipMalloc(&cs->d_send, 16);
  if (e == hipSuccess) e = hipMalloc(&cs->d_recv, 16 * (size_t)n_ranks);
  if (e != hipSuccess) {
    if (cs->d_send) hipFree(cs->d_send);
    delete cs;
    return SET_ERR(COPR_ERR_OOM, "comm scratch alloc");
  }
  ncclUniqueId nid;
  memcpy(&nid, id, sizeof(nid));
  ncclResult_t r = ncclCommInitRank(&cs->comm, n_ranks, nid, rank);
  if (r != ncclSuccess) {
    hipFree(cs->d_send);
    hipFree(cs->d_recv);
    delete cs;
    copr::tls_err() = std::string("ncclCommInitRank: ") + ncclGetErrorString(r);
    return COPR_ERR_INTERNAL;
  }
  eng->comm_state = cs;
  return COPR_OK;
}

extern "C" void copr_comm_destroy(copr_engine *eng) { comm_free(eng); }

extern "C" copr_status copr_merge_count(copr_engine *eng, uint64_t *inout) {
  CommState *cs = get_cs(eng);
  if (!cs) return SET_ERR(COPR_ERR_INVALID_REQUEST, "no communicator");
  HIP_TRY(hipSetDevice(eng->device), "hipSetDevice");
  HIP_TRY(hipMemcpyAsync(cs->d_send, inout, 8, hipMemcpyHostToDevice,
                         eng->stream), "merge H2D");
  NCCL_TRY(ncclAllReduce(cs->d_send, cs->d_recv, 1, ncclUint64, ncclSum,
                         cs->comm, eng->stream), "ncclAllReduce u64");
  HIP_TRY(hipMemcpyAsync(inout, cs->d_recv, 8, hipMemcpyDeviceToHost,
                         eng->stream), "merge D2H");
  HIP_TRY(hipStreamSynchronize(eng->stream), "merge sync");
  return COPR_OK;
}

extern "C" copr_status copr_merge_sum_f64(copr_engine *eng, double *inout) {
  CommState *cs = get_cs(eng);
  if (!cs) return SET_ERR(COPR_ERR_INVALID_REQUEST, "no communicator");
  HIP_TRY(hipSetDevice(eng->device), "hipSetDevice");
  HIP_TRY(hipMemcpyAsync(cs->d_send, inout, 8, hipMemcpyHostToDevice,
                         eng->stream), "merge H2D");
  NCCL_TRY(ncclAllReduce(cs->d_send, cs->d_recv, 1, ncclDouble, ncclSum,
                         cs->comm, eng->stream), "ncclAllReduce f64");
  HIP_TRY(hipMemcpyAsync(inout, cs->d_recv, 8, hipMemcpyDeviceToHost,
                         eng->stream), "merge D2H");
  HIP_TRY(hipStreamSynchronize(eng->stream), "merge sync");
  return COPR_OK;
}

extern "C" copr_status copr_merge_checksum(copr_engine *eng, uint64_t *inout) {
  CommState *cs = get_cs(eng);
  if (!cs) return SET_ERR(COPR_ERR_INVALID_REQUEST, "no communicator");
  HIP_TRY(hipSetDevice(eng->device), "hipSetDevice");
  HIP_TRY(hipMemcpyAsync(cs->d_send, inout, 8, hipMemcpyHostToDevice,
                         eng->stream), "merge H2D");
  NCCL_TRY(ncclAllGather(cs->d_send, cs->d_recv, 1, ncclUint64, cs->comm,
                         eng->stream), "ncclAllGather u64");
  std::vector<uint64_t> all(cs->n_ranks);
  HIP_TRY(hipMemcpyAsync(all.data(), cs->d_recv, 8 * (size_t)cs->n_ranks,
                         hipMemcpyDeviceToHost, eng->stream), "merge D2H");
  HIP_TRY(hipStreamSynchronize(eng->stream), "merge sync");
  uint64_t acc = 0;
  for (uint64_t v : all) acc ^= v;
  *inout = acc;
  return COPR_OK;
}

extern "C" copr_status copr_merge_sum_i128(copr_engine *eng, uint64_t *lo,
                                           uint64_t *hi) {
  CommState *cs = get_cs(eng);
  if (!cs) return SET_ERR(COPR_ERR_INVALID_REQUEST, "no communicator");
  HIP_TRY(hipSetDevice(eng->device), "hipSetDevice");
  uint64_t limbs[2] = {*lo, *hi};
  HIP_TRY(hipMemcpyAsync(cs->d_send, limbs, 16, hipMemcpyHostToDevice,
                         eng->stream), "merge H2D");
  NCCL_TRY(ncclAllGather(cs->d_send, cs->d_recv, 2, ncclUint64, cs->comm,
                         eng->stream), "ncclAllGather i128");
  std::vector<uint64_t> all(2 * (size_t)cs->n_ranks);
  HIP_TRY(hipMemcpyAsync(all.data(), cs->d_recv, 16 * (size_t)cs->n_ranks,
                         hipMemcpyDeviceToHost, eng->stream), "merge D2H");
  HIP_TRY(hipStreamSynchronize(eng->stream), "merge sync");
  /* exact two's-complement i128 fold */
  unsigned __int128 acc = 0;
  for (int r = 0; r < cs->n_ranks; r++) {
    unsigned __int128 v =
        ((unsigned __int128)all[2 * r + 1] << 64) | all[2 * r];
    acc += v;
  }
  *lo = (uint64_t)acc;
  *hi = (uint64_t)(acc >> 64);
  return COPR_OK;
}
