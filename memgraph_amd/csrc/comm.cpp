// RCCL over xGMI — the multi-GPU transport for the sharded PageRank path
// (SURVEY.md §8e). The reference has no analytics-parallel transport at all
// (its only cross-process stack is replication RPC, src/rpc + src/slk,
// which ships WAL deltas, never parallelizes analytics); this is the
// MI355X-native equivalent: one process per GPU, ncclAllGather of the owned
// contrib/rank slices per iteration.

#include <cstring>

#include <rccl/rccl.h>

#include "mgx_internal.h"

struct mgx_comm_state {
  ncclComm_t comm = nullptr;
  int rank = -1;
  int world = 0;
};

#define MGX_NCCL_TRY(expr)                                                    \
  do {                                                                        \
    ncclResult_t _r = (expr);                                                 \
    if (_r != ncclSuccess) {                                                  \
      mgx_set_error("%s:%d: %s failed: %s", __FILE__, __LINE__, #expr,        \
                    ncclGetErrorString(_r));                                  \
      return MGX_ERR_NCCL;                                                    \
    }                                                                         \
  } while (0)

extern "C" mgx_status mgx_comm_unique_id(void *out_bytes) {
  static_assert(sizeof(ncclUniqueId) == MGX_UNIQUE_ID_BYTES, "ncclUniqueId size");
  ncclUniqueId id;
  MGX_NCCL_TRY(ncclGetUniqueId(&id));
  memcpy(out_bytes, &id, sizeof(id));
  return MGX_OK;
}

extern "C" mgx_status mgx_comm_init(mgx_context *ctx, int rank, int world_size,
                                    const void *id_bytes) {
  MGX_HIP_TRY(hipSetDevice(ctx->device));
  ncclUniqueId id;
  memcpy(&id, id_bytes, sizeof(id));
  auto *st = new mgx_comm_state();
  ncclResult_t r = ncclCommInitRank(&st->comm, world_size, id, rank);
  if (r != ncclSuccess) {
    mgx_set_error("ncclCommInitRank failed: %s", ncclGetErrorString(r));
    delete st;
    return MGX_ERR_NCCL;
  }
  st->rank = rank;
  st->world = world_size;
  ctx->comm = st;
  return MGX_OK;
}

extern "C" mgx_status mgx_comm_destroy(mgx_context *ctx) {
  if (!ctx->comm) return MGX_OK;
  (void)ncclCommDestroy(ctx->comm->comm);
  delete ctx->comm;
  ctx->comm = nullptr;
  return MGX_OK;
}

int mgx_comm_world(mgx_context *ctx) { return ctx->comm ? ctx->comm->world : 0; }

mgx_status mgx_comm_allgather_f32(mgx_context *ctx, const float *send, float *recv,
                                  size_t per_rank_count) {
  MGX_NCCL_TRY(ncclAllGather(send, recv, per_rank_count, ncclFloat32, ctx->comm->comm,
                             ctx->stream));
  return MGX_OK;
}

mgx_status mgx_comm_allreduce_max_f32(mgx_context *ctx, const float *send, float *recv) {
  MGX_NCCL_TRY(ncclAllReduce(send, recv, 1, ncclFloat32, ncclMax, ctx->comm->comm,
                             ctx->stream));
  return MGX_OK;
}
