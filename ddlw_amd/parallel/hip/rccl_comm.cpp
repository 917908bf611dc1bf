// Native RCCL communicator layer — the Horovod-core (C++) equivalent
// (SURVEY.md §2.5 "Horovod core" row: fusion-buffered collectives on a side
// HIP stream; reference call sites §2.3 C1-C4).
//
// Design: a thin C ABI over RCCL that ddlw_amd.parallel.native binds with
// ctypes. Rendezvous (exchanging the ncclUniqueId) is done by the Python
// side over the torch.distributed store; everything data-plane — communicator
// lifetime, allreduce (ncclAvg in one pass), broadcast — runs here, enqueued
// on whatever HIP stream the caller passes (ddlw uses a dedicated side
// stream ordered against compute by events, so collectives overlap backward).
//
// Built standalone by hipcc (no torch headers): tensors cross the boundary
// as raw device pointers + dtype tags.

#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <cstring>

#define DDLW_EXPORT extern "C" __attribute__((visibility("default")))

namespace {

ncclDataType_t to_nccl(int dtype, bool* ok) {
    *ok = true;
    switch (dtype) {
        case 0: return ncclFloat32;
        case 1: return ncclBfloat16;
        case 2: return ncclFloat64;
        case 3: return ncclInt64;
        case 4: return ncclUint8;
        case 5: return ncclFloat16;
        case 6: return ncclInt32;
        default: *ok = false; return ncclFloat32;
    }
}

thread_local ncclResult_t g_last = ncclSuccess;

int ret(ncclResult_t r) {
    g_last = r;
    return static_cast<int>(r);
}

}  // namespace

DDLW_EXPORT int ddlw_rccl_unique_id_bytes() { return NCCL_UNIQUE_ID_BYTES; }

DDLW_EXPORT int ddlw_rccl_get_unique_id(char* out /* NCCL_UNIQUE_ID_BYTES */) {
    ncclUniqueId id;
    ncclResult_t r = ncclGetUniqueId(&id);
    if (r == ncclSuccess) std::memcpy(out, id.internal, NCCL_UNIQUE_ID_BYTES);
    return ret(r);
}

// Returns an opaque communicator handle (0 on failure). The caller must have
// made the right HIP device current (one process per GPU).
DDLW_EXPORT long long ddlw_rccl_comm_init(int nranks, int rank, const char* id_bytes) {
    ncclUniqueId id;
    std::memcpy(id.internal, id_bytes, NCCL_UNIQUE_ID_BYTES);
    ncclComm_t comm = nullptr;
    ncclResult_t r = ncclCommInitRank(&comm, nranks, id, rank);
    if (ret(r) != 0) return 0;
    return reinterpret_cast<long long>(comm);
}

DDLW_EXPORT int ddlw_rccl_comm_destroy(long long comm) {
    return ret(ncclCommDestroy(reinterpret_cast<ncclComm_t>(comm)));
}

// In-place all-reduce on `stream`. op: 0=sum, 1=avg (single fused pass).
DDLW_EXPORT int ddlw_rccl_allreduce(long long comm, void* buf, long long count,
                                    int dtype, int op, void* stream) {
    bool ok;
    ncclDataType_t t = to_nccl(dtype, &ok);
    if (!ok) return -1;
    return ret(ncclAllReduce(buf, buf, static_cast<size_t>(count), t,
                             op == 1 ? ncclAvg : ncclSum,
                             reinterpret_cast<ncclComm_t>(comm),
                             reinterpret_cast<hipStream_t>(stream)));
}

DDLW_EXPORT int ddlw_rccl_broadcast(long long comm, void* buf, long long count,
                                    int dtype, int root, void* stream) {
    bool ok;
    ncclDataType_t t = to_nccl(dtype, &ok);
    if (!ok) return -1;
    return ret(ncclBroadcast(buf, buf, static_cast<size_t>(count), t, root,
                             reinterpret_cast<ncclComm_t>(comm),
                             reinterpret_cast<hipStream_t>(stream)));
}

DDLW_EXPORT int ddlw_rccl_reduce_scatter(long long comm, const void* sendbuf,
                                         void* recvbuf, long long recv_count,
                                         int dtype, int op, void* stream) {
    bool ok;
    ncclDataType_t t = to_nccl(dtype, &ok);
    if (!ok) return -1;
    return ret(ncclReduceScatter(sendbuf, recvbuf, static_cast<size_t>(recv_count), t,
                                 op == 1 ? ncclAvg : ncclSum,
                                 reinterpret_cast<ncclComm_t>(comm),
                                 reinterpret_cast<hipStream_t>(stream)));
}

DDLW_EXPORT int ddlw_rccl_allgather(long long comm, const void* sendbuf, void* recvbuf,
                                    long long send_count, int dtype, void* stream) {
    bool ok;
    ncclDataType_t t = to_nccl(dtype, &ok);
    if (!ok) return -1;
    return ret(ncclAllGather(sendbuf, recvbuf, static_cast<size_t>(send_count), t,
                             reinterpret_cast<ncclComm_t>(comm),
                             reinterpret_cast<hipStream_t>(stream)));
}

DDLW_EXPORT const char* ddlw_rccl_error_string(int code) {
    return ncclGetErrorString(static_cast<ncclResult_t>(code));
}

DDLW_EXPORT const char* ddlw_rccl_last_error() {
    return ncclGetErrorString(g_last);
}
