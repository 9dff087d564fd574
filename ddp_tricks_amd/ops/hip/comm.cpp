// comm.cpp — native RCCL communicator layer.
//
// The reference leans on torch's ProcessGroupNCCL (invoked at reference
// run.py:28); here the DDP reducer talks to RCCL directly: a ncclUniqueId
// rendezvous (bootstrapped over any byte channel — the Python side uses the
// torch TCP store), then ncclAllReduce/ncclBroadcast enqueued on whatever
// HIP stream is current (the reducer's side stream), with the 1/N average
// fused into the reduction via ncclAvg.  Collectives for multiple buckets
// can be batched inside a ncclGroupStart/End window so RCCL schedules them
// across the 7 xGMI links together.
#include <torch/extension.h>

#ifdef USE_ROCM
#include <ATen/hip/HIPContext.h>
#include <rccl/rccl.h>

#include <mutex>
#include <vector>

#define RCCL_CHECK(cmd)                                                    \
    do {                                                                   \
        ncclResult_t r = (cmd);                                            \
        TORCH_CHECK(r == ncclSuccess, "RCCL error: ",                      \
                    ncclGetErrorString(r), " at ", __FILE__, ":", __LINE__); \
    } while (0)

static std::vector<ncclComm_t> g_comms;
static std::mutex g_mu;

static ncclDataType_t nccl_dtype(const at::Tensor& t) {
    switch (t.scalar_type()) {
        case at::kFloat: return ncclFloat32;
        case at::kBFloat16: return ncclBfloat16;
        case at::kHalf: return ncclFloat16;
        case at::kLong: return ncclInt64;
        case at::kInt: return ncclInt32;
        case at::kDouble: return ncclFloat64;
        default: TORCH_CHECK(false, "unsupported dtype for RCCL collective");
    }
}

py::bytes rccl_unique_id() {
    ncclUniqueId id;
    RCCL_CHECK(ncclGetUniqueId(&id));
    return py::bytes(reinterpret_cast<const char*>(&id), sizeof(id));
}

int64_t rccl_comm_init(int64_t nranks, int64_t rank, py::bytes id_bytes) {
    std::string s = id_bytes;
    TORCH_CHECK(s.size() == sizeof(ncclUniqueId), "bad unique id size");
    ncclUniqueId id;
    std::memcpy(&id, s.data(), sizeof(id));
    ncclComm_t comm;
    RCCL_CHECK(ncclCommInitRank(&comm, (int)nranks, id, (int)rank));
    std::lock_guard<std::mutex> lk(g_mu);
    g_comms.push_back(comm);
    return (int64_t)g_comms.size() - 1;
}

static ncclComm_t get_comm(int64_t h) {
    std::lock_guard<std::mutex> lk(g_mu);
    TORCH_CHECK(h >= 0 && (size_t)h < g_comms.size() && g_comms[h],
                "invalid RCCL comm handle");
    return g_comms[h];
}

void rccl_all_reduce(at::Tensor t, int64_t handle, bool avg) {
    TORCH_CHECK(t.is_cuda() && t.is_contiguous());
    auto stream = at::hip::getCurrentHIPStream();
    RCCL_CHECK(ncclAllReduce(t.data_ptr(), t.data_ptr(), t.numel(),
                             nccl_dtype(t), avg ? ncclAvg : ncclSum,
                             get_comm(handle), stream.stream()));
}

void rccl_broadcast(at::Tensor t, int64_t root, int64_t handle) {
    TORCH_CHECK(t.is_cuda() && t.is_contiguous());
    auto stream = at::hip::getCurrentHIPStream();
    RCCL_CHECK(ncclBroadcast(t.data_ptr(), t.data_ptr(), t.numel(),
                             nccl_dtype(t), (int)root, get_comm(handle),
                             stream.stream()));
}

void rccl_group_start() { RCCL_CHECK(ncclGroupStart()); }
void rccl_group_end() { RCCL_CHECK(ncclGroupEnd()); }

void rccl_comm_destroy(int64_t handle) {
    std::lock_guard<std::mutex> lk(g_mu);
    if (handle >= 0 && (size_t)handle < g_comms.size() && g_comms[handle]) {
        ncclCommDestroy(g_comms[handle]);
        g_comms[handle] = nullptr;
    }
}
#endif  // USE_ROCM

void register_comm(py::module_& m) {
#ifdef USE_ROCM
    m.def("rccl_unique_id", &rccl_unique_id);
    m.def("rccl_comm_init", &rccl_comm_init);
    m.def("rccl_all_reduce", &rccl_all_reduce,
          py::arg("t"), py::arg("handle"), py::arg("avg") = false);
    m.def("rccl_broadcast", &rccl_broadcast);
    m.def("rccl_group_start", &rccl_group_start);
    m.def("rccl_group_end", &rccl_group_end);
    m.def("rccl_comm_destroy", &rccl_comm_destroy);
#endif
}
