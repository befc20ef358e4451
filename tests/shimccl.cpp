// shimccl — TEST INFRASTRUCTURE ONLY.  An LD_PRELOAD mock of the RCCL
// symbol subset libconflux_lu.so uses, so the engine's distributed
// (!sim) branches — grouped send/recv choreography, depth reduces, the
// dual-comm lookahead, distributed validation — can EXECUTE as real
// multi-process runs on a single-GPU box (RCCL itself refuses two ranks
// on one device: "Duplicate GPU detected").  Transport is /dev/shm file
// mailboxes with rename-atomic publication; device buffers are staged
// through host memory.  Never linked into the product: the product links
// real librccl; this library only interposes in tests that set
// LD_PRELOAD.  It mirrors NCCL semantics the engine relies on:
//   * per-(comm, src->dst) FIFO matching in posted order,
//   * group semantics: no send blocks on its matching recv (all sends of
//     a group are staged+published before any recv is waited on),
//   * a size mismatch between a matched send/recv pair is a loud error
//     (a deliberate bug detector the real RCCL does not give you),
//   * ncclCommSplit(color 0, key = rank) clones the communicator into a
//     fresh namespace (the engine's pcomm).
// Cleanup: the test harness owns SHIMCCL_DIR and removes it afterwards.
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>
#include <sys/stat.h>
#include <unistd.h>

#include <atomic>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <map>
#include <mutex>
#include <string>
#include <vector>

namespace {

struct Comm {
    int rank = 0, world = 1, idx = 0;
    int nsplit = 0;
    std::string dir;
    std::vector<uint64_t> sseq, rseq;  // per-peer FIFO counters
    uint64_t arseq = 0, bcseq = 0;
};

struct Op {
    bool is_send;
    const void *sbuf;
    void *rbuf;
    size_t bytes;
    int peer;
    Comm *comm;
    hipStream_t stream;
};

thread_local int g_depth = 0;
thread_local std::vector<Op> g_ops;

size_t dtsize(ncclDataType_t t) {
    switch (t) {
        case ncclInt8:
        case ncclUint8:
            return 1;
        case ncclFloat16:
        case ncclBfloat16:
            return 2;
        case ncclInt32:
        case ncclUint32:
        case ncclFloat32:
            return 4;
        default:
            return 8;
    }
}

std::string msgpath(Comm *c, int src, int dst, uint64_t seq) {
    char b[256];
    std::snprintf(b, sizeof b, "%s/c%d_s%d_d%d_%llu", c->dir.c_str(), c->idx,
                  src, dst, (unsigned long long)seq);
    return b;
}

int write_atomic(const std::string &path, const void *data, size_t bytes) {
    const std::string tmp = path + ".tmp";
    FILE *f = std::fopen(tmp.c_str(), "wb");
    if (!f) return -1;
    if (bytes && std::fwrite(data, 1, bytes, f) != bytes) {
        std::fclose(f);
        return -1;
    }
    std::fclose(f);
    return std::rename(tmp.c_str(), path.c_str());
}

// Poll for `path` to appear with EXACTLY `bytes` bytes; a different size is
// a matched-pair size mismatch and fails loudly.  `remove`: consume it.
int read_blocking(const std::string &path, void *data, size_t bytes,
                  bool remove) {
    for (int t = 0; t < 1200000; ++t) {  // ~120 s at 100 us
        struct stat st;
        if (stat(path.c_str(), &st) == 0) {
            if ((size_t)st.st_size != bytes) {
                std::fprintf(stderr,
                             "[shimccl] SIZE MISMATCH %s: posted %zu, "
                             "published %lld\n",
                             path.c_str(), bytes, (long long)st.st_size);
                return -1;
            }
            FILE *f = std::fopen(path.c_str(), "rb");
            if (!f) return -1;
            const size_t got = bytes ? std::fread(data, 1, bytes, f) : 0;
            std::fclose(f);
            if (got != bytes) return -1;
            if (remove) unlink(path.c_str());
            return 0;
        }
        usleep(100);
    }
    std::fprintf(stderr, "[shimccl] TIMEOUT waiting for %s\n", path.c_str());
    return -1;
}

ncclResult_t stage_d2h(const void *dev, void *host, size_t bytes,
                       hipStream_t s) {
    if (!bytes) return ncclSuccess;
    if (hipMemcpyAsync(host, dev, bytes, hipMemcpyDeviceToHost, s) !=
            hipSuccess ||
        hipStreamSynchronize(s) != hipSuccess)
        return ncclUnhandledCudaError;
    return ncclSuccess;
}

ncclResult_t stage_h2d(const void *host, void *dev, size_t bytes,
                       hipStream_t s) {
    if (!bytes) return ncclSuccess;
    if (hipMemcpyAsync(dev, host, bytes, hipMemcpyHostToDevice, s) !=
            hipSuccess ||
        hipStreamSynchronize(s) != hipSuccess)
        return ncclUnhandledCudaError;
    return ncclSuccess;
}

ncclResult_t flush_ops_async();
bool shim_async();

ncclResult_t flush_ops() {
    if (shim_async()) return flush_ops_async();
    // sends first (stage + publish), then recvs — NCCL group semantics:
    // no send blocks on its matching recv
    for (auto &op : g_ops) {
        if (!op.is_send) continue;
        std::vector<char> h(op.bytes);
        ncclResult_t r = stage_d2h(op.sbuf, h.data(), op.bytes, op.stream);
        if (r != ncclSuccess) return r;
        const uint64_t seq = op.comm->sseq[op.peer]++;
        if (write_atomic(msgpath(op.comm, op.comm->rank, op.peer, seq),
                         h.data(), op.bytes))
            return ncclSystemError;
    }
    for (auto &op : g_ops) {
        if (op.is_send) continue;
        std::vector<char> h(op.bytes);
        const uint64_t seq = op.comm->rseq[op.peer]++;
        if (read_blocking(msgpath(op.comm, op.peer, op.comm->rank, seq),
                          h.data(), op.bytes, /*remove=*/true))
            return ncclSystemError;
        ncclResult_t r = stage_h2d(h.data(), op.rbuf, op.bytes, op.stream);
        if (r != ncclSuccess) return r;
    }
    g_ops.clear();
    return ncclSuccess;
}

ncclResult_t enqueue(Op op) {
    g_ops.push_back(op);
    if (g_depth == 0) return flush_ops();
    return ncclSuccess;
}

// ---------------------------------------------------------------------------
// SHIMCCL_ASYNC=1: stream-enqueued transport via hipLaunchHostFunc, matching
// real RCCL's completion semantics — send/recv return immediately and the
// blocking exchange happens when the STREAM reaches the host function, so
// the engine's multi-stream choreography (dual-comm lookahead, event
// gating) is exercised with true asynchronous ordering instead of the
// host-blocking default.  Data stages through a pinned-buffer pool; a
// callback that times out or hits I/O failure abort()s the rank (loud).
// Collectives (allreduce/broadcast) stay host-blocking: they sit on barrier
// paths where the host is about to synchronize anyway.
// ---------------------------------------------------------------------------
struct SlotPool {
    std::mutex m;
    std::multimap<size_t, void *> free_;
    void *get(size_t n) {
        {
            std::lock_guard<std::mutex> g(m);
            auto it = free_.lower_bound(n);
            if (it != free_.end() && it->first <= 2 * n + 4096) {
                void *p = it->second;
                free_.erase(it);
                return p;
            }
        }
        void *p = nullptr;
        if (hipHostMalloc(&p, n ? n : 8) != hipSuccess) return nullptr;
        return p;
    }
    void put(void *p, size_t n) {
        std::lock_guard<std::mutex> g(m);
        free_.emplace(n, p);
    }
};
SlotPool g_pool;

struct HostOp {
    std::string path;
    size_t bytes;
    void *slot;
    bool remove;
};

void cb_publish(void *ud) {  // send: slot -> file, release slot
    auto *h = (HostOp *)ud;
    if (write_atomic(h->path, h->slot, h->bytes)) {
        std::fprintf(stderr, "[shimccl async] publish failed %s\n",
                     h->path.c_str());
        abort();
    }
    g_pool.put(h->slot, h->bytes);
    delete h;
}

void cb_wait_read(void *ud) {  // recv: file -> slot (slot released later)
    auto *h = (HostOp *)ud;
    if (read_blocking(h->path, h->slot, h->bytes, h->remove)) {
        std::fprintf(stderr, "[shimccl async] recv failed %s\n",
                     h->path.c_str());
        abort();
    }
    delete h;
}

void cb_release(void *ud) {
    auto *h = (HostOp *)ud;
    g_pool.put(h->slot, h->bytes);
    delete h;
}

bool shim_async() {
    static int v = -1;
    if (v < 0) {
        const char *e = getenv("SHIMCCL_ASYNC");
        v = e ? atoi(e) : 0;
    }
    return v != 0;
}

ncclResult_t flush_ops_async() {
    // sends first, then recvs (NCCL group semantics), each as a chain of
    // stream ops: no host blocking here at all
    for (auto &op : g_ops) {
        if (!op.is_send) continue;
        void *slot = g_pool.get(op.bytes);
        if (!slot) return ncclSystemError;
        const uint64_t seq = op.comm->sseq[op.peer]++;
        if (op.bytes &&
            hipMemcpyAsync(slot, op.sbuf, op.bytes, hipMemcpyDeviceToHost,
                           op.stream) != hipSuccess)
            return ncclUnhandledCudaError;
        auto *h = new HostOp{msgpath(op.comm, op.comm->rank, op.peer, seq),
                             op.bytes, slot, false};
        if (hipLaunchHostFunc(op.stream, cb_publish, h) != hipSuccess)
            return ncclUnhandledCudaError;
    }
    for (auto &op : g_ops) {
        if (op.is_send) continue;
        void *slot = g_pool.get(op.bytes);
        if (!slot) return ncclSystemError;
        const uint64_t seq = op.comm->rseq[op.peer]++;
        auto *h = new HostOp{msgpath(op.comm, op.peer, op.comm->rank, seq),
                             op.bytes, slot, true};
        if (hipLaunchHostFunc(op.stream, cb_wait_read, h) != hipSuccess)
            return ncclUnhandledCudaError;
        if (op.bytes &&
            hipMemcpyAsync(op.rbuf, slot, op.bytes, hipMemcpyHostToDevice,
                           op.stream) != hipSuccess)
            return ncclUnhandledCudaError;
        auto *h2 = new HostOp{"", op.bytes, slot, false};
        if (hipLaunchHostFunc(op.stream, cb_release, h2) != hipSuccess)
            return ncclUnhandledCudaError;
    }
    g_ops.clear();
    return ncclSuccess;
}

}  // namespace

extern "C" {

ncclResult_t ncclGetUniqueId(ncclUniqueId *id) {
    static int counter = 0;
    char dir[100];
    if (const char *base = getenv("SHIMCCL_DIR"))
        std::snprintf(dir, sizeof dir, "%s", base);
    else
        std::snprintf(dir, sizeof dir, "/dev/shm/shimccl_%d_%d",
                      (int)getpid(), counter++);
    mkdir(dir, 0777);
    std::memset(id, 0, sizeof *id);
    std::snprintf(id->internal, sizeof id->internal, "%s", dir);
    return ncclSuccess;
}

ncclResult_t ncclCommInitRank(ncclComm_t *comm, int nranks, ncclUniqueId id,
                              int rank) {
    auto *c = new Comm;
    c->rank = rank;
    c->world = nranks;
    c->idx = 0;
    c->dir = id.internal;
    mkdir(c->dir.c_str(), 0777);
    c->sseq.assign(nranks, 0);
    c->rseq.assign(nranks, 0);
    *comm = (ncclComm_t)c;
    return ncclSuccess;
}

ncclResult_t ncclCommSplit(ncclComm_t comm, int color, int key,
                           ncclComm_t *newcomm, ncclConfig_t *) {
    auto *p = (Comm *)comm;
    if (color != 0 || key != p->rank) return ncclInvalidArgument;  // engine use
    auto *c = new Comm;
    c->rank = p->rank;
    c->world = p->world;
    c->dir = p->dir;
    c->idx = 100 * (p->idx + 1) + p->nsplit++;  // deterministic namespace
    c->sseq.assign(p->world, 0);
    c->rseq.assign(p->world, 0);
    *newcomm = (ncclComm_t)c;
    return ncclSuccess;
}

ncclResult_t ncclCommDestroy(ncclComm_t comm) {
    delete (Comm *)comm;  // files belong to the harness's SHIMCCL_DIR
    return ncclSuccess;
}

ncclResult_t ncclGroupStart() {
    ++g_depth;
    return ncclSuccess;
}

ncclResult_t ncclGroupEnd() {
    if (g_depth > 0) --g_depth;
    if (g_depth == 0) return flush_ops();
    return ncclSuccess;
}

ncclResult_t ncclSend(const void *sendbuff, size_t count,
                      ncclDataType_t datatype, int peer, ncclComm_t comm,
                      hipStream_t stream) {
    return enqueue({true, sendbuff, nullptr, count * dtsize(datatype), peer,
                    (Comm *)comm, stream});
}

ncclResult_t ncclRecv(void *recvbuff, size_t count, ncclDataType_t datatype,
                      int peer, ncclComm_t comm, hipStream_t stream) {
    return enqueue({false, nullptr, recvbuff, count * dtsize(datatype), peer,
                    (Comm *)comm, stream});
}

ncclResult_t ncclAllReduce(const void *sendbuff, void *recvbuff, size_t count,
                           ncclDataType_t datatype, ncclRedOp_t op,
                           ncclComm_t comm, hipStream_t stream) {
    auto *c = (Comm *)comm;
    if (datatype != ncclDouble || op != ncclSum) return ncclInvalidArgument;
    const size_t bytes = count * 8;
    std::vector<double> mine(count), acc(count, 0.0), other(count);
    ncclResult_t r = stage_d2h(sendbuff, mine.data(), bytes, stream);
    if (r != ncclSuccess) return r;
    const uint64_t seq = c->arseq++;
    char b[256];
    std::snprintf(b, sizeof b, "%s/ar%d_%llu_r%d", c->dir.c_str(), c->idx,
                  (unsigned long long)seq, c->rank);
    if (write_atomic(b, mine.data(), bytes)) return ncclSystemError;
    for (int p = 0; p < c->world; ++p) {  // deterministic rank-ascending sum
        std::snprintf(b, sizeof b, "%s/ar%d_%llu_r%d", c->dir.c_str(), c->idx,
                      (unsigned long long)seq, p);
        if (read_blocking(b, other.data(), bytes, /*remove=*/false))
            return ncclSystemError;
        for (size_t i = 0; i < count; ++i) acc[i] += other[i];
    }
    return stage_h2d(acc.data(), recvbuff, bytes, stream);
}

ncclResult_t ncclBroadcast(const void *sendbuff, void *recvbuff, size_t count,
                           ncclDataType_t datatype, int root, ncclComm_t comm,
                           hipStream_t stream) {
    auto *c = (Comm *)comm;
    const size_t bytes = count * dtsize(datatype);
    const uint64_t seq = c->bcseq++;
    char b[256];
    std::snprintf(b, sizeof b, "%s/bc%d_%llu", c->dir.c_str(), c->idx,
                  (unsigned long long)seq);
    std::vector<char> h(bytes);
    if (c->rank == root) {
        ncclResult_t r = stage_d2h(sendbuff, h.data(), bytes, stream);
        if (r != ncclSuccess) return r;
        if (write_atomic(b, h.data(), bytes)) return ncclSystemError;
        return stage_h2d(h.data(), recvbuff, bytes, stream);
    }
    if (read_blocking(b, h.data(), bytes, /*remove=*/false))
        return ncclSystemError;
    return stage_h2d(h.data(), recvbuff, bytes, stream);
}

const char *ncclGetErrorString(ncclResult_t code) {
    static char b[64];
    std::snprintf(b, sizeof b, "shimccl mock error %d", (int)code);
    return b;
}

}  // extern "C"
