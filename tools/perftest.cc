/* ucc_perftest — collective micro-benchmark over the public C API.
 *
 * Drop-in equivalent of the reference tools/perf/ucc_perftest
 * (ucc_pt_benchmark.cc size sweep + per-coll bus-BW formulas
 * ucc_pt_coll_allreduce.cc:84-93 / ucc_pt_coll_alltoall.cc:187-196),
 * MPI-free: bootstrap is either
 *   (a) fork mode (-p N): N processes, POSIX-shm allgather OOB
 *       (tools/shm_oob.h), HIP device = rank % device_count — the
 *       production single-node MI355X topology, or
 *   (b) in-process mode (default): N full lib/ctx/team stacks driven
 *       round-robin in one process (the gtest UccJob design) — runs
 *       anywhere, including CPU-only.
 *
 * Usage: ucc_perftest [-c coll] [-b min] [-e max] [-n iters] [-w warmup]
 *                     [-m host|cuda] [-d dtype] [-o op] [-p nprocs]
 *                     [-j nranks] [-F] [-i] [-C]
 */
#include <algorithm>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <map>
#include <ctime>
#include <string>
#include <vector>

#include <sys/wait.h>
#include <unistd.h>

#include "../src/api/ucc.h"
#include "shm_oob.h"

#ifdef UCC_AMD_HAS_HIP
#include <hip/hip_runtime.h>
#endif

using uccperf::ShmOob;

#ifdef UCC_AMD_HAS_HIP
static inline void pt_hip_warn(hipError_t e, const char *what)
{
    if (e != hipSuccess) {
        fprintf(stderr, "hip warning: %s: %s\n", what,
                hipGetErrorString(e));
    }
}
#define HIPWARN(expr) pt_hip_warn((expr), #expr)
#endif


/* ------------------------------------------------------------------ opts */
struct Opts {
    std::string coll = "allreduce";
    size_t      min_b = 8, max_b = 1 << 22;
    int         iters = 200, warmup = 20, large_iters = 40;
    std::string mem = "host";
    std::string dtype = "float32";
    std::string op    = "sum";
    int         nprocs = 0; /* 0 = in-process */
    int         nranks = 2;
    bool        skew = false; /* alltoallv: MoE-like skewed matrix */
    bool        root_shift = false; /* rooted colls: rotate root */
    bool        persistent = false, inplace = false, check = false;
    bool        triggered = false;
};

static ucc_datatype_t dt_from_name(const std::string &s)
{
    if (s == "int8") return UCC_DT_INT8;
    if (s == "uint8") return UCC_DT_UINT8;
    if (s == "int16") return UCC_DT_INT16;
    if (s == "int32") return UCC_DT_INT32;
    if (s == "int64") return UCC_DT_INT64;
    if (s == "float16" || s == "fp16") return UCC_DT_FLOAT16;
    if (s == "bfloat16" || s == "bf16") return UCC_DT_BFLOAT16;
    if (s == "float32" || s == "fp32") return UCC_DT_FLOAT32;
    if (s == "float64" || s == "fp64") return UCC_DT_FLOAT64;
    if (s == "float8_e4m3" || s == "fp8") return UCC_DT_FLOAT8_E4M3;
    fprintf(stderr, "unknown dtype %s\n", s.c_str());
    exit(1);
}

static ucc_reduction_op_t op_from_name(const std::string &s)
{
    if (s == "sum") return UCC_OP_SUM;
    if (s == "prod") return UCC_OP_PROD;
    if (s == "max") return UCC_OP_MAX;
    if (s == "min") return UCC_OP_MIN;
    if (s == "avg") return UCC_OP_AVG;
    fprintf(stderr, "unknown op %s\n", s.c_str());
    exit(1);
}

static const struct {
    const char     *name;
    ucc_coll_type_t ct;
} kColls[] = {
    {"allreduce", UCC_COLL_TYPE_ALLREDUCE},
    {"allgather", UCC_COLL_TYPE_ALLGATHER},
    {"allgatherv", UCC_COLL_TYPE_ALLGATHERV},
    {"alltoall", UCC_COLL_TYPE_ALLTOALL},
    {"alltoallv", UCC_COLL_TYPE_ALLTOALLV},
    {"barrier", UCC_COLL_TYPE_BARRIER},
    {"bcast", UCC_COLL_TYPE_BCAST},
    {"gather", UCC_COLL_TYPE_GATHER},
    {"reduce", UCC_COLL_TYPE_REDUCE},
    {"reduce_scatter", UCC_COLL_TYPE_REDUCE_SCATTER},
    {"reduce_scatterv", UCC_COLL_TYPE_REDUCE_SCATTERV},
    {"scatter", UCC_COLL_TYPE_SCATTER},
    {"gatherv", UCC_COLL_TYPE_GATHERV},
    {"scatterv", UCC_COLL_TYPE_SCATTERV},
};

static ucc_coll_type_t coll_from_name(const std::string &s)
{
    for (auto &c : kColls) {
        if (s == c.name) {
            return c.ct;
        }
    }
    fprintf(stderr, "unknown coll %s\n", s.c_str());
    exit(1);
}

/* bus-BW factor: allreduce 2(N-1)/N, alltoall/allgather/rs/bcast (N-1)/N
 * (reference ucc_pt_coll_*.cc formulas) */
static double busbw_factor(ucc_coll_type_t ct, int n)
{
    switch (ct) {
    case UCC_COLL_TYPE_ALLREDUCE:
        return 2.0 * (n - 1) / n;
    case UCC_COLL_TYPE_ALLGATHER:
    case UCC_COLL_TYPE_ALLGATHERV:
    case UCC_COLL_TYPE_ALLTOALL:
    case UCC_COLL_TYPE_ALLTOALLV:
    case UCC_COLL_TYPE_REDUCE_SCATTER:
    case UCC_COLL_TYPE_REDUCE_SCATTERV:
        return (double)(n - 1) / n;
    default:
        return 1.0;
    }
}

static double now_s()
{
    struct timespec ts;
    clock_gettime(CLOCK_MONOTONIC, &ts);
    return ts.tv_sec + 1e-9 * ts.tv_nsec;
}

/* --------------------------------------------------------------- buffers */
struct Bufs {
    void  *src = nullptr, *dst = nullptr;
    bool   dev = false;
    size_t cap = 0;

    void alloc(size_t bytes, bool device)
    {
        dev = device;
        cap = bytes;
        if (device) {
#ifdef UCC_AMD_HAS_HIP
            if (hipMalloc(&src, bytes) != hipSuccess ||
                hipMalloc(&dst, bytes) != hipSuccess) {
                fprintf(stderr, "hipMalloc failed\n");
                exit(1);
            }
            HIPWARN(hipMemset(src, 1, bytes));
            HIPWARN(hipMemset(dst, 0, bytes));
            HIPWARN(hipDeviceSynchronize());
#else
            fprintf(stderr, "built without HIP\n");
            exit(1);
#endif
        } else {
            src = malloc(bytes);
            dst = malloc(bytes);
            memset(src, 1, bytes);
            memset(dst, 0, bytes);
        }
    }
    ~Bufs()
    {
        if (dev) {
#ifdef UCC_AMD_HAS_HIP
            HIPWARN(hipFree(src));
            HIPWARN(hipFree(dst));
#endif
        } else {
            free(src);
            free(dst);
        }
    }
};

/* ------------------------------------------------------------- rank state */
struct Rank {
    ucc_lib_h     lib = nullptr;
    ucc_context_h ctx = nullptr;
    ucc_team_h    team = nullptr;
    Bufs          bufs;
};

/* in-process OOB: memcpy allgather across the rank array. Rounds are
 * keyed by per-rank sequence number (ranks advance through team-create
 * rounds at different paces, so a single blob slot per rank would get
 * overwritten before slower ranks copied it). */
struct LocalOobState {
    struct Round {
        std::vector<std::vector<uint8_t>> blobs;
        int                               arrived = 0;
    };
    std::map<uint64_t, Round> rounds;
    std::vector<uint64_t>     next_round;
    int                       n;
};
static LocalOobState g_loob;

struct LocalOobReq {
    int      rank;
    uint64_t round;
    void    *recv;
    size_t   size;
};

static ucc_status_t loob_allgather(void *src, void *recv, size_t size,
                                   void *info, void **req)
{
    int      rank  = (int)(intptr_t)info;
    uint64_t round = g_loob.next_round[rank]++;
    auto    &r     = g_loob.rounds[round];
    if (r.blobs.empty()) {
        r.blobs.resize(g_loob.n);
    }
    r.blobs[rank].assign((uint8_t *)src, (uint8_t *)src + size);
    r.arrived++;
    *req = new LocalOobReq{rank, round, recv, size};
    return UCC_OK;
}
static ucc_status_t loob_test(void *req)
{
    auto *r  = (LocalOobReq *)req;
    auto &rd = g_loob.rounds[r->round];
    if (rd.arrived < g_loob.n) {
        return UCC_INPROGRESS;
    }
    for (int i = 0; i < g_loob.n; i++) {
        memcpy((uint8_t *)r->recv + (size_t)i * r->size,
               rd.blobs[i].data(), r->size);
    }
    return UCC_OK;
}
static ucc_status_t loob_free(void *req)
{
    delete (LocalOobReq *)req;
    return UCC_OK;
}

/* fork-mode OOB wrappers */
static ShmOob g_shm_oob;
static ucc_status_t soob_allgather(void *src, void *recv, size_t size,
                                   void *info, void **req)
{
    (void)info;
    g_shm_oob.allgather(src, recv, size);
    *req = (void *)1;
    return UCC_OK;
}
static ucc_status_t soob_test(void *req)
{
    (void)req;
    return UCC_OK;
}
static ucc_status_t soob_free(void *req)
{
    (void)req;
    return UCC_OK;
}

static void setup_rank(Rank &r, int rank, int nranks, bool forked)
{
    ucc_lib_params_t lp{};
    lp.mask        = UCC_LIB_PARAM_FIELD_THREAD_MODE;
    lp.thread_mode = UCC_THREAD_SINGLE;
    if (ucc_init(&lp, nullptr, &r.lib) != UCC_OK) {
        fprintf(stderr, "ucc_init failed\n");
        exit(1);
    }
    ucc_context_params_t cp{};
    cp.mask = 0;
    if (ucc_context_create(r.lib, &cp, nullptr, &r.ctx) != UCC_OK) {
        fprintf(stderr, "ucc_context_create failed\n");
        exit(1);
    }
    ucc_team_params_t tp{};
    tp.mask          = UCC_TEAM_PARAM_FIELD_OOB;
    tp.oob.allgather = forked ? soob_allgather : loob_allgather;
    tp.oob.req_test  = forked ? soob_test : loob_test;
    tp.oob.req_free  = forked ? soob_free : loob_free;
    tp.oob.coll_info = forked ? nullptr : (void *)(intptr_t)rank;
    tp.oob.n_oob_eps = nranks;
    tp.oob.oob_ep    = rank;
    if (ucc_team_create_post(&r.ctx, 1, &tp, &r.team) != UCC_OK) {
        fprintf(stderr, "team_create_post failed\n");
        exit(1);
    }
}

static ucc_coll_args_t make_args(const Opts &o, Rank &r, size_t bytes,
                                 int rank, int nranks,
                                 std::vector<uint64_t> &cnts,
                                 std::vector<uint64_t> &dsps,
                                 std::vector<uint64_t> &rcnts,
                                 std::vector<uint64_t> &rdsps,
                                 int root = 0)
{
    ucc_coll_type_t    ct = coll_from_name(o.coll);
    ucc_datatype_t     dt = dt_from_name(o.dtype);
    ucc_reduction_op_t op = op_from_name(o.op);
    size_t             ds = ucc_dt_size(dt);
    size_t             count = bytes / ds;
    ucc_memory_type_t  mt = o.mem == "cuda" ? UCC_MEMORY_TYPE_CUDA
                                            : UCC_MEMORY_TYPE_HOST;
    /* block colls need count divisible by nranks (equal blocks is the
     * allgather/alltoall contract; reference perftest sizes the same
     * way) — round down, keeping at least one element per rank */
    switch (ct) {
    case UCC_COLL_TYPE_ALLGATHER:
    case UCC_COLL_TYPE_ALLTOALL:
    case UCC_COLL_TYPE_GATHER:
    case UCC_COLL_TYPE_SCATTER:
    case UCC_COLL_TYPE_REDUCE_SCATTER:
    case UCC_COLL_TYPE_ALLGATHERV:
    case UCC_COLL_TYPE_ALLTOALLV:
        count -= count % (size_t)nranks;
        if (count == 0) {
            count = (size_t)nranks;
        }
        break;
    default:
        break;
    }
    ucc_coll_args_t a{};
    a.mask      = UCC_COLL_ARGS_FIELD_FLAGS;
    a.flags     = (o.persistent ? UCC_COLL_ARGS_FLAG_PERSISTENT : 0) |
                  (o.inplace ? UCC_COLL_ARGS_FLAG_IN_PLACE : 0) |
                  UCC_COLL_ARGS_FLAG_COUNT_64BIT |
                  UCC_COLL_ARGS_FLAG_DISPLACEMENTS_64BIT;
    a.coll_type = ct;
    a.op        = op;
    a.root      = (uint64_t)root;
    a.src.info.buffer   = r.bufs.src;
    a.src.info.count    = count;
    a.src.info.datatype = dt;
    a.src.info.mem_type = mt;
    a.dst.info.buffer   = r.bufs.dst;
    a.dst.info.count    = count;
    a.dst.info.datatype = dt;
    a.dst.info.mem_type = mt;
    switch (ct) {
    case UCC_COLL_TYPE_ALLGATHER:
    case UCC_COLL_TYPE_ALLTOALL:
    case UCC_COLL_TYPE_GATHER:
    case UCC_COLL_TYPE_SCATTER:
        /* count = total (count per rank * nranks) per reference perftest */
        a.src.info.count = count;
        a.dst.info.count = count;
        break;
    case UCC_COLL_TYPE_REDUCE_SCATTER:
        a.dst.info.count = count / nranks;
        break;
    case UCC_COLL_TYPE_REDUCE_SCATTERV: {
        cnts.assign(nranks, count / nranks);
        dsps.resize(nranks);
        for (int i = 0; i < nranks; i++) {
            dsps[i] = (uint64_t)i * (count / nranks);
        }
        a.dst.info_v.buffer        = r.bufs.dst;
        a.dst.info_v.counts        = cnts.data();
        a.dst.info_v.displacements = dsps.data();
        a.dst.info_v.datatype      = dt;
        a.dst.info_v.mem_type      = mt;
        a.src.info.count           = count;
        break;
    }
    case UCC_COLL_TYPE_GATHERV: {
        cnts.assign(nranks, count / nranks);
        dsps.resize(nranks);
        for (int i = 0; i < nranks; i++) {
            dsps[i] = (uint64_t)i * (count / nranks);
        }
        a.src.info.count           = count / nranks;
        a.dst.info_v.buffer        = r.bufs.dst;
        a.dst.info_v.counts        = cnts.data();
        a.dst.info_v.displacements = dsps.data();
        a.dst.info_v.datatype      = dt;
        a.dst.info_v.mem_type      = mt;
        break;
    }
    case UCC_COLL_TYPE_SCATTERV: {
        cnts.assign(nranks, count / nranks);
        dsps.resize(nranks);
        for (int i = 0; i < nranks; i++) {
            dsps[i] = (uint64_t)i * (count / nranks);
        }
        a.src.info_v.buffer        = r.bufs.src;
        a.src.info_v.counts        = cnts.data();
        a.src.info_v.displacements = dsps.data();
        a.src.info_v.datatype      = dt;
        a.src.info_v.mem_type      = mt;
        a.dst.info.count           = count / nranks;
        break;
    }
    case UCC_COLL_TYPE_ALLGATHERV: {
        cnts.assign(nranks, count / nranks);
        dsps.resize(nranks);
        for (int i = 0; i < nranks; i++) {
            dsps[i] = (uint64_t)i * (count / nranks);
        }
        a.dst.info_v.buffer        = r.bufs.dst;
        a.dst.info_v.counts        = cnts.data();
        a.dst.info_v.displacements = dsps.data();
        a.dst.info_v.datatype      = dt;
        a.dst.info_v.mem_type      = mt;
        a.src.info.count           = count / nranks;
        break;
    }
    case UCC_COLL_TYPE_ALLTOALLV: {
        /* -M: skewed per-peer counts (traffic-matrix generator role,
         * reference tools/perf/generator): rank r sends to peer d a
         * share proportional to (r+1)*(d+1), normalized to count */
        cnts.resize(nranks);
        dsps.resize(nranks);
        if (o.skew) {
            size_t denom = 0;
            for (int d = 0; d < nranks; d++) {
                denom += (size_t)(rank + 1) * (d + 1);
            }
            size_t off2 = 0;
            for (int d = 0; d < nranks; d++) {
                cnts[d] = count * ((size_t)(rank + 1) * (d + 1)) / denom;
                dsps[d] = off2;
                off2 += cnts[d];
            }
        } else {
            for (int i = 0; i < nranks; i++) {
                cnts[i] = count / nranks;
                dsps[i] = (uint64_t)i * (count / nranks);
            }
        }
        a.src.info_v.buffer        = r.bufs.src;
        a.src.info_v.counts        = cnts.data();
        a.src.info_v.displacements = dsps.data();
        a.src.info_v.datatype      = dt;
        a.src.info_v.mem_type      = mt;
        rcnts.resize(nranks);
        rdsps.resize(nranks);
        if (o.skew) {
            size_t off2 = 0;
            for (int s = 0; s < nranks; s++) {
                size_t denom = 0;
                for (int d = 0; d < nranks; d++) {
                    denom += (size_t)(s + 1) * (d + 1);
                }
                rcnts[s] = count * ((size_t)(s + 1) * (rank + 1)) / denom;
                rdsps[s] = off2;
                off2 += rcnts[s];
            }
        } else {
            rcnts = cnts;
            rdsps = dsps;
        }
        a.dst.info_v.buffer        = r.bufs.dst;
        a.dst.info_v.counts        = rcnts.data();
        a.dst.info_v.displacements = rdsps.data();
        a.dst.info_v.datatype      = dt;
        a.dst.info_v.mem_type      = mt;
        break;
    }
    default:
        break;
    }
    return a;
}

/* golden fp32 patterns shared by the fork- and inproc-mode -C checks */
static void vpattern(ucc_coll_type_t ct, int rank, size_t count,
                     std::vector<float> &h)
{
    h.resize(count);
    switch (ct) {
    case UCC_COLL_TYPE_ALLREDUCE:
        for (size_t i = 0; i < count; i++) {
            h[i] = (float)(rank + 1) + 0.25f * (float)(i % 7);
        }
        break;
    case UCC_COLL_TYPE_BCAST:
        for (size_t i = 0; i < count; i++) {
            h[i] = rank == 0 ? (float)(i % 97) : -1.0f;
        }
        break;
    default: /* allgather / reduce_scatter family */
        for (size_t i = 0; i < count; i++) {
            h[i] = (float)(rank + 1) + 0.5f * (float)(i % 5);
        }
        break;
    }
}

static size_t vcheck(ucc_coll_type_t ct, int rank, int nranks,
                     size_t count, const float *out)
{
    size_t bad = 0;
    if (ct == UCC_COLL_TYPE_ALLREDUCE) {
        for (size_t i = 0; i < count; i++) {
            float s = 0;
            for (int k = 0; k < nranks; k++) {
                s += (float)(k + 1) + 0.25f * (float)(i % 7);
            }
            bad += out[i] != s;
        }
    } else if (ct == UCC_COLL_TYPE_ALLGATHER) {
        size_t per = count / nranks;
        for (size_t i = 0; i < per * (size_t)nranks; i++) {
            int    src = (int)(i / per);
            size_t li  = i % per;
            bad += out[i] !=
                   (float)(src + 1) + 0.5f * (float)(li % 5);
        }
    } else if (ct == UCC_COLL_TYPE_REDUCE_SCATTER) {
        size_t per = count / nranks;
        for (size_t i = 0; i < per; i++) {
            size_t gi = (size_t)rank * per + i;
            float  e  = 0;
            for (int k = 0; k < nranks; k++) {
                e += (float)(k + 1) + 0.5f * (float)(gi % 5);
            }
            bad += out[i] != e;
        }
    } else if (ct == UCC_COLL_TYPE_BCAST) {
        for (size_t i = 0; i < count; i++) {
            bad += out[i] != (float)(i % 97);
        }
    }
    return bad;
}

/* -C validation: run one iteration with a known fp32 pattern and check
 * the result (allreduce/allgather/bcast/reduce_scatter). Returns false
 * on mismatch. Works for host and device memory (staged via host). */
static bool validate_once(const Opts &o, Rank &r, ucc_coll_req_h req,
                          ucc_coll_type_t ct, size_t bytes, int rank,
                          int nranks)
{
    if (o.dtype != "float32" || o.inplace) {
        return true; /* only the fp32 non-inplace patterns are checked */
    }
    size_t count = bytes / 4;
    if (count == 0) {
        return true;
    }
    if (ct != UCC_COLL_TYPE_ALLREDUCE && ct != UCC_COLL_TYPE_ALLGATHER &&
        ct != UCC_COLL_TYPE_REDUCE_SCATTER && ct != UCC_COLL_TYPE_BCAST) {
        return true;
    }
    std::vector<float> h, out(count, 0.0f);
    vpattern(ct, rank, count, h);
#ifdef UCC_AMD_HAS_HIP
    if (o.mem == "cuda") {
        HIPWARN(hipMemcpy(r.bufs.src, h.data(), bytes, hipMemcpyHostToDevice));
        HIPWARN(hipMemset(r.bufs.dst, 0, bytes));
        HIPWARN(hipDeviceSynchronize());
    } else
#endif
    {
        memcpy(r.bufs.src, h.data(), bytes);
        memset(r.bufs.dst, 0, bytes);
    }
    if (ct == UCC_COLL_TYPE_BCAST) {
#ifdef UCC_AMD_HAS_HIP
        if (o.mem == "cuda") {
            HIPWARN(hipMemcpy(r.bufs.src, h.data(), bytes,
                              hipMemcpyHostToDevice));
        }
#endif
    }
    ucc_collective_post(req);
    while (ucc_collective_test(req) == UCC_INPROGRESS) {
        ucc_context_progress(r.ctx);
    }
#ifdef UCC_AMD_HAS_HIP
    if (o.mem == "cuda") {
        HIPWARN(hipDeviceSynchronize());
        HIPWARN(hipMemcpy(out.data(),
                          ct == UCC_COLL_TYPE_BCAST ? r.bufs.src
                                                    : r.bufs.dst,
                          bytes, hipMemcpyDeviceToHost));
    } else
#endif
    {
        memcpy(out.data(),
               ct == UCC_COLL_TYPE_BCAST ? r.bufs.src : r.bufs.dst,
               bytes);
    }
    size_t bad = vcheck(ct, rank, nranks, count, out.data());
    if (bad) {
        fprintf(stderr, "[rank %d] VALIDATION FAILED: %zu/%zu bad at %zu "
                        "bytes\n", rank, bad, count, bytes);
        return false;
    }
    return true;
}

/* ---------------------------------------------------------- fork driver */
static int run_forked_child(const Opts &o, int rank)
{
#ifdef UCC_AMD_HAS_HIP
    if (o.mem == "cuda") {
        int ndev = 0;
        HIPWARN(hipGetDeviceCount(&ndev));
        if (ndev > 0) {
            HIPWARN(hipSetDevice(rank % ndev));
        }
    }
#endif
    g_shm_oob.set_rank(rank);
    Rank r;
    setup_rank(r, rank, o.nprocs, true);
    while (true) {
        ucc_status_t st = ucc_team_create_test(r.team);
        if (st == UCC_OK) {
            break;
        }
        if (st != UCC_INPROGRESS) {
            fprintf(stderr, "team create failed: %d\n", st);
            return 1;
        }
    }
    ucc_coll_type_t ct = coll_from_name(o.coll);
    r.bufs.alloc(std::max(o.max_b * 2, (size_t)4096), o.mem == "cuda");
    /* align ranks before the first collective: team create time is
     * rank-skewed (e.g. a several-second vendor-lib fallback on one
     * rank) and triggered posts have no host-side entry gate */
    g_shm_oob.max_double(0.0);
    if (rank == 0) {
        printf("# ucc_perftest  coll=%s mem=%s dt=%s op=%s procs=%d%s%s\n",
               o.coll.c_str(), o.mem.c_str(), o.dtype.c_str(),
               o.op.c_str(), o.nprocs, o.persistent ? " persistent" : "",
               o.inplace ? " inplace" : "");
        printf("%12s %8s %12s %12s %12s\n", "bytes", "iters", "avg_us",
               "algbw_GBps", "busbw_GBps");
    }
    for (size_t bytes = o.min_b; bytes <= o.max_b; bytes *= 2) {
        int iters = bytes >= (64 << 10) ? o.large_iters : o.iters;
        std::vector<uint64_t> cnts, dsps, rcnts, rdsps;
        ucc_coll_args_t a = make_args(o, r, bytes, rank, o.nprocs, cnts,
                                      dsps, rcnts, rdsps);
        ucc_coll_req_h  req;
        if (ucc_collective_init(&a, &req, r.team) != UCC_OK) {
            if (rank == 0) {
                printf("%12zu        - coll_init unsupported\n", bytes);
            }
            continue;
        }
#ifdef UCC_AMD_HAS_HIP
        hipStream_t tstream = nullptr;
        ucc_ee_h    ee      = nullptr;
        if (o.triggered) {
            HIPWARN(hipStreamCreateWithFlags(&tstream, hipStreamNonBlocking));
            ucc_ee_params_t ep{};
            ep.ee_type         = UCC_EE_ROCM_STREAM;
            ep.ee_context      = (void *)tstream;
            ep.ee_context_size = sizeof(void *);
            if (ucc_ee_create(r.team, &ep, &ee) != UCC_OK) {
                fprintf(stderr, "ee create failed\n");
                ucc_collective_finalize(req);
                continue;
            }
        }
#endif
        bool trig_fail = false;
        auto iter = [&]() {
#ifdef UCC_AMD_HAS_HIP
            if (o.triggered) {
                ucc_ev_t ev{};
                ev.ev_type = UCC_EVENT_COMPUTE_COMPLETE;
                ev.req     = req;
                if (ucc_collective_triggered_post(ee, &ev) != UCC_OK) {
                    trig_fail = true;
                    return;
                }
                HIPWARN(hipStreamSynchronize(tstream));
                return;
            }
#endif
            if (ucc_collective_post(req) != UCC_OK) {
                trig_fail = true; /* reuse the unsupported-row path */
                return;
            }
            ucc_status_t ts;
            while ((ts = ucc_collective_test(req)) == UCC_INPROGRESS) {
                ucc_context_progress(r.ctx);
            }
            if (ts != UCC_OK) {
                trig_fail = true;
            }
        };
        if (o.check &&
            !validate_once(o, r, req, ct, bytes, rank, o.nprocs)) {
            ucc_collective_finalize(req);
            g_shm_oob.max_double(0.0);
            g_shm_oob.max_double(0.0);
            exit(2);
        }
        for (int i = 0; i < o.warmup && !trig_fail; i++) {
            iter();
        }
        if (trig_fail) {
            if (rank == 0) {
                printf("%12zu        - unsupported at this size\n",
                       bytes);
            }
            ucc_collective_finalize(req);
#ifdef UCC_AMD_HAS_HIP
            if (ee) { ucc_ee_destroy(ee); ee = nullptr; }
            if (tstream) { HIPWARN(hipStreamDestroy(tstream)); tstream = nullptr; }
#endif
            g_shm_oob.max_double(0.0);
            g_shm_oob.max_double(0.0);
            continue;
        }
#ifdef UCC_AMD_HAS_HIP
        if (o.mem == "cuda") {
            HIPWARN(hipDeviceSynchronize());
        }
#endif
        g_shm_oob.max_double(0.0); /* barrier */
        double t0 = now_s();
        for (int i = 0; i < iters; i++) {
            iter();
        }
#ifdef UCC_AMD_HAS_HIP
        if (o.mem == "cuda") {
            HIPWARN(hipDeviceSynchronize());
        }
#endif
        double t  = g_shm_oob.max_double(now_s() - t0) / iters;
        if (trig_fail) { /* a timed iteration errored: no number */
            if (rank == 0) {
                printf("%12zu        - failed during timing\n", bytes);
            }
            ucc_collective_finalize(req);
            continue;
        }
        ucc_collective_finalize(req);
#ifdef UCC_AMD_HAS_HIP
        if (ee) {
            ucc_ee_destroy(ee);
        }
        if (tstream) {
            HIPWARN(hipStreamDestroy(tstream));
        }
#endif
        if (rank == 0) {
            double algbw = bytes / t / 1e9;
            double busbw = algbw * busbw_factor(ct, o.nprocs);
            printf("%12zu %8d %12.2f %12.2f %12.2f\n", bytes, iters,
                   t * 1e6, algbw, busbw);
        }
    }
    return 0;
}

static int run_forked(const Opts &o)
{
    char name[64];
    snprintf(name, sizeof(name), "/uccperf-oob-%d", (int)getpid());
    if (!g_shm_oob.create(name, o.nprocs)) {
        fprintf(stderr, "shm oob create failed\n");
        return 1;
    }
    std::vector<pid_t> kids;
    for (int rnk = 1; rnk < o.nprocs; rnk++) {
        pid_t pid = fork();
        if (pid == 0) {
            if (!g_shm_oob.open_(name, o.nprocs, rnk)) {
                _exit(1);
            }
            _exit(run_forked_child(o, rnk));
        }
        kids.push_back(pid);
    }
    g_shm_oob.set_rank(0);
    int rc = run_forked_child(o, 0);
    for (pid_t pid : kids) {
        int st = 0;
        waitpid(pid, &st, 0);
        if (!WIFEXITED(st) || WEXITSTATUS(st) != 0) {
            rc = 1;
        }
    }
    g_shm_oob.unlink_();
    return rc;
}

/* ----------------------------------------------------- in-process driver */
static int run_inproc(const Opts &o)
{
    int n = o.nranks;
    g_loob.n = n;
    g_loob.next_round.assign(n, 0);
    std::vector<Rank> ranks(n);
    for (int i = 0; i < n; i++) {
        setup_rank(ranks[i], i, n, false);
    }
    while (true) {
        bool all = true;
        for (auto &r : ranks) {
            ucc_status_t st = ucc_team_create_test(r.team);
            if (st == UCC_INPROGRESS) {
                all = false;
            } else if (st != UCC_OK) {
                fprintf(stderr, "team create failed: %d\n", st);
                return 1;
            }
        }
        if (all) {
            break;
        }
    }
    ucc_coll_type_t ct = coll_from_name(o.coll);
    for (auto &r : ranks) {
        r.bufs.alloc(std::max(o.max_b * 2, (size_t)4096), o.mem == "cuda");
    }
    printf("# ucc_perftest  coll=%s mem=%s dt=%s op=%s inproc_ranks=%d%s%s\n",
           o.coll.c_str(), o.mem.c_str(), o.dtype.c_str(), o.op.c_str(), n,
           o.persistent ? " persistent" : "", o.inplace ? " inplace" : "");
    printf("%12s %8s %12s %12s %12s\n", "bytes", "iters", "avg_us",
           "algbw_GBps", "busbw_GBps");
    for (size_t bytes = o.min_b; bytes <= o.max_b; bytes *= 2) {
        int iters = bytes >= (64 << 10) ? o.large_iters : o.iters;
        std::vector<std::vector<uint64_t>> cnts(n), dsps(n), rcnts(n),
            rdsps(n);
        std::vector<ucc_coll_req_h>        reqs(n);
        bool                               ok = true;
        for (int i = 0; i < n; i++) {
            ucc_coll_args_t a = make_args(o, ranks[i], bytes, i, n,
                                          cnts[i], dsps[i], rcnts[i],
                                          rdsps[i]);
            if (ucc_collective_init(&a, &reqs[i], ranks[i].team) !=
                UCC_OK) {
                ok = false;
                break;
            }
        }
        if (!ok) {
            printf("%12zu        - coll_init unsupported\n", bytes);
            continue;
        }
        bool coll_fail = false;
        int  rs_iter   = 0;
        const bool rooted = ct == UCC_COLL_TYPE_BCAST ||
                            ct == UCC_COLL_TYPE_GATHER ||
                            ct == UCC_COLL_TYPE_SCATTER ||
                            ct == UCC_COLL_TYPE_REDUCE;
        auto iter = [&]() {
            if (o.root_shift && rooted) {
                /* -R: rotate the root every iteration (reference
                 * root-shift mode) — re-init so placement effects
                 * average out; the timing includes the re-init */
                int root = rs_iter++ % n;
                for (int i = 0; i < n; i++) {
                    ucc_collective_finalize(reqs[i]);
                    ucc_coll_args_t a2 = make_args(
                        o, ranks[i], bytes, i, n, cnts[i], dsps[i],
                        rcnts[i], rdsps[i], root);
                    if (ucc_collective_init(&a2, &reqs[i],
                                            ranks[i].team) != UCC_OK) {
                        coll_fail = true;
                        return;
                    }
                }
            }
            for (int i = 0; i < n; i++) {
                if (ucc_collective_post(reqs[i]) != UCC_OK) {
                    coll_fail = true;
                }
            }
            bool done = false;
            while (!done) {
                done = true;
                for (int i = 0; i < n; i++) {
                    ucc_status_t ts = ucc_collective_test(reqs[i]);
                    if (ts == UCC_INPROGRESS) {
                        done = false;
                        ucc_context_progress(ranks[i].ctx);
                    } else if (ts != UCC_OK) {
                        coll_fail = true;
                    }
                }
            }
        };
        const bool checkable =
            o.check && o.dtype == "float32" && !o.inplace &&
            o.mem == "host" && bytes >= 4 &&
            (ct == UCC_COLL_TYPE_ALLREDUCE ||
             ct == UCC_COLL_TYPE_ALLGATHER ||
             ct == UCC_COLL_TYPE_REDUCE_SCATTER ||
             ct == UCC_COLL_TYPE_BCAST);
        if (checkable) {
            std::vector<float> h;
            for (int i = 0; i < n; i++) {
                vpattern(ct, i, bytes / 4, h);
                memcpy(ranks[i].bufs.src, h.data(), bytes);
                memset(ranks[i].bufs.dst, 0, bytes);
            }
            iter();
            size_t bad = 0;
            for (int i = 0; i < n && !coll_fail; i++) {
                const float *out = (const float *)(
                    ct == UCC_COLL_TYPE_BCAST ? ranks[i].bufs.src
                                              : ranks[i].bufs.dst);
                /* note: make_args rounded the count for block colls */
                size_t cnt2 =
                    ct == UCC_COLL_TYPE_REDUCE_SCATTER
                        ? ((bytes / 4) - (bytes / 4) % n) / n
                        : ct == UCC_COLL_TYPE_ALLGATHER
                              ? (bytes / 4) - (bytes / 4) % n
                              : bytes / 4;
                bad += vcheck(ct, i, n, ct == UCC_COLL_TYPE_REDUCE_SCATTER
                                            ? cnt2 * n
                                            : cnt2,
                              out);
            }
            if (bad) {
                printf("%12zu        - VALIDATION FAILED (%zu bad)\n",
                       bytes, bad);
                for (int i = 0; i < n; i++) {
                    ucc_collective_finalize(reqs[i]);
                }
                exit(2);
            }
        }
        for (int i = 0; i < o.warmup && !coll_fail; i++) {
            iter();
        }
        if (coll_fail) {
            printf("%12zu        - unsupported at this size\n", bytes);
            for (int i = 0; i < n; i++) {
                ucc_collective_finalize(reqs[i]);
            }
            continue;
        }
#ifdef UCC_AMD_HAS_HIP
        if (o.mem == "cuda") {
            HIPWARN(hipDeviceSynchronize());
        }
#endif
        double t0 = now_s();
        for (int i = 0; i < iters; i++) {
            iter();
        }
#ifdef UCC_AMD_HAS_HIP
        if (o.mem == "cuda") {
            HIPWARN(hipDeviceSynchronize());
        }
#endif
        double t = (now_s() - t0) / iters;
        for (int i = 0; i < n; i++) {
            ucc_collective_finalize(reqs[i]);
        }
        if (coll_fail) {
            printf("%12zu        - failed during timing\n", bytes);
            continue;
        }
        double algbw = bytes / t / 1e9;
        double busbw = algbw * busbw_factor(ct, n);
        printf("%12zu %8d %12.2f %12.2f %12.2f\n", bytes, iters, t * 1e6,
               algbw, busbw);
    }
    for (auto &r : ranks) { /* full teardown (leak-checked under ASAN) */
        ucc_team_destroy(r.team);
    }
    for (auto &r : ranks) {
        ucc_context_destroy(r.ctx);
        ucc_finalize(r.lib);
    }
    return 0;
}

int main(int argc, char **argv)
{
    setvbuf(stdout, nullptr, _IOLBF, 0); /* line-buffer under pipes */
    Opts o;
    int  c;
    while ((c = getopt(argc, argv, "c:b:e:n:w:m:d:o:p:j:FiCTMRh")) != -1) {
        switch (c) {
        case 'c': o.coll = optarg; break;
        case 'b': o.min_b = strtoull(optarg, nullptr, 0); break;
        case 'e': o.max_b = strtoull(optarg, nullptr, 0); break;
        case 'n': o.iters = atoi(optarg); break;
        case 'w': o.warmup = atoi(optarg); break;
        case 'm': o.mem = optarg; break;
        case 'd': o.dtype = optarg; break;
        case 'o': o.op = optarg; break;
        case 'p': o.nprocs = atoi(optarg); break;
        case 'j': o.nranks = atoi(optarg); break;
        case 'F': o.persistent = true; break;
        case 'i': o.inplace = true; break;
        case 'C': o.check = true; break;
        case 'T': o.triggered = true; o.persistent = true; break;
        case 'M': o.skew = true; break;
        case 'R': o.root_shift = true; break;
        case 'h':
        default:
            printf("ucc_perftest [-c coll] [-b min] [-e max] [-n iters] "
                   "[-w warmup] [-m host|cuda] [-d dtype] [-o op] "
                   "[-p nprocs(fork)] [-j inproc_ranks] [-F persistent] "
                   "[-i inplace] [-T triggered] [-M skewed-alltoallv] "
                   "[-R root-shift]\n");
            return c == 'h' ? 0 : 1;
        }
    }
    if (o.min_b < 1) {
        o.min_b = 1;
    }
    return o.nprocs > 1 ? run_forked(o) : run_inproc(o);
}
