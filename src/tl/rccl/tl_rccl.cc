/* TL "rccl": vendor-collective comparison/fallback transport over RCCL.
 *
 * Reference parity: tl/nccl (tl_nccl_coll.c call-site inventory, SURVEY
 * §2.8) / tl/rccl (tl_rccl_coll.c), re-derived: team create exchanges the
 * ncclUniqueId through the team's combined OOB exchange round (rank 0
 * contributes it) and calls blocking ncclCommInitRank (the reference's
 * fallback-friendly blocking init, tl_nccl_team.c:147-166); each
 * collective is one RCCL call (or a ncclGroupStart/End send/recv loop for
 * alltoall(v)/gather/scatter) on a dedicated nonblocking HIP stream,
 * completed via hipEvent query from the progress engine.
 *
 * Scores below tl/cdna4 (20 vs 80): the native xGMI transport is the
 * primary path; RCCL is the A/B baseline (UCC_TL_CDNA4_ENABLE=0 flips a
 * team to RCCL for comparison) and the fallback for shapes cdna4 declines.
 */
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include "../../core/core.h"
#include "../../mc/mc.h"

namespace ucc {
namespace {

/* fire-and-forget HIP calls (cleanup paths, stream ops whose failures
 * surface later through event queries): log-and-continue. */
static inline void hip_warn_on_err(hipError_t e, const char *what)
{
    if (e != hipSuccess) {
        ucc_warn("%s: %s", what, hipGetErrorString(e));
    }
}
#define HIPWARN(expr) hip_warn_on_err((expr), #expr)

#define RCCLCHK(expr)                                                        \
    do {                                                                     \
        ncclResult_t _r = (expr);                                            \
        if (_r != ncclSuccess) {                                             \
            ucc_error("%s failed: %s", #expr, ncclGetErrorString(_r));       \
            return UCC_ERR_NO_RESOURCE;                                      \
        }                                                                    \
    } while (0)

class RcclTl;

class RcclTlContext final : public TlContext {
  public:
    RcclTlContext(Context *ctx, Tl *tl) : TlContext(ctx), tl_(tl)
    {
        HIPWARN(hipStreamCreateWithFlags(&stream_, hipStreamNonBlocking));
    }
    ~RcclTlContext() override
    {
        if (stream_) {
            HIPWARN(hipStreamDestroy(stream_));
        }
    }
    Tl *iface() override;

    Tl         *tl_;
    hipStream_t stream_ = nullptr;
};

class RcclTlTeam final : public TlTeam {
  public:
    RcclTlTeam(TlContext *tlc, Team *team) : TlTeam(tlc, team) {}
    ~RcclTlTeam() override
    {
        if (comm_) {
            ncclCommDestroy(comm_);
        }
    }

    size_t exchg_size() override { return sizeof(ncclUniqueId); }

    void exchg_pack(void *buf) override
    {
        ncclUniqueId id{};
        if (team_->rank == 0) {
            if (ncclGetUniqueId(&id) != ncclSuccess) {
                ucc_error("ncclGetUniqueId failed");
            }
        }
        memcpy(buf, &id, sizeof(id));
    }

    ucc_status_t exchg_unpack(const void *all, size_t stride) override
    {
        memcpy(&uid_, (const uint8_t *)all + 0 * stride, sizeof(uid_));
        return UCC_OK;
    }

    ucc_status_t create_test() override
    {
        if (!comm_) {
            /* blocking collective init (all ranks reach here after the
             * exchange round) */
            RCCLCHK(ncclCommInitRank(&comm_, (int)team_->size, uid_,
                                     (int)team_->rank));
        }
        return UCC_OK;
    }

    void get_scores(Team *team, ScoreMap &map) override;

    ncclComm_t   comm_ = nullptr;
    ncclUniqueId uid_{};
};

static bool nccl_dt(ucc_datatype_t dt, ncclDataType_t *out)
{
    switch (dt) {
    case UCC_DT_INT8: *out = ncclInt8; return true;
    case UCC_DT_UINT8: *out = ncclUint8; return true;
    case UCC_DT_INT32: *out = ncclInt32; return true;
    case UCC_DT_UINT32: *out = ncclUint32; return true;
    case UCC_DT_INT64: *out = ncclInt64; return true;
    case UCC_DT_UINT64: *out = ncclUint64; return true;
    case UCC_DT_FLOAT16: *out = ncclFloat16; return true;
    case UCC_DT_BFLOAT16: *out = ncclBfloat16; return true;
    case UCC_DT_FLOAT32: *out = ncclFloat32; return true;
    case UCC_DT_FLOAT64: *out = ncclFloat64; return true;
    default: return false;
    }
}

static bool nccl_op(ucc_reduction_op_t op, ncclRedOp_t *out)
{
    switch (op) {
    case UCC_OP_SUM: *out = ncclSum; return true;
    case UCC_OP_PROD: *out = ncclProd; return true;
    case UCC_OP_MAX: *out = ncclMax; return true;
    case UCC_OP_MIN: *out = ncclMin; return true;
    case UCC_OP_AVG: *out = ncclAvg; return true;
    default: return false;
    }
}

static inline size_t vcnt(const ucc_coll_args_t &a, const void *c,
                          uint32_t i)
{
    return (a.flags & UCC_COLL_ARGS_FLAG_COUNT_64BIT)
               ? (size_t)((const uint64_t *)c)[i]
               : (size_t)((const uint32_t *)c)[i];
}
static inline size_t vdsp(const ucc_coll_args_t &a, const void *d,
                          uint32_t i)
{
    return (a.flags & UCC_COLL_ARGS_FLAG_DISPLACEMENTS_64BIT)
               ? (size_t)((const uint64_t *)d)[i]
               : (size_t)((const uint32_t *)d)[i];
}

class RcclTask final : public Task {
  public:
    RcclTask(Context *ctx, RcclTlTeam *tt, const ucc_coll_args_t &args)
        : Task(ctx), tt_(tt), a_(args)
    {
        HIPWARN(hipEventCreateWithFlags(&ev_, hipEventDisableTiming));
    }
    ~RcclTask() override
    {
        if (ev_) {
            HIPWARN(hipEventDestroy(ev_));
        }
        if (barrier_buf_) {
            HIPWARN(hipFree(barrier_buf_));
        }
    }

    ucc_status_t post() override
    {
        ucc_status_t st = issue();
        if (st != UCC_OK) {
            status = st;
            return st;
        }
        HIPWARN(hipEventRecord(ev_, stream()));
        status = UCC_INPROGRESS;
        return UCC_OK;
    }

    ucc_status_t progress() override
    {
        if (status != UCC_INPROGRESS) {
            return status;
        }
        hipError_t e = hipEventQuery(ev_);
        if (e == hipErrorNotReady) {
            /* failure detection: a comm that hit a transport error never
             * completes the event — surface the async error instead of
             * spinning forever (reference tl_nccl optional
             * ncclCommGetAsyncError polling) */
            ncclResult_t ar = ncclSuccess;
            if (ncclCommGetAsyncError(tt_->comm_, &ar) == ncclSuccess &&
                ar != ncclSuccess && ar != ncclInProgress) {
                ucc_error("rccl async error: %s",
                          ncclGetErrorString(ar));
                status = UCC_ERR_NO_RESOURCE;
                return status;
            }
            return UCC_INPROGRESS;
        }
        status = e == hipSuccess ? UCC_OK : UCC_ERR_NO_RESOURCE;
        return status;
    }

  private:
    hipStream_t stream() { return ((RcclTlContext *)tt_->tlc_)->stream_; }

    ucc_status_t issue()
    {
        ncclComm_t     comm = tt_->comm_;
        hipStream_t    s    = stream();
        const uint32_t n    = tt_->team_->size;
        const uint32_t me   = tt_->team_->rank;
        const bool inplace  = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        ncclDataType_t dt;
        ncclRedOp_t    op;
        switch (a_.coll_type) {
        case UCC_COLL_TYPE_ALLREDUCE: {
            if (!nccl_dt(a_.dst.info.datatype, &dt) ||
                !nccl_op(a_.op, &op)) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            const void *src =
                inplace ? a_.dst.info.buffer : a_.src.info.buffer;
            RCCLCHK(ncclAllReduce(src, a_.dst.info.buffer,
                                  a_.dst.info.count, dt, op, comm, s));
            return UCC_OK;
        }
        case UCC_COLL_TYPE_ALLGATHER: {
            if (!nccl_dt(a_.dst.info.datatype, &dt)) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            size_t      per = a_.dst.info.count / n;
            const void *src =
                inplace ? (const uint8_t *)a_.dst.info.buffer +
                              me * per * ucc_dt_size(a_.dst.info.datatype)
                        : a_.src.info.buffer;
            RCCLCHK(ncclAllGather(src, a_.dst.info.buffer, per, dt, comm,
                                  s));
            return UCC_OK;
        }
        case UCC_COLL_TYPE_REDUCE_SCATTER: {
            if (!nccl_dt(a_.dst.info.datatype, &dt) ||
                !nccl_op(a_.op, &op)) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            if (inplace) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            RCCLCHK(ncclReduceScatter(a_.src.info.buffer,
                                      a_.dst.info.buffer,
                                      a_.dst.info.count, dt, op, comm, s));
            return UCC_OK;
        }
        case UCC_COLL_TYPE_BCAST: {
            if (!nccl_dt(a_.src.info.datatype, &dt)) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            RCCLCHK(ncclBroadcast(a_.src.info.buffer, a_.src.info.buffer,
                                  a_.src.info.count, dt, (int)a_.root,
                                  comm, s));
            return UCC_OK;
        }
        case UCC_COLL_TYPE_REDUCE: {
            if (!nccl_dt(a_.src.info.datatype, &dt) ||
                !nccl_op(a_.op, &op)) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            const void *src = (inplace && me == a_.root)
                                  ? a_.dst.info.buffer
                                  : a_.src.info.buffer;
            RCCLCHK(ncclReduce(src, a_.dst.info.buffer, a_.src.info.count,
                               dt, op, (int)a_.root, comm, s));
            return UCC_OK;
        }
        case UCC_COLL_TYPE_BARRIER: {
            /* 4-byte allreduce (reference tl_nccl barrier) */
            if (!barrier_buf_ &&
                hipMalloc(&barrier_buf_, 4) != hipSuccess) {
                return UCC_ERR_NO_MEMORY;
            }
            RCCLCHK(ncclAllReduce(barrier_buf_, barrier_buf_, 1,
                                  ncclFloat32, ncclSum, comm, s));
            return UCC_OK;
        }
        case UCC_COLL_TYPE_ALLTOALL: {
            size_t ds  = ucc_dt_size(a_.dst.info.datatype);
            size_t per = a_.dst.info.count / n * ds;
            auto  *sb  = (const uint8_t *)a_.src.info.buffer;
            auto  *db  = (uint8_t *)a_.dst.info.buffer;
            RCCLCHK(ncclGroupStart());
            for (uint32_t r = 0; r < n; r++) {
                RCCLCHK(ncclSend(sb + r * per, per, ncclInt8, (int)r, comm,
                                 s));
                RCCLCHK(ncclRecv(db + r * per, per, ncclInt8, (int)r, comm,
                                 s));
            }
            RCCLCHK(ncclGroupEnd());
            return UCC_OK;
        }
        case UCC_COLL_TYPE_ALLTOALLV: {
            size_t sds = ucc_dt_size(a_.src.info_v.datatype);
            size_t dds = ucc_dt_size(a_.dst.info_v.datatype);
            auto  *sb  = (const uint8_t *)a_.src.info_v.buffer;
            auto  *db  = (uint8_t *)a_.dst.info_v.buffer;
            RCCLCHK(ncclGroupStart());
            for (uint32_t r = 0; r < n; r++) {
                RCCLCHK(ncclSend(
                    sb + vdsp(a_, a_.src.info_v.displacements, r) * sds,
                    vcnt(a_, a_.src.info_v.counts, r) * sds, ncclInt8,
                    (int)r, comm, s));
                RCCLCHK(ncclRecv(
                    db + vdsp(a_, a_.dst.info_v.displacements, r) * dds,
                    vcnt(a_, a_.dst.info_v.counts, r) * dds, ncclInt8,
                    (int)r, comm, s));
            }
            RCCLCHK(ncclGroupEnd());
            return UCC_OK;
        }
        case UCC_COLL_TYPE_ALLGATHERV: {
            /* no NCCL primitive: grouped p2p ring of own-block sends
             * (reference tl_nccl allgatherv role) */
            size_t ds  = ucc_dt_size(a_.dst.info_v.datatype);
            auto  *db  = (uint8_t *)a_.dst.info_v.buffer;
            size_t mine = vcnt(a_, a_.dst.info_v.counts, me) * ds;
            const void *src =
                inplace ? db + vdsp(a_, a_.dst.info_v.displacements, me) *
                                   ds
                        : a_.src.info.buffer;
            RCCLCHK(ncclGroupStart());
            for (uint32_t r = 0; r < n; r++) {
                RCCLCHK(ncclSend(src, mine, ncclInt8, (int)r, comm, s));
                RCCLCHK(ncclRecv(
                    db + vdsp(a_, a_.dst.info_v.displacements, r) * ds,
                    vcnt(a_, a_.dst.info_v.counts, r) * ds, ncclInt8,
                    (int)r, comm, s));
            }
            RCCLCHK(ncclGroupEnd());
            return UCC_OK;
        }
        case UCC_COLL_TYPE_GATHER:
        case UCC_COLL_TYPE_GATHERV: {
            const bool is_v = a_.coll_type == UCC_COLL_TYPE_GATHERV;
            size_t     ds   = ucc_dt_size(
                me == a_.root && !is_v ? a_.dst.info.datatype
                                           : a_.src.info.datatype);
            RCCLCHK(ncclGroupStart());
            if (me == a_.root) {
                auto *db = (uint8_t *)(is_v ? a_.dst.info_v.buffer
                                            : a_.dst.info.buffer);
                size_t per =
                    is_v ? 0 : a_.dst.info.count / n * ds;
                for (uint32_t r = 0; r < n; r++) {
                    uint8_t *dst =
                        is_v ? db + vdsp(a_, a_.dst.info_v.displacements,
                                         r) *
                                        ds
                             : db + (size_t)r * per;
                    size_t len =
                        is_v ? vcnt(a_, a_.dst.info_v.counts, r) * ds
                             : per;
                    RCCLCHK(ncclRecv(dst, len, ncclInt8, (int)r, comm,
                                     s));
                }
            }
            {
                const void *src =
                    (inplace && me == a_.root)
                        ? nullptr /* in-place root: block already home */
                        : a_.src.info.buffer;
                size_t len = a_.src.info.count * ds;
                if (inplace && me == a_.root) {
                    /* self send still required to satisfy the posted
                     * recv: source it from the dst block */
                    auto *db = (uint8_t *)(is_v ? a_.dst.info_v.buffer
                                                : a_.dst.info.buffer);
                    src = is_v ? db + vdsp(a_,
                                           a_.dst.info_v.displacements,
                                           me) *
                                          ds
                               : db + (size_t)me *
                                          (a_.dst.info.count / n) * ds;
                    len = is_v ? vcnt(a_, a_.dst.info_v.counts, me) * ds
                               : a_.dst.info.count / n * ds;
                }
                RCCLCHK(ncclSend(src, len, ncclInt8, (int)a_.root, comm,
                                 s));
            }
            RCCLCHK(ncclGroupEnd());
            return UCC_OK;
        }
        case UCC_COLL_TYPE_SCATTER:
        case UCC_COLL_TYPE_SCATTERV: {
            const bool is_v = a_.coll_type == UCC_COLL_TYPE_SCATTERV;
            size_t     ds   = ucc_dt_size(
                me == a_.root && !is_v ? a_.src.info.datatype
                                           : a_.dst.info.datatype);
            RCCLCHK(ncclGroupStart());
            if (me == a_.root) {
                auto *sb = (const uint8_t *)(is_v ? a_.src.info_v.buffer
                                                  : a_.src.info.buffer);
                size_t per = is_v ? 0 : a_.src.info.count / n * ds;
                for (uint32_t r = 0; r < n; r++) {
                    if (inplace && r == me) {
                        continue; /* root's block stays in place */
                    }
                    const uint8_t *src =
                        is_v ? sb + vdsp(a_, a_.src.info_v.displacements,
                                         r) *
                                        ds
                             : sb + (size_t)r * per;
                    size_t len =
                        is_v ? vcnt(a_, a_.src.info_v.counts, r) * ds
                             : per;
                    RCCLCHK(ncclSend(src, len, ncclInt8, (int)r, comm,
                                     s));
                }
            }
            if (!(inplace && me == a_.root)) {
                void  *dst = a_.dst.info.buffer;
                size_t len = a_.dst.info.count * ds;
                RCCLCHK(ncclRecv(dst, len, ncclInt8, (int)a_.root, comm,
                                 s));
            }
            RCCLCHK(ncclGroupEnd());
            return UCC_OK;
        }
        default:
            return UCC_ERR_NOT_SUPPORTED;
        }
    }

    RcclTlTeam     *tt_;
    ucc_coll_args_t a_;
    hipEvent_t      ev_ = nullptr;
    void           *barrier_buf_ = nullptr;
};

class RcclTl final : public Tl {
  public:
    const char *name() const override { return "rccl"; }
    int         default_score() const override { return 20; }

    TlContext *context_create(Context *ctx) override
    {
        auto &cfg = Config::instance();
        cfg.declare("TL_RCCL", "ENABLE", "1",
                    "enable the RCCL comparison/fallback transport");
        cfg.declare("TL_RCCL", "SCORE", "20",
                    "score for rccl ranges (cdna4 default is 80)");
        if (!cfg.get_bool("TL_RCCL", "ENABLE", true) ||
            !mc::hip_available()) {
            return nullptr;
        }
        return new RcclTlContext(ctx, this);
    }

    TlTeam *team_create(TlContext *tlc, Team *team) override
    {
        if (team->size < 2 || !team->all_have_device()) {
            return nullptr;
        }
        /* RCCL cannot run N blocking single-device comm inits from one
         * thread: decline in-process multi-rank jigs. */
        int same = 0;
        for (auto &p : team->procs) {
            same += (p.pid == team->ctx->proc.pid);
        }
        if (same > 1) {
            return nullptr;
        }
        return new RcclTlTeam(tlc, team);
    }
};

static RcclTl g_rccl_tl;

Tl *RcclTlContext::iface() { return &g_rccl_tl; }

void RcclTlTeam::get_scores(Team *team, ScoreMap &map)
{
    (void)team;
    RcclTlTeam *self = this;
    int         sc   = (int)Config::instance().get_int("TL_RCCL", "SCORE",
                                                       g_rccl_tl.default_score());
    auto add = [&](ucc_coll_type_t ct) {
        ScoreRange r;
        r.start    = 0;
        r.end      = SIZE_MAX;
        r.score    = sc;
        r.tl_name  = "rccl";
        r.alg_name = "rccl";
        r.init     = [self](const ucc_coll_args_t &args, Team *t,
                        Task **task) -> ucc_status_t {
            *task = new RcclTask(t->ctx, self, args);
            return UCC_OK;
        };
        for (auto mt : {UCC_MEMORY_TYPE_CUDA, UCC_MEMORY_TYPE_CUDA_MANAGED}) {
            map.add(ct, mt, r);
        }
    };
    add(UCC_COLL_TYPE_ALLREDUCE);
    add(UCC_COLL_TYPE_ALLGATHER);
    add(UCC_COLL_TYPE_REDUCE_SCATTER);
    add(UCC_COLL_TYPE_BCAST);
    add(UCC_COLL_TYPE_REDUCE);
    add(UCC_COLL_TYPE_BARRIER);
    add(UCC_COLL_TYPE_ALLTOALL);
    add(UCC_COLL_TYPE_ALLTOALLV);
    add(UCC_COLL_TYPE_ALLGATHERV);
    add(UCC_COLL_TYPE_GATHER);
    add(UCC_COLL_TYPE_GATHERV);
    add(UCC_COLL_TYPE_SCATTER);
    add(UCC_COLL_TYPE_SCATTERV);
}

} // namespace

Tl *tl_rccl_iface() { return &g_rccl_tl; }

} // namespace ucc
