/* TL "shm": intra-node host-memory collectives through a POSIX shared
 * memory segment with a slotted, sense-free monotonic step protocol.
 *
 * Reference parity: this TL replaces the reference's UCX shm path for host
 * buffers (tl/ucp over UCX shared memory) and borrows the slot/sync design
 * of tl/cuda (tl_cuda_coll.h: max_concurrent slots, per-rank seq
 * publication, shm barrier) re-derived for host data movement:
 *
 *  - segment = header | per-slot per-rank step counters | per-slot data
 *    areas (2x chunk per rank, double buffered) | per-slot result areas.
 *  - a collective is the u-th use of slot s = (tl coll seq) % nslots; all
 *    step targets are base = u*CAP plus a deterministic program counter,
 *    published with release stores and awaited with acquire loads, so no
 *    sense reversal and no resets are needed.
 *  - every collective runs rounds of <= chunk bytes; data staged through
 *    the writer's area, reduced/copied out by readers, with consumption
 *    publications gating buffer reuse (parity j%2).
 *
 * All 16 collective types run on this one machinery (alltoall(v) and the
 * rooted v-variants stage per-peer cells of chunk/nranks bytes and agree on
 * the round count via a metadata round through the result area).
 */
#include "../../core/core.h"
#include "../../ec/ec_cpu.h"
#include "../../mc/mc.h"

#include "slot_seg.h"

#include <atomic>
#include <unistd.h>

namespace ucc {
namespace {

class ShmTlTeam; /* fwd */

/* ------------------------------------------------------------ coll task */
class ShmCollTask final : public Task {
  public:
    ShmCollTask(Context *ctx, ShmTlTeam *tt, const ucc_coll_args_t &args);
    ucc_status_t post() override;
    ucc_status_t progress() override;

  private:
    /* helpers */
    inline void publish(uint64_t k);
    inline bool all_ge(uint64_t k);
    inline bool rank_ge(uint32_t r, uint64_t k);
    inline bool root_ge(uint64_t k) { return rank_ge((uint32_t)a_.root, k); }

    void calc_socket(const Team *team, uint32_t root)
    {
        if (!Config::instance().get_bool("TL_SHM", "SOCKET_STAGING",
                                         true)) {
            return;
        }
        const auto &procs = team->procs;
        /* group by CPU socket; if socket ids are unavailable on this
         * platform, NUMA ids carry the same locality signal (both are
         * reference ucc_sbgp kinds) */
        auto gid = [&](uint32_t r) -> int16_t {
            return procs[r].socket_id >= 0 ? procs[r].socket_id
                                           : procs[r].numa_id;
        };
        int16_t rsck  = gid(root);
        bool    known = rsck >= 0;
        bool    multi = false;
        sck_ldr_      = me_;
        for (uint32_t r = 0; r < n_ && known; r++) {
            if (gid(r) < 0) {
                known = false;
            }
            if (gid(r) != rsck) {
                multi = true;
            }
            if (gid(r) == gid(me_) && r < sck_ldr_) {
                sck_ldr_ = r;
            }
        }
        sck_      = known && multi;
        root_sck_ = gid(me_) == rsck;
    }

    ucc_status_t prog_allreduce();
    ucc_status_t prog_reduce_scatter(); /* also v */
    ucc_status_t prog_allgather();      /* also v */
    ucc_status_t prog_bcast();
    ucc_status_t prog_reduce();
    ucc_status_t prog_barrier();
    ucc_status_t prog_alltoall();       /* also v, gather(v), scatter(v) */

    ShmTlTeam      *tt_;
    ucc_coll_args_t a_;
    /* normalized views */
    const uint8_t *sbuf_ = nullptr;
    uint8_t       *dbuf_ = nullptr;
    ucc_datatype_t dt_   = UCC_DT_INT8;
    size_t         dtsz_ = 1;
    ucc_reduction_op_t op_ = UCC_OP_SUM;
    size_t         total_ = 0; /* bytes of the logical vector           */
    std::vector<size_t> cnt_, dsp_;   /* per-rank bytes (v colls)       */
    std::vector<size_t> rcnt_, rdsp_; /* recv side for alltoallv        */
    bool   inplace_ = false;

    uint32_t me_ = 0, n_ = 1, slot_ = 0;
    uint64_t base_ = 0, k_ = 0;
    size_t   round_ = 0, nrounds_ = 0, cell_ = 0;
    int      phase_ = 0;
    double   alpha_ = 1.0;
    /* socket-aware bcast staging */
    bool     sck_ = false, sck_leader_ = false, root_sck_ = true;
    uint32_t sck_ldr_ = 0;
};

/* -------------------------------------------------------------- TL team */
class ShmTlTeam final : public TlTeam {
  public:
    ShmTlTeam(TlContext *tlc, Team *team, uint32_t nslots, size_t chunk)
        : TlTeam(tlc, team), nslots_(nslots), chunk_(chunk)
    {
        char buf[96];
        snprintf(buf, sizeof(buf), "/uccamd-%016llx-%u",
                 (unsigned long long)team->team_uid,
                 (unsigned)(team->team_uid >> 48));
        name_ = buf;
        if (team->rank == 0) {
            create_st_ = seg_.create(name_, team->size, nslots_, chunk_);
        }
    }

    ucc_status_t create_test() override
    {
        if (team_->rank == 0) {
            return create_st_;
        }
        if (!attached_) {
            ucc_status_t st =
                seg_.attach(name_, team_->size, nslots_, chunk_);
            if (st == UCC_INPROGRESS) {
                return UCC_INPROGRESS;
            }
            if (st != UCC_OK) {
                return st;
            }
            attached_ = true;
        }
        return seg_.ready() ? UCC_OK : UCC_INPROGRESS;
    }

    void get_scores(Team *team, ScoreMap &map) override
    {
        ShmTlTeam *self = this;
        for (int ci = 0; ci < UCC_COLL_TYPE_NUM; ci++) {
            ScoreRange r;
            r.score    = 40; /* above self(50)? no: below self for size1 */
            r.tl_name  = "shm";
            r.alg_name = "slotted";
            r.init     = [self](const ucc_coll_args_t &args, Team *t,
                            Task **task) -> ucc_status_t {
                if (args.mask & UCC_COLL_ARGS_FIELD_ACTIVE_SET) {
                    /* slot protocol needs every rank: subset colls go
                     * to tl/tcp (score fallback) */
                    return UCC_ERR_NOT_SUPPORTED;
                }
                const ucc_generic_dt_ops_t *g =
                    ucc_dt_generic_ops(args.src.info.datatype);
                if (g && !(g->flags & UCC_GENERIC_DT_OPS_FLAG_CONTIG)) {
                    /* non-contig generic dtypes: tcp pack/unpack path */
                    return UCC_ERR_NOT_SUPPORTED;
                }
                *task = new ShmCollTask(t->ctx, self, args);
                return UCC_OK;
            };
            map.add((ucc_coll_type_t)(1u << ci), UCC_MEMORY_TYPE_HOST, r);
        }
        (void)team;
    }

    ShmSeg   seg_;
    uint64_t seq_ = 0; /* posts routed to this TL, defines slot order    */
    uint32_t nslots_;
    size_t   chunk_;

  private:
    std::string  name_;
    ucc_status_t create_st_ = UCC_OK;
    bool         attached_  = false;
};

/* ----------------------------------------------------- task bodies      */
ShmCollTask::ShmCollTask(Context *ctx, ShmTlTeam *tt,
                         const ucc_coll_args_t &args)
    : Task(ctx), tt_(tt), a_(args)
{
}

inline void ShmCollTask::publish(uint64_t k)
{
    k_ = k;
    tt_->seg_.step(slot_, me_)->store(base_ + k,
                                      std::memory_order_release);
}

inline bool ShmCollTask::all_ge(uint64_t k)
{
    for (uint32_t r = 0; r < n_; r++) {
        if (tt_->seg_.step(slot_, r)->load(std::memory_order_acquire) <
            base_ + k) {
            return false;
        }
    }
    return true;
}

inline bool ShmCollTask::rank_ge(uint32_t r, uint64_t k)
{
    return tt_->seg_.step(slot_, r)->load(std::memory_order_acquire) >=
           base_ + k;
}

static size_t cnt_at(const ucc_coll_args_t &a, const ucc_count_t *c, int i)
{
    return (a.flags & UCC_COLL_ARGS_FLAG_COUNT_64BIT)
               ? (size_t)((const uint64_t *)c)[i]
               : (size_t)((const uint32_t *)c)[i];
}
static size_t dsp_at(const ucc_coll_args_t &a, const ucc_aint_t *d, int i)
{
    return (a.flags & UCC_COLL_ARGS_FLAG_DISPLACEMENTS_64BIT)
               ? (size_t)((const uint64_t *)d)[i]
               : (size_t)((const uint32_t *)d)[i];
}

ucc_status_t ShmCollTask::post()
{
    Team *team = tt_->team_;
    me_        = team->rank;
    n_         = team->size;
    inplace_   = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
    uint64_t use = tt_->seq_++;
    slot_        = (uint32_t)(use % tt_->nslots_);
    base_        = (use / tt_->nslots_) * kCap;
    k_           = 0;
    round_       = 0;
    phase_       = 0;
    alpha_       = 1.0;

    const size_t chunk = tt_->seg_.chunk();
    cell_              = (chunk / n_) & ~(size_t)15;

    switch (a_.coll_type) {
    case UCC_COLL_TYPE_ALLREDUCE:
        dt_    = a_.dst.info.datatype;
        op_    = a_.op;
        dtsz_  = ucc_dt_size(dt_);
        total_ = a_.dst.info.count * dtsz_;
        dbuf_  = (uint8_t *)a_.dst.info.buffer;
        sbuf_  = inplace_ ? dbuf_ : (const uint8_t *)a_.src.info.buffer;
        if (op_ == UCC_OP_AVG) {
            alpha_ = 1.0 / n_;
        }
        nrounds_ = (total_ + chunk - 1) / chunk;
        /* phase-C result reads relay through one leader per socket
         * (the result area is written scattered by every rank, so all
         * sockets benefit symmetrically) */
        calc_socket(team, me_);
        sck_leader_ = sck_ && sck_ldr_ == me_;
        break;
    case UCC_COLL_TYPE_REDUCE:
        dt_    = a_.src.info.datatype;
        op_    = a_.op;
        dtsz_  = ucc_dt_size(dt_);
        total_ = a_.src.info.count * dtsz_;
        dbuf_  = (uint8_t *)a_.dst.info.buffer;
        sbuf_  = (inplace_ && me_ == a_.root)
                     ? dbuf_
                     : (const uint8_t *)a_.src.info.buffer;
        if (op_ == UCC_OP_AVG) {
            alpha_ = 1.0 / n_;
        }
        nrounds_ = (total_ + chunk - 1) / chunk;
        break;
    case UCC_COLL_TYPE_BCAST:
        dt_      = a_.src.info.datatype;
        dtsz_    = ucc_dt_size(dt_);
        total_   = a_.src.info.count * dtsz_;
        dbuf_    = (uint8_t *)a_.src.info.buffer;
        sbuf_    = dbuf_;
        nrounds_ = (total_ + chunk - 1) / chunk;
        /* socket-aware two-level staging (reference ucc_sbgp SOCKET
         * kind consumption): when the node team spans CPU sockets,
         * one leader per non-root socket copies each chunk of the
         * root's result area into its own data area and its socket
         * peers read THAT — one cross-socket transfer per socket per
         * chunk instead of one per remote reader. */
        calc_socket(team, (uint32_t)a_.root);
        sck_leader_ = sck_ && !root_sck_ && sck_ldr_ == me_;
        break;
    case UCC_COLL_TYPE_BARRIER:
    case UCC_COLL_TYPE_FANIN:
    case UCC_COLL_TYPE_FANOUT:
        nrounds_ = 1;
        break;
    case UCC_COLL_TYPE_ALLGATHER: {
        dt_          = a_.dst.info.datatype;
        dtsz_        = ucc_dt_size(dt_);
        size_t block = a_.dst.info.count * dtsz_ / n_;
        cnt_.assign(n_, block);
        dsp_.resize(n_);
        for (uint32_t r = 0; r < n_; r++) {
            dsp_[r] = (size_t)r * block;
        }
        dbuf_ = (uint8_t *)a_.dst.info.buffer;
        sbuf_ = inplace_ ? dbuf_ + dsp_[me_]
                         : (const uint8_t *)a_.src.info.buffer;
        nrounds_ = (block + chunk - 1) / chunk;
        break;
    }
    case UCC_COLL_TYPE_ALLGATHERV: {
        dt_   = a_.dst.info_v.datatype;
        dtsz_ = ucc_dt_size(dt_);
        cnt_.resize(n_);
        dsp_.resize(n_);
        size_t maxb = 0;
        for (uint32_t r = 0; r < n_; r++) {
            cnt_[r] = cnt_at(a_, a_.dst.info_v.counts, r) * dtsz_;
            dsp_[r] = dsp_at(a_, a_.dst.info_v.displacements, r) * dtsz_;
            maxb    = cnt_[r] > maxb ? cnt_[r] : maxb;
        }
        dbuf_ = (uint8_t *)a_.dst.info_v.buffer;
        sbuf_ = inplace_ ? dbuf_ + dsp_[me_]
                         : (const uint8_t *)a_.src.info.buffer;
        nrounds_ = (maxb + chunk - 1) / chunk;
        break;
    }
    case UCC_COLL_TYPE_REDUCE_SCATTER: {
        dt_    = a_.dst.info.datatype;
        op_    = a_.op;
        dtsz_  = ucc_dt_size(dt_);
        if (op_ == UCC_OP_AVG) {
            alpha_ = 1.0 / n_;
        }
        size_t out_bytes;
        if (inplace_) {
            /* src vector lives in dst buffer (size = count) */
            total_    = a_.dst.info.count * dtsz_;
            out_bytes = total_ / n_;
            sbuf_     = (uint8_t *)a_.dst.info.buffer;
            dbuf_     = (uint8_t *)a_.dst.info.buffer + me_ * out_bytes;
        } else {
            out_bytes = a_.dst.info.count * dtsz_;
            total_    = out_bytes * n_;
            sbuf_     = (const uint8_t *)a_.src.info.buffer;
            dbuf_     = (uint8_t *)a_.dst.info.buffer;
        }
        cnt_.assign(n_, out_bytes);
        dsp_.resize(n_);
        for (uint32_t r = 0; r < n_; r++) {
            dsp_[r] = (size_t)r * out_bytes;
        }
        if (cell_ == 0) {
            return UCC_ERR_NOT_SUPPORTED;
        }
        size_t maxb = out_bytes;
        nrounds_    = (maxb + cell_ - 1) / cell_;
        break;
    }
    case UCC_COLL_TYPE_REDUCE_SCATTERV: {
        dt_    = a_.dst.info_v.datatype;
        op_    = a_.op;
        dtsz_  = ucc_dt_size(dt_);
        if (op_ == UCC_OP_AVG) {
            alpha_ = 1.0 / n_;
        }
        cnt_.resize(n_);
        dsp_.resize(n_);
        size_t off = 0, maxb = 0;
        for (uint32_t r = 0; r < n_; r++) {
            cnt_[r] = cnt_at(a_, a_.dst.info_v.counts, r) * dtsz_;
            dsp_[r] = off;
            off += cnt_[r];
            maxb = cnt_[r] > maxb ? cnt_[r] : maxb;
        }
        total_ = off;
        sbuf_  = inplace_ ? (const uint8_t *)a_.dst.info_v.buffer
                          : (const uint8_t *)a_.src.info.buffer;
        dbuf_  = inplace_
                     ? (uint8_t *)a_.dst.info_v.buffer + dsp_[me_]
                     : (uint8_t *)a_.dst.info_v.buffer;
        if (cell_ == 0) {
            return UCC_ERR_NOT_SUPPORTED;
        }
        nrounds_ = (maxb + cell_ - 1) / cell_;
        break;
    }
    case UCC_COLL_TYPE_ALLTOALL: {
        dt_          = a_.dst.info.datatype;
        dtsz_        = ucc_dt_size(dt_);
        size_t block = a_.dst.info.count * dtsz_ / n_;
        cnt_.assign(n_, block);
        rcnt_.assign(n_, block);
        dsp_.resize(n_);
        rdsp_.resize(n_);
        for (uint32_t r = 0; r < n_; r++) {
            dsp_[r] = rdsp_[r] = (size_t)r * block;
        }
        dbuf_ = (uint8_t *)a_.dst.info.buffer;
        sbuf_ = inplace_ ? dbuf_ : (const uint8_t *)a_.src.info.buffer;
        if (cell_ == 0) {
            return UCC_ERR_NOT_SUPPORTED;
        }
        nrounds_ = (block + cell_ - 1) / cell_;
        break;
    }
    case UCC_COLL_TYPE_ALLTOALLV: {
        dt_   = a_.src.info_v.datatype;
        dtsz_ = ucc_dt_size(dt_);
        cnt_.resize(n_);
        dsp_.resize(n_);
        rcnt_.resize(n_);
        rdsp_.resize(n_);
        size_t maxb = 0;
        for (uint32_t r = 0; r < n_; r++) {
            cnt_[r]  = cnt_at(a_, a_.src.info_v.counts, r) * dtsz_;
            dsp_[r]  = dsp_at(a_, a_.src.info_v.displacements, r) * dtsz_;
            rcnt_[r] = cnt_at(a_, a_.dst.info_v.counts, r) *
                       ucc_dt_size(a_.dst.info_v.datatype);
            rdsp_[r] = dsp_at(a_, a_.dst.info_v.displacements, r) *
                       ucc_dt_size(a_.dst.info_v.datatype);
            maxb = cnt_[r] > maxb ? cnt_[r] : maxb;
            maxb = rcnt_[r] > maxb ? rcnt_[r] : maxb;
        }
        sbuf_ = (const uint8_t *)a_.src.info_v.buffer;
        dbuf_ = (uint8_t *)a_.dst.info_v.buffer;
        if (cell_ == 0) {
            return UCC_ERR_NOT_SUPPORTED;
        }
        /* per-rank round counts differ: agree via metadata round */
        nrounds_ = (maxb + cell_ - 1) / cell_;
        break;
    }
    case UCC_COLL_TYPE_GATHER:
    case UCC_COLL_TYPE_GATHERV:
    case UCC_COLL_TYPE_SCATTER:
    case UCC_COLL_TYPE_SCATTERV: {
        bool gather = a_.coll_type == UCC_COLL_TYPE_GATHER ||
                      a_.coll_type == UCC_COLL_TYPE_GATHERV;
        bool is_v = a_.coll_type == UCC_COLL_TYPE_GATHERV ||
                    a_.coll_type == UCC_COLL_TYPE_SCATTERV;
        cnt_.assign(n_, 0);
        dsp_.assign(n_, 0);
        rcnt_.assign(n_, 0);
        rdsp_.assign(n_, 0);
        if (gather) {
            /* dst at root holds all blocks */
            if (me_ == a_.root) {
                dt_   = is_v ? a_.dst.info_v.datatype : a_.dst.info.datatype;
                dtsz_ = ucc_dt_size(dt_);
                if (is_v) {
                    for (uint32_t r = 0; r < n_; r++) {
                        rcnt_[r] = cnt_at(a_, a_.dst.info_v.counts, r) * dtsz_;
                        rdsp_[r] =
                            dsp_at(a_, a_.dst.info_v.displacements, r) * dtsz_;
                    }
                } else {
                    size_t block = a_.dst.info.count * dtsz_ / n_;
                    for (uint32_t r = 0; r < n_; r++) {
                        rcnt_[r] = block;
                        rdsp_[r] = (size_t)r * block;
                    }
                }
                dbuf_ = is_v ? (uint8_t *)a_.dst.info_v.buffer
                             : (uint8_t *)a_.dst.info.buffer;
                sbuf_ = inplace_ ? dbuf_ + rdsp_[me_]
                                 : (const uint8_t *)a_.src.info.buffer;
                cnt_[me_] = rcnt_[me_];
            } else {
                dt_   = a_.src.info.datatype;
                dtsz_ = ucc_dt_size(dt_);
                sbuf_ = (const uint8_t *)a_.src.info.buffer;
                cnt_[a_.root] = a_.src.info.count * dtsz_;
            }
        } else { /* scatter */
            if (me_ == a_.root) {
                dt_   = is_v ? a_.src.info_v.datatype : a_.src.info.datatype;
                dtsz_ = ucc_dt_size(dt_);
                if (is_v) {
                    for (uint32_t r = 0; r < n_; r++) {
                        cnt_[r] = cnt_at(a_, a_.src.info_v.counts, r) * dtsz_;
                        dsp_[r] =
                            dsp_at(a_, a_.src.info_v.displacements, r) * dtsz_;
                    }
                } else {
                    size_t block = a_.src.info.count * dtsz_ / n_;
                    for (uint32_t r = 0; r < n_; r++) {
                        cnt_[r] = block;
                        dsp_[r] = (size_t)r * block;
                    }
                }
                sbuf_ = is_v ? (const uint8_t *)a_.src.info_v.buffer
                             : (const uint8_t *)a_.src.info.buffer;
                dbuf_ = inplace_ ? (uint8_t *)(uintptr_t)(sbuf_ + dsp_[me_])
                                 : (uint8_t *)a_.dst.info.buffer;
                rcnt_[me_] = cnt_[me_];
            } else {
                dt_   = a_.dst.info.datatype;
                dtsz_ = ucc_dt_size(dt_);
                dbuf_ = (uint8_t *)a_.dst.info.buffer;
                rcnt_[a_.root] = a_.dst.info.count * dtsz_;
            }
        }
        /* rounds agreed via metadata round (root may not know leaf counts
         * for v; leaves do not know total) */
        nrounds_ = 0;
        break;
    }
    default: return UCC_ERR_NOT_SUPPORTED;
    }
    status = UCC_INPROGRESS;
    ucc_status_t st = progress();
    return st;
}

/* ---- allreduce: per round j (parity p=j%2), steps 3j+1..3j+3:
 *  A: [wait all >= 3(j-2)+2 (parity reuse)] copy my frag -> data[me][p]
 *  B: [wait all >= 3j+1] reduce my 1/n sub-slice of frag -> result[p]
 *  C: [wait all >= 3j+2] copy result frag -> dst; publish 3j+3.
 * Entry condition: all >= 0 (previous slot use finished).             */
ucc_status_t ShmCollTask::prog_allreduce()
{
    auto  &seg   = tt_->seg_;
    size_t chunk = seg.chunk();
    while (round_ < nrounds_) {
        size_t   j   = round_;
        uint32_t p   = j & 1;
        size_t   off = j * chunk;
        size_t   len = total_ - off < chunk ? total_ - off : chunk;
        switch (phase_) {
        case 0: /* A */
            /* sck_: leaders' data areas double as phase-C relays, so
             * reuse waits for the full previous parity round (+3) */
            if (j < 2 ? !all_ge(0)
                      : !all_ge(3 * (j - 2) + (sck_ ? 3 : 2))) {
                return UCC_INPROGRESS;
            }
            memcpy(seg.data(slot_, p, me_), sbuf_ + off, len);
            publish(3 * j + 1);
            phase_ = 1;
            break;
        case 1: { /* B: reduce my sub-slice across ranks */
            if (!all_ge(3 * j + 1)) {
                return UCC_INPROGRESS;
            }
            size_t nelem = len / dtsz_;
            size_t per   = nelem / n_;
            size_t b     = me_ * per;
            size_t e     = (me_ == n_ - 1) ? nelem : b + per;
            if (e > b) {
                const void *srcs[256];
                for (uint32_t r = 0; r < n_; r++) {
                    srcs[r] = seg.data(slot_, p, r) + b * dtsz_;
                }
                ec_cpu::reduce(seg.result(slot_, p) + b * dtsz_, srcs, n_,
                               e - b, dt_, op_, alpha_);
            }
            /* tail bytes (len not multiple of dtsz_ cannot happen) */
            publish(3 * j + 2);
            phase_ = 2;
            break;
        }
        case 2: /* C */
            if (sck_ && !sck_leader_) {
                /* read my socket leader's local relay copy */
                if (!rank_ge(sck_ldr_, 3 * j + 3)) {
                    return UCC_INPROGRESS;
                }
                memcpy(dbuf_ + off, seg.data(slot_, p, sck_ldr_), len);
            } else {
                if (!all_ge(3 * j + 2)) {
                    return UCC_INPROGRESS;
                }
                memcpy(dbuf_ + off, seg.result(slot_, p), len);
                if (sck_leader_) { /* publish the relay for my socket */
                    memcpy(seg.data(slot_, p, me_), dbuf_ + off, len);
                }
            }
            publish(3 * j + 3);
            phase_ = 0;
            round_++;
            break;
        }
    }
    /* close the slot use */
    tt_->seg_.step(slot_, me_)->store(base_ + kCap,
                                      std::memory_order_release);
    return UCC_OK;
}

/* ---- reduce_scatter(v): cells. Round j, steps 2j+1,2j+2:
 *  W: [reuse: all >= 2(j-2)+2] write cell (me->d) = frag j of slice d
 *     for all d, into data[me][p] cell d; publish.
 *  R: [all >= 2j+1] reduce my cells across ranks into dst frag j.     */
ucc_status_t ShmCollTask::prog_reduce_scatter()
{
    auto &seg = tt_->seg_;
    while (round_ < nrounds_) {
        size_t   j = round_;
        uint32_t p = j & 1;
        switch (phase_) {
        case 0: {
            if (j < 2 ? !all_ge(0) : !all_ge(2 * (j - 2) + 2)) {
                return UCC_INPROGRESS;
            }
            uint8_t *out = seg.data(slot_, p, me_);
            for (uint32_t d = 0; d < n_; d++) {
                size_t off = j * cell_;
                if (off >= cnt_[d]) {
                    continue;
                }
                size_t len = cnt_[d] - off < cell_ ? cnt_[d] - off : cell_;
                memcpy(out + (size_t)d * cell_, sbuf_ + dsp_[d] + off, len);
            }
            publish(2 * j + 1);
            phase_ = 1;
            break;
        }
        case 1: {
            if (!all_ge(2 * j + 1)) {
                return UCC_INPROGRESS;
            }
            size_t off = j * cell_;
            if (off < cnt_[me_]) {
                size_t len =
                    cnt_[me_] - off < cell_ ? cnt_[me_] - off : cell_;
                const void *srcs[256];
                for (uint32_t r = 0; r < n_; r++) {
                    srcs[r] = seg.data(slot_, p, r) + (size_t)me_ * cell_;
                }
                ec_cpu::reduce(dbuf_ + off, srcs, n_, len / dtsz_, dt_, op_,
                               alpha_);
            }
            publish(2 * j + 2);
            phase_ = 0;
            round_++;
            break;
        }
        }
    }
    tt_->seg_.step(slot_, me_)->store(base_ + kCap,
                                      std::memory_order_release);
    return UCC_OK;
}

/* ---- allgather(v): round j: W writes frag j of my block; R reads all. */
ucc_status_t ShmCollTask::prog_allgather()
{
    auto  &seg   = tt_->seg_;
    size_t chunk = seg.chunk();
    while (round_ < nrounds_) {
        size_t   j = round_;
        uint32_t p = j & 1;
        switch (phase_) {
        case 0: {
            if (j < 2 ? !all_ge(0) : !all_ge(2 * (j - 2) + 2)) {
                return UCC_INPROGRESS;
            }
            size_t off = j * chunk;
            if (off < cnt_[me_]) {
                size_t len =
                    cnt_[me_] - off < chunk ? cnt_[me_] - off : chunk;
                memcpy(seg.data(slot_, p, me_), sbuf_ + off, len);
            }
            publish(2 * j + 1);
            phase_ = 1;
            break;
        }
        case 1: {
            if (!all_ge(2 * j + 1)) {
                return UCC_INPROGRESS;
            }
            size_t off = j * chunk;
            for (uint32_t r = 0; r < n_; r++) {
                if (off >= cnt_[r]) {
                    continue;
                }
                size_t len =
                    cnt_[r] - off < chunk ? cnt_[r] - off : chunk;
                if (r == me_ && inplace_) {
                    continue; /* already in place */
                }
                memcpy(dbuf_ + dsp_[r] + off, seg.data(slot_, p, r), len);
            }
            publish(2 * j + 2);
            phase_ = 0;
            round_++;
            break;
        }
        }
    }
    tt_->seg_.step(slot_, me_)->store(base_ + kCap,
                                      std::memory_order_release);
    return UCC_OK;
}

/* ---- bcast: root stages frag into result[p]; all consume.            */
ucc_status_t ShmCollTask::prog_bcast()
{
    auto  &seg   = tt_->seg_;
    size_t chunk = seg.chunk();
    while (round_ < nrounds_) {
        size_t   j   = round_;
        uint32_t p   = j & 1;
        size_t   off = j * chunk;
        size_t   len = total_ - off < chunk ? total_ - off : chunk;
        switch (phase_) {
        case 0:
            if (me_ == a_.root) {
                if (j < 2 ? !all_ge(0) : !all_ge(2 * (j - 2) + 2)) {
                    return UCC_INPROGRESS;
                }
                memcpy(seg.result(slot_, p), sbuf_ + off, len);
            } else {
                if (j >= 2 && !all_ge(2 * (j - 2) + 2)) {
                    return UCC_INPROGRESS;
                }
                if (sck_leader_) {
                    /* relay: ONE cross-socket read of the root chunk,
                     * written to my dst and my data area for my
                     * socket's peers (they poll my 2j+1 publication) */
                    if (!root_ge(2 * j + 1)) {
                        return UCC_INPROGRESS;
                    }
                    memcpy(dbuf_ + off, seg.result(slot_, p), len);
                    memcpy(seg.data(slot_, p, me_), dbuf_ + off, len);
                }
            }
            publish(2 * j + 1);
            phase_ = 1;
            break;
        case 1:
            if (me_ == a_.root) {
                if (!all_ge(2 * j + 1)) {
                    return UCC_INPROGRESS;
                }
            } else if (sck_ && !root_sck_) {
                if (!sck_leader_) { /* read my socket leader's relay */
                    if (!rank_ge(sck_ldr_, 2 * j + 1)) {
                        return UCC_INPROGRESS;
                    }
                    memcpy(dbuf_ + off, seg.data(slot_, p, sck_ldr_),
                           len);
                } /* leader already copied in phase 0 */
            } else {
                if (!root_ge(2 * j + 1)) {
                    return UCC_INPROGRESS;
                }
                memcpy(dbuf_ + off, seg.result(slot_, p), len);
            }
            publish(2 * j + 2);
            phase_ = 0;
            round_++;
            break;
        }
    }
    tt_->seg_.step(slot_, me_)->store(base_ + kCap,
                                      std::memory_order_release);
    return UCC_OK;
}

/* ---- reduce to root: W stage frag; root reduces into dst.            */
ucc_status_t ShmCollTask::prog_reduce()
{
    auto  &seg   = tt_->seg_;
    size_t chunk = seg.chunk();
    while (round_ < nrounds_) {
        size_t   j   = round_;
        uint32_t p   = j & 1;
        size_t   off = j * chunk;
        size_t   len = total_ - off < chunk ? total_ - off : chunk;
        switch (phase_) {
        case 0:
            if (j < 2 ? !all_ge(0) : !all_ge(2 * (j - 2) + 2)) {
                return UCC_INPROGRESS;
            }
            memcpy(seg.data(slot_, p, me_), sbuf_ + off, len);
            publish(2 * j + 1);
            phase_ = 1;
            break;
        case 1:
            if (me_ == a_.root) {
                if (!all_ge(2 * j + 1)) {
                    return UCC_INPROGRESS;
                }
                const void *srcs[256];
                for (uint32_t r = 0; r < n_; r++) {
                    srcs[r] = seg.data(slot_, p, r);
                }
                ec_cpu::reduce(dbuf_ + off, srcs, n_, len / dtsz_, dt_, op_,
                               alpha_);
            }
            publish(2 * j + 2);
            phase_ = 0;
            round_++;
            break;
        }
    }
    tt_->seg_.step(slot_, me_)->store(base_ + kCap,
                                      std::memory_order_release);
    return UCC_OK;
}

ucc_status_t ShmCollTask::prog_barrier()
{
    switch (phase_) {
    case 0:
        if (!all_ge(0)) {
            return UCC_INPROGRESS;
        }
        publish(1);
        phase_ = 1;
        /* fallthrough */
    case 1:
        if (a_.coll_type == UCC_COLL_TYPE_FANIN && me_ != a_.root) {
            /* leaves do not wait */
        } else if (a_.coll_type == UCC_COLL_TYPE_FANOUT && me_ != a_.root) {
            if (!root_ge(1)) {
                return UCC_INPROGRESS;
            }
        } else {
            if (!all_ge(1)) {
                return UCC_INPROGRESS;
            }
        }
        break;
    }
    tt_->seg_.step(slot_, me_)->store(base_ + kCap,
                                      std::memory_order_release);
    return UCC_OK;
}

/* ---- alltoall(v) + gather(v)/scatter(v): cell-staged rounds with a
 * metadata round 0 agreeing on the global round count.
 * meta: result[0] holds n u64s; each rank writes its max local rounds.
 * Then round j (steps: meta=1, W=2+2j, R=3+2j):
 *  W: write cell (me->d) frag j into data[me][p] cell d (d==me: direct)
 *  R: read cell (s->me) frag j from data[s][p] cell me.                */
ucc_status_t ShmCollTask::prog_alltoall()
{
    auto &seg = tt_->seg_;
    switch (phase_) {
    case 0: { /* publish my needed rounds */
        if (!all_ge(0)) {
            return UCC_INPROGRESS;
        }
        size_t mine = 0;
        for (uint32_t r = 0; r < n_; r++) {
            size_t need_s = (cnt_[r] + cell_ - 1) / cell_;
            size_t need_r = (rcnt_[r] + cell_ - 1) / cell_;
            mine = need_s > mine ? need_s : mine;
            mine = need_r > mine ? need_r : mine;
        }
        ((std::atomic<uint64_t> *)seg.result(slot_, 0))[me_].store(
            mine, std::memory_order_relaxed);
        publish(1);
        phase_ = 1;
        break;
    }
    case 1: {
        if (!all_ge(1)) {
            return UCC_INPROGRESS;
        }
        size_t g = 0;
        for (uint32_t r = 0; r < n_; r++) {
            uint64_t v = ((std::atomic<uint64_t> *)seg.result(slot_, 0))[r]
                             .load(std::memory_order_relaxed);
            g = v > g ? v : g;
        }
        nrounds_ = g;
        round_   = 0;
        phase_   = 2;
        break;
    }
    default: break;
    }
    if (phase_ < 2) {
        return UCC_INPROGRESS;
    }
    while (round_ < nrounds_) {
        size_t   j = round_;
        uint32_t p = j & 1;
        if (phase_ == 2) { /* W */
            if (j >= 2 && !all_ge(1 + 2 * (j - 2) + 2)) {
                return UCC_INPROGRESS;
            }
            uint8_t *out = seg.data(slot_, p, me_);
            size_t   off = j * cell_;
            for (uint32_t d = 0; d < n_; d++) {
                if (off >= cnt_[d]) {
                    continue;
                }
                size_t len = cnt_[d] - off < cell_ ? cnt_[d] - off : cell_;
                memcpy(out + (size_t)d * cell_, sbuf_ + dsp_[d] + off, len);
            }
            publish(1 + 2 * j + 1);
            phase_ = 3;
        } else { /* R */
            size_t off = j * cell_;
            /* need every sender whose block to me reaches this round */
            for (uint32_t s = 0; s < n_; s++) {
                if (off < rcnt_[s] && !rank_ge(s, 1 + 2 * j + 1)) {
                    return UCC_INPROGRESS;
                }
            }
            for (uint32_t s = 0; s < n_; s++) {
                if (off >= rcnt_[s]) {
                    continue;
                }
                size_t len =
                    rcnt_[s] - off < cell_ ? rcnt_[s] - off : cell_;
                memcpy(dbuf_ + rdsp_[s] + off,
                       seg.data(slot_, p, s) + (size_t)me_ * cell_, len);
            }
            publish(1 + 2 * j + 2);
            phase_ = 2;
            round_++;
        }
    }
    tt_->seg_.step(slot_, me_)->store(base_ + kCap,
                                      std::memory_order_release);
    return UCC_OK;
}

ucc_status_t ShmCollTask::progress()
{
    switch (a_.coll_type) {
    case UCC_COLL_TYPE_ALLREDUCE: return prog_allreduce();
    case UCC_COLL_TYPE_REDUCE_SCATTER:
    case UCC_COLL_TYPE_REDUCE_SCATTERV: return prog_reduce_scatter();
    case UCC_COLL_TYPE_ALLGATHER:
    case UCC_COLL_TYPE_ALLGATHERV: return prog_allgather();
    case UCC_COLL_TYPE_BCAST: return prog_bcast();
    case UCC_COLL_TYPE_REDUCE: return prog_reduce();
    case UCC_COLL_TYPE_BARRIER:
    case UCC_COLL_TYPE_FANIN:
    case UCC_COLL_TYPE_FANOUT: return prog_barrier();
    case UCC_COLL_TYPE_ALLTOALL:
    case UCC_COLL_TYPE_ALLTOALLV:
    case UCC_COLL_TYPE_GATHER:
    case UCC_COLL_TYPE_GATHERV:
    case UCC_COLL_TYPE_SCATTER:
    case UCC_COLL_TYPE_SCATTERV: return prog_alltoall();
    default: return UCC_ERR_NOT_SUPPORTED;
    }
}

/* ------------------------------------------------------------ component */
class ShmTlContext final : public TlContext {
  public:
    ShmTlContext(Context *ctx, Tl *tl) : TlContext(ctx), tl_(tl) {}
    Tl *iface() override { return tl_; }
    Tl *tl_;
};

class ShmTl final : public Tl {
  public:
    const char *name() const override { return "shm"; }
    int         default_score() const override { return 40; }
    TlContext  *context_create(Context *ctx) override
    {
        auto &cfg = Config::instance();
        cfg.declare("TL_SHM", "ENABLE", "1", "enable the shm TL");
        cfg.declare("TL_SHM", "CHUNK_SIZE", "512k",
                    "shm staging chunk bytes per round");
        cfg.declare("TL_SHM", "MAX_CONCURRENT", "4",
                    "concurrent shm collective slots");
        cfg.declare("TL_SHM", "SOCKET_STAGING", "1",
                    "bcast: relay chunks through one leader per "
                    "non-root CPU socket (one cross-socket read per "
                    "socket instead of per rank)");
        if (!cfg.get_bool("TL_SHM", "ENABLE", true)) {
            return nullptr;
        }
        return new ShmTlContext(ctx, this);
    }
    TlTeam *team_create(TlContext *tlc, Team *team) override
    {
        if (team->size < 2 || team->size > 256 || !team->all_same_node()) {
            return nullptr;
        }
        auto  &cfg    = Config::instance();
        size_t chunk  = cfg.get_size("TL_SHM", "CHUNK_SIZE", 512 * 1024);
        size_t nslots = (size_t)cfg.get_int("TL_SHM", "MAX_CONCURRENT", 4);
        return new ShmTlTeam(tlc, team, (uint32_t)nslots, chunk);
    }
};

} // namespace

Tl *tl_shm_iface()
{
    static ShmTl tl;
    return &tl;
}

} // namespace ucc
