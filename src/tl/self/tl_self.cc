/* TL "self": team-size-1 fast path — every collective is a local copy or a
 * no-op. Parity: reference components/tl/self/. Also what makes single-rank
 * tests and 1-GPU bench runs hardware-independent. */
#include "../../core/core.h"
#include "../../mc/mc.h"

namespace ucc {
namespace {

size_t counts_elem(const ucc_coll_args_t &a, const ucc_count_t *c, int idx)
{
    if (a.flags & UCC_COLL_ARGS_FLAG_COUNT_64BIT) {
        return ((const uint64_t *)c)[idx];
    }
    return ((const uint32_t *)c)[idx];
}

size_t displ_elem(const ucc_coll_args_t &a, const ucc_aint_t *d, int idx)
{
    if (a.flags & UCC_COLL_ARGS_FLAG_DISPLACEMENTS_64BIT) {
        return (size_t)((const uint64_t *)d)[idx];
    }
    return (size_t)((const uint32_t *)d)[idx];
}

class SelfTask final : public Task {
  public:
    SelfTask(Context *ctx, const ucc_coll_args_t &args) : Task(ctx), a_(args)
    {
    }

    ucc_status_t post() override
    {
        const bool inplace = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        switch (a_.coll_type) {
        case UCC_COLL_TYPE_BARRIER:
        case UCC_COLL_TYPE_FANIN:
        case UCC_COLL_TYPE_FANOUT:
        case UCC_COLL_TYPE_BCAST: /* single buffer: root already has it
                                   * (dst is unused/null for bcast) */
            return UCC_OK;
        case UCC_COLL_TYPE_ALLTOALLV: {
            if (inplace) {
                return UCC_OK;
            }
            size_t dt = ucc_dt_size(a_.src.info_v.datatype);
            size_t cnt = counts_elem(a_, a_.src.info_v.counts, 0);
            size_t sd  = displ_elem(a_, a_.src.info_v.displacements, 0);
            size_t dd  = displ_elem(a_, a_.dst.info_v.displacements, 0);
            return mc::copy((uint8_t *)a_.dst.info_v.buffer + dd * dt,
                            a_.dst.info_v.mem_type,
                            (const uint8_t *)a_.src.info_v.buffer + sd * dt,
                            a_.src.info_v.mem_type, cnt * dt);
        }
        case UCC_COLL_TYPE_ALLGATHERV:
        case UCC_COLL_TYPE_GATHERV:
        case UCC_COLL_TYPE_REDUCE_SCATTERV: {
            if (inplace) {
                return UCC_OK;
            }
            size_t dt  = ucc_dt_size(a_.dst.info_v.datatype);
            size_t cnt = counts_elem(a_, a_.dst.info_v.counts, 0);
            size_t dd  = displ_elem(a_, a_.dst.info_v.displacements, 0);
            return mc::copy((uint8_t *)a_.dst.info_v.buffer + dd * dt,
                            a_.dst.info_v.mem_type, a_.src.info.buffer,
                            a_.src.info.mem_type, cnt * dt);
        }
        case UCC_COLL_TYPE_SCATTERV: {
            if (inplace) {
                return UCC_OK;
            }
            size_t dt  = ucc_dt_size(a_.src.info_v.datatype);
            size_t cnt = counts_elem(a_, a_.src.info_v.counts, 0);
            size_t sd  = displ_elem(a_, a_.src.info_v.displacements, 0);
            return mc::copy(a_.dst.info.buffer, a_.dst.info.mem_type,
                            (const uint8_t *)a_.src.info_v.buffer + sd * dt,
                            a_.src.info_v.mem_type, cnt * dt);
        }
        default: {
            if (inplace) {
                return UCC_OK;
            }
            /* contiguous copy: min(src,dst) bytes */
            size_t sb =
                a_.src.info.count * ucc_dt_size(a_.src.info.datatype);
            size_t db =
                a_.dst.info.count * ucc_dt_size(a_.dst.info.datatype);
            size_t bytes = sb < db ? sb : db;
            if (bytes == 0) {
                return UCC_OK;
            }
            return mc::copy(a_.dst.info.buffer, a_.dst.info.mem_type,
                            a_.src.info.buffer, a_.src.info.mem_type, bytes);
        }
        }
    }

  private:
    ucc_coll_args_t a_;
};

class SelfTlTeam final : public TlTeam {
  public:
    using TlTeam::TlTeam;
    void get_scores(Team *team, ScoreMap &map) override
    {
        (void)team;
        for (int ci = 0; ci < UCC_COLL_TYPE_NUM; ci++) {
            for (int mi = 0; mi < UCC_MEMORY_TYPE_LAST; mi++) {
                ScoreRange r;
                r.score    = 50; /* reference tl/self score */
                r.tl_name  = "self";
                r.alg_name = "self";
                r.init     = [](const ucc_coll_args_t &args, Team *t,
                            Task **task) {
                    *task = new SelfTask(t->ctx, args);
                    return UCC_OK;
                };
                map.add((ucc_coll_type_t)(1u << ci), (ucc_memory_type_t)mi,
                        r);
            }
        }
    }
};

class SelfTlContext final : public TlContext {
  public:
    SelfTlContext(Context *ctx, Tl *tl) : TlContext(ctx), tl_(tl) {}
    Tl *iface() override { return tl_; }
    Tl *tl_;
};

class SelfTl final : public Tl {
  public:
    const char *name() const override { return "self"; }
    int         default_score() const override { return 50; }
    TlContext  *context_create(Context *ctx) override
    {
        return new SelfTlContext(ctx, this);
    }
    TlTeam *team_create(TlContext *tlc, Team *team) override
    {
        if (team->size != 1) {
            return nullptr;
        }
        return new SelfTlTeam(tlc, team);
    }
};

} // namespace

Tl *tl_self_iface()
{
    static SelfTl tl;
    return &tl;
}

} // namespace ucc
