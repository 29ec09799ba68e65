/* Algorithm selection: per-(coll_type x mem_type) msg-size ranges with
 * scores, user tuning-string overrides and highest-score-first dispatch
 * with fallback. Parity: reference src/coll_score/ (ucc_coll_score.h range
 * lists, UCC_TL_*_TUNE string grammar, score-map O(1) dispatch). */
#include "../core/core.h"

#include <algorithm>
#include <cstdlib>
#include <cstring>
#include <sstream>

namespace ucc {

int coll_type_index(ucc_coll_type_t ct)
{
    int idx = 0;
    unsigned v = (unsigned)ct;
    while (v > 1) {
        v >>= 1;
        idx++;
    }
    return idx;
}

static const char *k_coll_names[UCC_COLL_TYPE_NUM] = {
    "allgather",     "allgatherv",      "allreduce", "alltoall",
    "alltoallv",     "barrier",         "bcast",     "fanin",
    "fanout",        "gather",          "gatherv",   "reduce",
    "reduce_scatter","reduce_scatterv", "scatter",   "scatterv",
};

const char *coll_type_name(ucc_coll_type_t ct)
{
    int i = coll_type_index(ct);
    return (i >= 0 && i < UCC_COLL_TYPE_NUM) ? k_coll_names[i] : "?";
}

ucc_status_t coll_type_from_name(const std::string &s, ucc_coll_type_t *ct)
{
    for (int i = 0; i < UCC_COLL_TYPE_NUM; i++) {
        if (s == k_coll_names[i]) {
            *ct = (ucc_coll_type_t)(1u << i);
            return UCC_OK;
        }
    }
    return UCC_ERR_NOT_FOUND;
}

extern "C" ucc_status_t ucc_amd_coll_from_name_c(const char *s,
                                                 ucc_coll_type_t *ct)
{
    return coll_type_from_name(s, ct);
}

const char *mem_type_name(ucc_memory_type_t mt)
{
    switch (mt) {
    case UCC_MEMORY_TYPE_HOST: return "host";
    case UCC_MEMORY_TYPE_CUDA: return "cuda";
    case UCC_MEMORY_TYPE_CUDA_MANAGED: return "cuda_managed";
    case UCC_MEMORY_TYPE_ROCM: return "rocm";
    case UCC_MEMORY_TYPE_ROCM_MANAGED: return "rocm_managed";
    default: return "unknown";
    }
}

void ScoreMap::add(ucc_coll_type_t ct, ucc_memory_type_t mt, ScoreRange r)
{
    auto &v = ranges[coll_type_index(ct)][mt];
    v.push_back(std::move(r));
    std::stable_sort(v.begin(), v.end(),
                     [](const ScoreRange &a, const ScoreRange &b) {
                         return a.score > b.score;
                     });
}

ucc_status_t ScoreMap::init_coll(const ucc_coll_args_t &args, Team *team,
                                 size_t msgsize, Task **task) const
{
    const auto  mt = coll_args_mem_type(args, team->rank);
    if (mt >= UCC_MEMORY_TYPE_LAST) {
        return UCC_ERR_INVALID_PARAM;
    }
    const auto &v  = ranges[coll_type_index(args.coll_type)][mt];
    ucc_status_t last = UCC_ERR_NOT_SUPPORTED;
    for (const auto &r : v) {
        if (msgsize < r.start || msgsize > r.end || r.score <= 0) {
            continue;
        }
        ucc_status_t st = r.init(args, team, task);
        if (st == UCC_OK) {
            /* UCC_COLL_TRACE=1: per-collective selection print at INFO
             * (reference UCC_COLL_TRACE, core/ucc_coll.c:329-345) */
            static const bool trace = [] {
                const char *e = getenv("UCC_COLL_TRACE");
                return e && *e && strcmp(e, "0") != 0;
            }();
            UCC_LOG(trace ? LogLevel::INFO : LogLevel::DEBUG, "score",
                    "coll %s mem %s size %zu -> %s/%s (score %d)",
                    coll_type_name(args.coll_type), mem_type_name(mt), msgsize,
                    r.tl_name.c_str(), r.alg_name.c_str(), r.score);
            return UCC_OK;
        }
        last = st;
        UCC_LOG(LogLevel::DEBUG, "score", "init %s/%s failed (%d), fallback",
                r.tl_name.c_str(), r.alg_name.c_str(), st);
    }
    return last;
}

/* Tuning string grammar (reference UCC_TL_*_TUNE compatible subset):
 *   entry[,entry...]
 *   entry := [coll:][msgrange:][mem:][@alg:]score
 *   msgrange := start-end with k/m/g suffixes, "inf" for open end
 *   score := integer | "inf"(=INT_MAX) | "0" (disable)
 * Entries match by TL alg name when @alg given, else all ranges of the
 * matching coll/mem. */
ucc_status_t ScoreMap::apply_str(const std::string &str)
{
    std::stringstream ss(str);
    std::string       entry;
    while (std::getline(ss, entry, ',')) {
        if (entry.empty()) {
            continue;
        }
        ucc_coll_type_t ct_filter   = (ucc_coll_type_t)0;
        bool            have_ct     = false;
        size_t          r_start = 0, r_end = SIZE_MAX;
        bool            have_range  = false;
        int             mt_filter   = -1;
        std::string     alg_filter;
        int             score       = -1;

        std::stringstream es(entry);
        std::string       tok;
        std::vector<std::string> toks;
        while (std::getline(es, tok, ':')) {
            toks.push_back(tok);
        }
        if (toks.empty()) {
            continue;
        }
        /* last token is always the score */
        std::string sc = toks.back();
        toks.pop_back();
        score = (sc == "inf") ? INT32_MAX : (int)strtol(sc.c_str(), nullptr, 0);
        for (auto &t : toks) {
            ucc_coll_type_t ct;
            if (t.empty()) {
                continue;
            }
            if (t[0] == '@') {
                alg_filter = t.substr(1);
            } else if (coll_type_from_name(t, &ct) == UCC_OK) {
                ct_filter = ct;
                have_ct   = true;
            } else if (t == "host" || t == "cuda" || t == "rocm") {
                mt_filter = (t == "host") ? UCC_MEMORY_TYPE_HOST
                                          : UCC_MEMORY_TYPE_CUDA;
            } else if (t.find('-') != std::string::npos) {
                size_t dash = t.find('-');
                r_start     = parse_size(t.substr(0, dash), 0);
                std::string e2 = t.substr(dash + 1);
                r_end = (e2 == "inf") ? SIZE_MAX : parse_size(e2, SIZE_MAX);
                have_range = true;
            } else {
                ucc_warn("unrecognized tuning token '%s'", t.c_str());
                return UCC_ERR_INVALID_PARAM;
            }
        }
        for (int ci = 0; ci < UCC_COLL_TYPE_NUM; ci++) {
            if (have_ct && ci != coll_type_index(ct_filter)) {
                continue;
            }
            for (int mi = 0; mi < UCC_MEMORY_TYPE_LAST; mi++) {
                if (mt_filter >= 0 && mi != mt_filter) {
                    continue;
                }
                /* ranged updates SPLIT partially-overlapping entries:
                 * the score changes only inside [r_start, r_end); the
                 * outside pieces keep their old score (reference
                 * ucc_coll_score_update_from_str semantics) */
                std::vector<ScoreRange> keep;
                for (auto &r : ranges[ci][mi]) {
                    if (!alg_filter.empty() && r.alg_name != alg_filter) {
                        continue;
                    }
                    if (!have_range) {
                        r.score = score;
                        continue;
                    }
                    size_t lo = r_start > r.start ? r_start : r.start;
                    size_t hi = r_end < r.end ? r_end : r.end;
                    if (lo >= hi) {
                        continue; /* disjoint: untouched */
                    }
                    if (lo > r.start) {
                        ScoreRange pre = r;
                        pre.end        = lo;
                        keep.push_back(pre);
                    }
                    if (hi < r.end) {
                        ScoreRange post = r;
                        post.start      = hi;
                        keep.push_back(post);
                    }
                    r.start = lo;
                    r.end   = hi;
                    r.score = score;
                }
                for (auto &k : keep) {
                    ranges[ci][mi].push_back(k);
                }
                std::stable_sort(ranges[ci][mi].begin(), ranges[ci][mi].end(),
                                 [](const ScoreRange &a, const ScoreRange &b) {
                                     return a.score > b.score;
                                 });
            }
        }
    }
    return UCC_OK;
}

std::string ScoreMap::to_string() const
{
    std::stringstream out;
    for (int ci = 0; ci < UCC_COLL_TYPE_NUM; ci++) {
        for (int mi = 0; mi < UCC_MEMORY_TYPE_LAST; mi++) {
            for (const auto &r : ranges[ci][mi]) {
                out << k_coll_names[ci] << ":"
                    << mem_type_name((ucc_memory_type_t)mi) << ":"
                    << r.start << "-";
                if (r.end == SIZE_MAX) {
                    out << "inf";
                } else {
                    out << r.end;
                }
                out << ":@" << r.tl_name << "/" << r.alg_name << ":"
                    << r.score << "\n";
            }
        }
    }
    return out.str();
}

/* -------- msgsize / memtype helpers (reference: utils/ucc_coll_utils.c) */
static size_t counts_total(const ucc_coll_args_t &args, const ucc_count_t *c,
                           uint32_t size)
{
    size_t total = 0;
    if (args.flags & UCC_COLL_ARGS_FLAG_COUNT_64BIT) {
        for (uint32_t i = 0; i < size; i++) {
            total += ((const uint64_t *)c)[i];
        }
    } else {
        for (uint32_t i = 0; i < size; i++) {
            total += ((const uint32_t *)c)[i];
        }
    }
    return total;
}

size_t coll_args_msgsize(const ucc_coll_args_t &args, uint32_t rank,
                         uint32_t size)
{
    switch (args.coll_type) {
    case UCC_COLL_TYPE_BARRIER:
    case UCC_COLL_TYPE_FANIN:
    case UCC_COLL_TYPE_FANOUT: return 0;
    case UCC_COLL_TYPE_ALLREDUCE:
    case UCC_COLL_TYPE_REDUCE:
    case UCC_COLL_TYPE_BCAST:
        return args.src.info.count * ucc_dt_size(args.src.info.datatype);
    case UCC_COLL_TYPE_REDUCE_SCATTER:
    case UCC_COLL_TYPE_ALLTOALL:
    case UCC_COLL_TYPE_ALLGATHER:
        /* total bytes moved through dst */
        if (args.flags & UCC_COLL_ARGS_FLAG_IN_PLACE &&
            args.coll_type != UCC_COLL_TYPE_ALLGATHER) {
            return args.src.info.count * ucc_dt_size(args.src.info.datatype);
        }
        return args.dst.info.count * ucc_dt_size(args.dst.info.datatype);
    case UCC_COLL_TYPE_GATHER:
        /* selection MUST be rank-invariant (all ranks must pick the
         * same algorithm): root passes the gathered TOTAL, leaves
         * their equal block — scale the leaf view by team size */
        if (rank == (uint32_t)args.root) {
            return args.dst.info.count *
                   ucc_dt_size(args.dst.info.datatype);
        }
        return args.src.info.count *
               ucc_dt_size(args.src.info.datatype) * size;
    case UCC_COLL_TYPE_SCATTER:
        if (rank == (uint32_t)args.root) {
            return args.src.info.count *
                   ucc_dt_size(args.src.info.datatype);
        }
        return args.dst.info.count *
               ucc_dt_size(args.dst.info.datatype) * size;
    case UCC_COLL_TYPE_ALLGATHERV:
        return counts_total(args, args.dst.info_v.counts, size) *
               ucc_dt_size(args.dst.info_v.datatype);
    case UCC_COLL_TYPE_GATHERV:
        /* v-args are significant at the root only (UCC semantics):
         * non-roots size by their own contiguous src. NOTE: this is
         * rank-divergent for ragged tables; keep every gatherv
         * algorithm registered over the FULL size range (currently
         * true: single linear task per transport). */
        if (rank != (uint32_t)args.root) {
            return args.src.info.count *
                   ucc_dt_size(args.src.info.datatype);
        }
        return counts_total(args, args.dst.info_v.counts, size) *
               ucc_dt_size(args.dst.info_v.datatype);
    case UCC_COLL_TYPE_ALLTOALLV:
        /* the per-rank count ROWS differ across ranks, so any local
         * byte sum is rank-DIVERGENT — ranks in different score bands
         * would run different algorithms against each other and
         * deadlock (caught by the randomized-sequence stress under
         * the hier composition). Selection therefore uses a constant;
         * size-adaptive behavior lives INSIDE the algorithms (hybrid
         * per-pair threshold, cdna4 exchanged global max/sum). */
        return 0;
    case UCC_COLL_TYPE_SCATTERV:
        if (rank != (uint32_t)args.root) {
            return args.dst.info.count *
                   ucc_dt_size(args.dst.info.datatype);
        }
        return counts_total(args, args.src.info_v.counts, size) *
               ucc_dt_size(args.src.info_v.datatype);
    case UCC_COLL_TYPE_REDUCE_SCATTERV:
        return counts_total(args, args.dst.info_v.counts, size) *
               ucc_dt_size(args.dst.info_v.datatype);
    default: return 0;
    }
}

ucc_memory_type_t coll_args_mem_type(const ucc_coll_args_t &args,
                                     uint32_t rank)
{
    ucc_memory_type_t mt;
    bool              v_dst =
        args.coll_type == UCC_COLL_TYPE_ALLGATHERV ||
        args.coll_type == UCC_COLL_TYPE_GATHERV ||
        args.coll_type == UCC_COLL_TYPE_ALLTOALLV ||
        args.coll_type == UCC_COLL_TYPE_REDUCE_SCATTERV;
    if (args.coll_type == UCC_COLL_TYPE_BARRIER ||
        args.coll_type == UCC_COLL_TYPE_FANIN ||
        args.coll_type == UCC_COLL_TYPE_FANOUT) {
        return UCC_MEMORY_TYPE_HOST;
    }
    if (args.coll_type == UCC_COLL_TYPE_SCATTERV) {
        /* scatterv's SOURCE uses the v-union member — at the root;
         * non-roots only have a plain dst */
        mt = rank == (uint32_t)args.root ? args.src.info_v.mem_type
                                         : args.dst.info.mem_type;
    } else if (args.coll_type == UCC_COLL_TYPE_BCAST) {
        mt = args.src.info.mem_type;
    } else if (args.coll_type == UCC_COLL_TYPE_GATHERV &&
               rank != (uint32_t)args.root) {
        mt = args.src.info.mem_type;
    } else {
        mt = v_dst ? args.dst.info_v.mem_type : args.dst.info.mem_type;
    }
    /* the device TLs serve one memory space; fold the aliases */
    if (mt == UCC_MEMORY_TYPE_ROCM) {
        mt = UCC_MEMORY_TYPE_CUDA;
    }
    if (mt == UCC_MEMORY_TYPE_ROCM_MANAGED) {
        mt = UCC_MEMORY_TYPE_CUDA_MANAGED;
    }
    return mt;
}

} // namespace ucc
