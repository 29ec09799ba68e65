/* Device execution component: gfx950 HIP kernels for n-source reduction,
 * gather-copy, and the fused single-kernel allreduce over xGMI peer
 * pointers. Parity targets: reference components/ec/rocm kernels
 * (ec_rocm_reduce.cu dtype x op family, executor copy_multi) and the NVLS
 * allreduce kernel role (tl/cuda/kernels/allreduce_kernel.cu) — re-designed
 * for CDNA4: 64-wide wavefronts, 16B vector lanes, system-scope atomic
 * arrival flags on fine-grained memory instead of multimem. */
#ifndef UCC_AMD_EC_HIP_H_
#define UCC_AMD_EC_HIP_H_

#include <cstddef>
#include <cstdint>
#include "../api/ucc.h"

struct ihipStream_t;
typedef struct ihipStream_t *hipStream_t;

namespace ucc {
namespace ec_hip {

constexpr int kMaxSrcs  = 8;
constexpr int kMaxRanks = 8;

/* dst[i] = alpha * op(srcs[0][i] .. srcs[n-1][i]); dt/op per ucc enums. */
struct ReduceArgs {
    void        *dst;
    const void  *srcs[kMaxSrcs];
    int          n_srcs;
    uint64_t     count;
    ucc_datatype_t     dt;
    ucc_reduction_op_t op;
    float        alpha;
};
ucc_status_t reduce(const ReduceArgs &a, hipStream_t stream);

/* Gather-copy: dst_base+off[i] <- srcs[i], len[i] bytes, i in [0,n). */
struct GatherArgs {
    void       *dst_base;
    const void *srcs[kMaxSrcs];
    uint64_t    offs[kMaxSrcs];
    uint64_t    lens[kMaxSrcs];
    int         n;
};
ucc_status_t gather_copy(const GatherArgs &a, hipStream_t stream);

/* Fused small-message allreduce: one kernel per rank that stages src into
 * its scratch, signals arrival on every peer's flag array (system-scope),
 * waits for all arrivals, then reduces all peers' scratch into dst.
 * flags layout: u64[slot * kMaxRanks + src_rank], monotonically increasing
 * seq values (never reset). */
struct FusedArgs {
    const void *src;
    void       *dst;
    uint64_t    count;
    void       *my_scratch;
    const void *peer_scratch[kMaxRanks];
    uint64_t   *local_flags;
    uint64_t   *peer_flags[kMaxRanks];
    int         rank, nranks, slot;
    uint64_t    seq;
    uint64_t    stage_target; /* cumulative block arrivals for this slot
                                 including this launch (host-tracked)   */
    ucc_datatype_t     dt;
    ucc_reduction_op_t op;
    float       alpha;
    uint64_t   *error_word; /* local fine-grained word set on spin timeout */
    int         nblocks;    /* workgroups; each handles a count slice     */
    uint64_t    spin_limit; /* 0 = default kSpinLimit                     */
    /* host-pinned completion flag: the LAST block (cumulative arrival
     * counter at flags[kFusedDoneBase + slot] reaching done_target)
     * release-stores done_seq here — the host polls plain memory
     * instead of hipEventRecord+Query (µs-class latency saving on the
     * small-message path). */
    uint64_t   *done_host;
    uint64_t    done_seq;
    uint64_t    done_target;
};
ucc_status_t fused_allreduce(const FusedArgs &a, hipStream_t stream);

/* Graph-capturable fused allreduce: identical to fused_allreduce but the
 * per-iteration sequence number is derived ON DEVICE from per-block launch
 * counters (flags u64[kGraphCntBase + slot*kMaxGraphBlocks + block]), so
 * the SAME launch can be captured once into a hipGraph and replayed: every
 * replay is a new collective iteration. Requires a slot dedicated to one
 * persistent request (all ranks replay in lockstep). */
constexpr int kMaxGraphBlocks = 64;
constexpr int kStageCntBase   = 8 * kMaxRanks;      /* u64 idx of stage cnts */
constexpr int kGraphCntBase   = 128;                /* u64 idx of blk cnts   */
constexpr int kFlagsBytes     = 131072;             /* flags alloc size
                                    (>= 8*(kGatedMirrorBase + 3*8*2*8)) */

struct GraphFusedArgs {
    const void *src;
    void       *dst;
    uint64_t    count;
    void       *my_scratch;
    const void *peer_scratch[kMaxRanks];
    uint64_t   *local_flags;
    uint64_t   *peer_flags[kMaxRanks];
    int         rank, nranks, slot;
    ucc_datatype_t     dt;
    ucc_reduction_op_t op;
    float       alpha;
    uint64_t   *error_word;
    int         nblocks;       /* <= kMaxGraphBlocks */
    uint64_t    parity_stride; /* bytes between parity-0 and parity-1
                                  staging areas: iteration i stages into
                                  base + (i&1)*parity_stride so a peer's
                                  replay i+1 never overwrites data still
                                  being read for replay i */
};
ucc_status_t fused_allreduce_graph(const GraphFusedArgs &a,
                                   hipStream_t stream);

bool dt_supported(ucc_datatype_t dt);
bool op_supported(ucc_datatype_t dt, ucc_reduction_op_t op);

/* ------------------------------------------------------------------------
 * Device-gated staged pipeline (large-message allreduce fast path).
 *
 * Replaces the host-polled shm gating of the staged-linear algorithm with
 * kernel-prologue spins on per-(slot,parity,phase) cumulative block
 * counters in each rank's fine-grained flags buffer: the host enqueues
 * stage/reduce/gather kernels for EVERY fragment up front (two streams)
 * and waits for one trailing event — no host round trips inside the
 * collective. Targets are host-computed cumulative launch counts
 * (collectives post in the same order on every rank), x kGatedBlocks
 * block-increments per launch.
 *
 * Counter layout in the flags buffer (u64 index):
 *   kGatedCntBase + (phase * kGatedSlots + slot) * 2 + parity
 *   phase: 0=stage 1=reduce 2=gather; slot < kGatedSlots; parity = frag&1.
 */
constexpr int kGatedBlocks    = 64;  /* default grid (UCC_TL_CDNA4_GATED_BLOCKS) */
constexpr int kGatedMaxBlocks = 512; /* flag-layout bound for the knob   */
constexpr int kGatedSlots     = 8;
constexpr int kGatedCntBase   = 704;  /* after graph counters (128..639) */
/* per-(slot,parity,block) launch counters for the graph-replayable gated
 * mode: u64 idx kGatedGraphBase + ((slot*2)+parity)*kGatedMaxBlocks + block
 * -> 1024..9215 */
constexpr int kGatedGraphBase = 1024;
/* push-model mirrors: rank r's phase counter value, WRITTEN BY RANK r
 * over xGMI into every peer's flag buffer when its last block arrives,
 * so waiters poll LOCAL memory only (B x n remote pollers would steal
 * link bandwidth - the fused kernel already works this way).
 * u64 idx kGatedMirrorBase + ((phase*kGatedSlots+slot)*2+parity)*kMaxRanks
 *         + src_rank -> 9216..9599 */
constexpr int kGatedMirrorBase = 9216;
/* fused-kernel completion arrival counters: u64 idx
 * kFusedDoneBase + slot -> 9600..9607 */
constexpr int kFusedDoneBase = 9600;

/* default device spin bound (iterations of the bounded wait loops);
 * host-side team setup scales it by the xGMI hop count of the farthest
 * peer before passing it down via GatedArgs/FusedArgs.spin_limit */
constexpr uint64_t kDefaultSpinLimit = 200u * 1000u * 1000u;

struct GatedArgs {
    /* data */
    const void *src;      /* stage: user src frag; others unused        */
    void       *dst;      /* gather: user dst frag; others unused       */
    void       *my_in;    /* my in[slot][parity] staging area           */
    void       *my_out;   /* my out[slot][parity] reduced-slice area    */
    const void *peer_in[kMaxRanks];  /* peers' in areas (xGMI)          */
    const void *peer_out[kMaxRanks]; /* peers' out areas (xGMI)         */
    uint64_t   *local_flags;
    uint64_t   *peer_flags[kMaxRanks];
    uint64_t   *error_word;
    uint64_t    len;     /* fragment bytes                              */
    uint64_t    sl_b, sl_e; /* my reduce slice [begin,end) bytes        */
    /* per-rank slice table for gather (byte offsets into out areas)    */
    uint64_t    slice_b[kMaxRanks], slice_e[kMaxRanks];
    int         rank, nranks, slot, parity;
    ucc_datatype_t     dt;
    ucc_reduction_op_t op;
    float       alpha;
    /* cumulative block-count targets (host-tracked). Counters are shared
     * by every gated collective type on a (slot,parity), so area-reuse
     * waits cover BOTH possible consumer phases:
     *   stage  waits reduce_cnt >= t_sw_reduce AND
     *                gather_cnt >= t_sw_gather   (in-area reuse)
     *   reduce waits stage_cnt  >= t_stage AND
     *                gather_cnt >= t_prev_gather (out-area reuse)
     *   gather waits phase gw_phase >= t_gather_wait
     *                (1=reduce for allreduce, 0=stage for ag/alltoall) */
    uint64_t    t_sw_reduce, t_sw_gather, t_prev_gather, t_stage,
                t_gather_wait;
    /* post-launch cumulative counter values for THIS launch's signal
     * phase (stage=0/reduce=1/gather=2): the block whose increment
     * reaches the value is the last arriver and pushes the mirrors.
     * Counters accumulate BLOCK counts, so collectives with different
     * grid sizes compose on the same (slot,parity). Derive mode
     * computes these on device as u*B instead. */
    uint64_t    t_sig_stage, t_sig_reduce, t_sig_gather;
    int         gw_phase;
    /* Graph-replayable mode (persistent colls on a dedicated slot): when
     * derive != 0, targets are computed ON DEVICE from per-block launch
     * counters at kGatedGraphBase (one counter per (slot,parity,block),
     * incremented by EVERY gated kernel of the pattern): this block's
     * count v gives the iteration u = (v-1)/pp + 1 and the host targets
     * above are ignored — so one capture replays forever. pp = kernels
     * per (slot,parity) iteration (3 for allreduce, 2 for rs/ag/a2a);
     * has_reduce/has_gather describe the pattern so waits on phases that
     * never launch resolve to zero. */
    int         derive, pp, has_reduce, has_gather;
    /* write-based distribution (zero-copy allreduce): reduce writes its
     * slice DIRECTLY into every rank's user dst (peer_out[r] =
     * mapped dst + frag_off + my_slice_off) instead of the local out
     * area; the gather launch degenerates to a pure wait+signal. */
    int         zc_write;
    int         nblocks;    /* gated grid size (team-lifetime constant,
                               same on every rank; <= kGatedMaxBlocks;
                               0 = kGatedBlocks default) */
    int         pull_wait;  /* 1 = poll peers' counters remotely instead
                               of local mirrors (debug fallback) */
    /* host-pinned completion (set ONLY on the final kernel of a
     * host-posted collective): the last arriving block release-stores
     * done_seq after its signal push — host polls plain memory instead
     * of a trailing hipEvent. */
    uint64_t   *done_host;
    uint64_t    done_seq;
    uint64_t    spin_limit; /* 0 = default kSpinLimit */
    /* per-dest cell staging (alltoall): my_in[c_dst_off[k]] <-
     * src[c_src_off[k]], c_len[k] bytes; 0 = contiguous stage of len   */
    int         n_cells;
    uint64_t    c_src_off[kMaxRanks], c_dst_off[kMaxRanks],
                c_len[kMaxRanks];
};

ucc_status_t staged_stage(const GatedArgs &a, hipStream_t s);
ucc_status_t staged_reduce(const GatedArgs &a, hipStream_t s);
ucc_status_t staged_gather(const GatedArgs &a, hipStream_t s);
/* copy-engine gating (SDMA data path): pure wait on gw_phase/t_gather_wait
 * (1 block), and signal-phase-2-then-wait-team (full gated grid). */
ucc_status_t gated_wait_only(const GatedArgs &a, hipStream_t s);
ucc_status_t gated_done(const GatedArgs &a, hipStream_t s);

} // namespace ec_hip
} // namespace ucc

#endif
