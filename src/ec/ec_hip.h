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
constexpr int kFlagsBytes     = 8192;               /* flags alloc size      */

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

} // namespace ec_hip
} // namespace ucc

#endif
