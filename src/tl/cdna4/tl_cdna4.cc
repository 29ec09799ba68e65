/* TL "cdna4": the xGMI device transport for up to 8 fully-connected
 * MI355X GPUs (one process per GPU, or the in-process multi-rank jig on a
 * single GPU).
 *
 * Reference parity: replaces tl/cuda (tl_cuda_team.c scratch+IPC exchange,
 * tl_cuda_coll.h slot protocol, *_linear.c algorithms) and the NVLS
 * allreduce (tl/cuda/kernels/allreduce_kernel.cu), re-designed for CDNA4:
 *
 *  - control plane: host POSIX-shm slot segment (slot_seg.h) with monotonic
 *    step counters — same protocol as tl/shm.
 *  - data plane: per-rank device scratch (hipMalloc, HIP-IPC exported) laid
 *    out as [slot][parity][in|out] areas; peers read each other's areas
 *    directly over xGMI from gfx950 kernels (ec_hip reduce/gather), so a
 *    staged linear collective touches all 7 links simultaneously:
 *    reduce-scatter phase + allgather phase each move S/8 per link =>
 *    aggregate bus bandwidth approaches 7 x 153 GB/s per GPU.
 *  - small messages: ONE fused kernel per rank (ec_hip::fused_allreduce)
 *    that stages, signals arrival on every peer's fine-grained flag word
 *    (system-scope atomics), spins bounded, reduces — no host round trips
 *    between ranks, single kernel-launch latency.
 *
 * Algorithms here (v1): allreduce (fused small / staged-linear large),
 * allgather(v), reduce_scatter(v), bcast, reduce. Host-memory colls and
 * everything else fall back per the score map.
 */
#include <atomic>
#include <map>
#include <mutex>

#include <hip/hip_runtime.h>

#include "../../core/core.h"
#include "../../ec/ec_hip.h"
#include "../../topo/topo.h"
#include "../../mc/mc.h"
#include "../shm/slot_seg.h"

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

using ec_hip::kMaxRanks;

/* v-coll count/displacement accessors honoring the 64-bit flags
 * (reference: ucc_coll_utils count conventions). */
static inline size_t coll_count_at(const ucc_coll_args_t &a, const void *c,
                                   uint32_t i)
{
    return (a.flags & UCC_COLL_ARGS_FLAG_COUNT_64BIT)
               ? (size_t)((const uint64_t *)c)[i]
               : (size_t)((const uint32_t *)c)[i];
}
static inline size_t coll_disp_at(const ucc_coll_args_t &a, const void *d,
                                  uint32_t i)
{
    return (a.flags & UCC_COLL_ARGS_FLAG_DISPLACEMENTS_64BIT)
               ? (size_t)((const uint64_t *)d)[i]
               : (size_t)((const uint32_t *)d)[i];
}

/* observability: zc exchanges that consumed user-registered mem_map
 * handles (honored src_memh/dst_memh) — regression hook for tests */
static std::atomic<uint64_t> g_memh_uses{0};
extern "C" uint64_t ucc_amd_cdna4_memh_uses()
{
    return g_memh_uses.load(std::memory_order_relaxed);
}

#define HIPCHK(expr)                                                         \
    do {                                                                     \
        hipError_t _e = (expr);                                              \
        if (_e != hipSuccess) {                                              \
            ucc_error("%s failed: %s", #expr, hipGetErrorString(_e));        \
            return UCC_ERR_NO_RESOURCE;                                      \
        }                                                                    \
    } while (0)

struct Cdna4Cfg {
    uint64_t spin_limit; /* device spin bound; 0 = kernel default */
    uint32_t nslots;
    uint32_t npers;     /* dedicated slots for persistent triggered
                           (graph-captured) collectives              */
    size_t   chunk;     /* staging fragment bytes                     */
    size_t   fused_max; /* msg sizes <= this take the fused kernel    */
    int      gated_blocks; /* gated kernel grid size (team constant)  */
    bool     ce_alltoall;  /* SDMA copy-engine alltoall data path     */
    size_t   ce_alltoall_min; /* min total msg bytes for the CE path  */
};

class Cdna4Tl;

class Cdna4TlContext final : public TlContext {
  public:
    Cdna4TlContext(Context *ctx, Tl *tl, int dev) : TlContext(ctx), tl_(tl),
                                                    dev_(dev)
    {
        HIPWARN(hipStreamCreateWithFlags(&comp_, hipStreamNonBlocking));
        HIPWARN(hipStreamCreateWithFlags(&copy_, hipStreamNonBlocking));
    }
    ~Cdna4TlContext() override
    {
        if (comp_) {
            HIPWARN(hipStreamDestroy(comp_));
        }
        if (copy_) {
            HIPWARN(hipStreamDestroy(copy_));
        }
        for (auto &s : ce_) {
            if (s) {
                HIPWARN(hipStreamDestroy(s));
            }
        }
    }
    Tl *iface() override;

    /* copy-engine streams: concurrent hipMemcpyAsync lanes so per-peer
     * SDMA transfers overlap (one stream would serialize the blits) */
    static constexpr int kNumCe = 4;
    hipStream_t ce(int i)
    {
        i &= kNumCe - 1;
        if (!ce_[i] &&
            hipStreamCreateWithFlags(&ce_[i], hipStreamNonBlocking) !=
                hipSuccess) {
            ce_[i] = nullptr;
        }
        return ce_[i];
    }

    Tl         *tl_;
    int         dev_;
    hipStream_t comp_ = nullptr, copy_ = nullptr;
    hipStream_t ce_[kNumCe] = {};
};

struct PeerRes {
    uint8_t  *scratch = nullptr;
    uint64_t *flags   = nullptr;
    bool      ipc_s = false, ipc_f = false;
};

struct ExchgBlob {
    int32_t           pid;
    int32_t           device;
    uint64_t          scratch_ptr;
    uint64_t          flags_ptr;
    hipIpcMemHandle_t sh;
    hipIpcMemHandle_t fh;
};

class Cdna4TlTeam final : public TlTeam {
  public:
    Cdna4TlTeam(TlContext *tlc, Team *team, const Cdna4Cfg &cfg)
        : TlTeam(tlc, team), cfg_(cfg)
    {
        char buf[96];
        snprintf(buf, sizeof(buf), "/uccamd-c4-%016llx-%u",
                 (unsigned long long)team->team_uid,
                 (unsigned)(team->team_uid >> 48));
        seg_name_ = buf;
        init_st_  = local_init();
        if (init_st_ == UCC_OK && team->rank == 0) {
            init_st_ = seg_.create(seg_name_, team->size, cfg_.nslots, 0);
        }
    }

    /* team-lifetime IPC import cache: user-buffer handles opened for
     * zero-copy stay mapped until the team dies, so re-created
     * persistent colls on the same tensors (e.g. a rebuilt DDP bucket)
     * skip the hipIpcOpenMemHandle on every rank (reference mem_map /
     * rcache role for the xGMI path).
     *
     * Lifecycle (reference tl_cuda_cache.c:113-140 invalidate role):
     * keyed by (exporter pid, exporter base VA). A hit re-validates the
     * HANDLE BYTES — if the exporter freed and a new allocation landed
     * on the same VA, the handle differs and the stale mapping is
     * closed and re-opened. Conversely a NEW (pid,va) whose handle
     * bytes match a cached entry at a DIFFERENT va means the driver
     * recycled the handle identity after a free: the old entry is
     * stale and must be closed BEFORE the open (an open of the same
     * handle can return the old refcounted mapping). */
    void *ipc_open_cached(int32_t pid, uint64_t remote_base,
                          const hipIpcMemHandle_t &h)
    {
        std::string hb((const char *)&h, sizeof(h));
        auto        key = std::make_pair(pid, remote_base);
        std::lock_guard<std::mutex> g(ipc_mu_);
        auto it = ipc_cache_.find(key);
        if (it != ipc_cache_.end()) {
            if (it->second.handle == hb) {
                return it->second.mapped;
            }
            HIPWARN(hipIpcCloseMemHandle(it->second.mapped));
            ipc_cache_.erase(it);
        }
        for (auto jt = ipc_cache_.begin(); jt != ipc_cache_.end();) {
            if (jt->second.handle == hb) {
                HIPWARN(hipIpcCloseMemHandle(jt->second.mapped));
                jt = ipc_cache_.erase(jt);
            } else {
                ++jt;
            }
        }
        void *m = nullptr;
        if (hipIpcOpenMemHandle(&m, h, hipIpcMemLazyEnablePeerAccess) !=
            hipSuccess) {
            return nullptr;
        }
        ipc_cache_[key] = IpcEnt{hb, m};
        return m;
    }
    struct IpcEnt {
        std::string handle;
        void       *mapped;
    };
    std::map<std::pair<int32_t, uint64_t>, IpcEnt> ipc_cache_;
    std::mutex                    ipc_mu_;

    ~Cdna4TlTeam() override
    {
        for (auto &kv : ipc_cache_) {
            HIPWARN(hipIpcCloseMemHandle(kv.second.mapped));
        }
        for (uint32_t r = 0; r < (uint32_t)peers_.size(); r++) {
            if (peers_[r].ipc_s) {
                HIPWARN(hipIpcCloseMemHandle(peers_[r].scratch));
            }
            if (peers_[r].ipc_f) {
                HIPWARN(hipIpcCloseMemHandle(peers_[r].flags));
            }
        }
        if (scratch_) {
            HIPWARN(hipFree(scratch_));
        }
        if (flags_) {
            HIPWARN(hipFree(flags_));
        }
        if (err_host_) {
            HIPWARN(hipHostFree(err_host_));
        }
    }

    ucc_status_t local_init()
    {
        /* scratch: [slot][parity][in|out] areas of chunk bytes each;
         * slots [0, nslots) rotate for host-driven colls, slots
         * [nslots, nslots+npers) are pinned to persistent triggered
         * (graph-replayable) requests */
        scratch_bytes_ =
            (size_t)(cfg_.nslots + cfg_.npers) * 2 * 2 * cfg_.chunk;
        HIPCHK(hipMalloc((void **)&scratch_, scratch_bytes_));
        pslot_used_.assign(cfg_.npers, false);
        pslot_kind_.assign(cfg_.npers, 0);
        /* flags: fine-grained for cross-GPU system-scope atomics */
        hipError_t e =
            hipExtMallocWithFlags((void **)&flags_, ec_hip::kFlagsBytes,
                                  hipDeviceMallocFinegrained);
        if (e != hipSuccess) {
            ucc_warn("finegrained alloc failed (%s), using hipMalloc",
                     hipGetErrorString(e));
            HIPCHK(hipMalloc((void **)&flags_, ec_hip::kFlagsBytes));
        }
        HIPCHK(hipMemset(flags_, 0, ec_hip::kFlagsBytes));
        HIPCHK(hipHostMalloc((void **)&err_host_, 192,
                             hipHostMallocDefault));
        memset((void *)err_host_, 0, 192);
        done_host_  = err_host_ + 8;  /* [8..15]: per-slot fused flags */
        gdone_host_ = err_host_ + 16; /* [16..23]: per-slot gated flags */
        gdone_seq_.assign(ec_hip::kGatedSlots, 0);
        HIPCHK(hipDeviceSynchronize());
        return UCC_OK;
    }

    size_t exchg_size() override { return sizeof(ExchgBlob); }

    void exchg_pack(void *buf) override
    {
        ExchgBlob b{};
        b.pid         = (int32_t)getpid();
        b.device      = ((Cdna4TlContext *)tlc_)->dev_;
        b.scratch_ptr = (uint64_t)(uintptr_t)scratch_;
        b.flags_ptr   = (uint64_t)(uintptr_t)flags_;
        if (init_st_ == UCC_OK) {
            HIPWARN(hipIpcGetMemHandle(&b.sh, scratch_));
            HIPWARN(hipIpcGetMemHandle(&b.fh, flags_));
        }
        memcpy(buf, &b, sizeof(b));
    }

    ucc_status_t exchg_unpack(const void *all, size_t stride) override
    {
        if (init_st_ != UCC_OK) {
            return init_st_;
        }
        const uint32_t n  = team_->size;
        const int32_t  me = (int32_t)getpid();
        peers_.resize(n);
        dev_map_.resize(n);
        /* Topology gate (reference tl_cuda_team_topo.c:56-100 role):
         * every pair must be xGMI/peer-access reachable or same-device,
         * or this TL refuses the team and the score fallback chain
         * (rccl, shm staging) takes over instead of hanging on an
         * unreachable IPC read. Spin bounds scale with the worst hop
         * count so remote-peer latency is not misread as a timeout. */
        {
            int mydev = ((Cdna4TlContext *)tlc_)->dev_;
            const topo::GpuLinks &gl = topo::gpu_links();
            max_hops_ = 1;
            for (uint32_t r = 0; r < n; r++) {
                ExchgBlob b;
                memcpy(&b, (const uint8_t *)all + r * stride, sizeof(b));
                dev_map_[r] = b.device;
                if (b.device == mydev || r == team_->rank) {
                    continue;
                }
                if (b.device < gl.ndev && mydev < gl.ndev &&
                    b.device >= 0) {
                    if (!gl.peer[mydev][b.device]) {
                        ucc_warn("cdna4: no peer access dev%d->dev%d "
                                 "(rank %u); dropping cdna4 team "
                                 "(fallback TLs take over)",
                                 mydev, b.device, r);
                        return UCC_ERR_NOT_SUPPORTED;
                    }
                    int h = gl.hops[mydev][b.device];
                    if (h > (int)max_hops_) {
                        max_hops_ = (uint32_t)h;
                    }
                }
            }
        }
        for (uint32_t r = 0; r < n; r++) {
            ExchgBlob b;
            memcpy(&b, (const uint8_t *)all + r * stride, sizeof(b));
            if (r == team_->rank) {
                peers_[r].scratch = scratch_;
                peers_[r].flags   = flags_;
            } else if (b.pid == me) {
                /* in-process jig: same address space */
                peers_[r].scratch = (uint8_t *)(uintptr_t)b.scratch_ptr;
                peers_[r].flags   = (uint64_t *)(uintptr_t)b.flags_ptr;
            } else {
                void *p = nullptr;
                hipError_t e = hipIpcOpenMemHandle(
                    &p, b.sh, hipIpcMemLazyEnablePeerAccess);
                if (e != hipSuccess) {
                    ucc_warn("IPC open scratch of rank %u failed: %s", r,
                             hipGetErrorString(e));
                    return UCC_ERR_NO_RESOURCE;
                }
                peers_[r].scratch = (uint8_t *)p;
                peers_[r].ipc_s   = true;
                e = hipIpcOpenMemHandle(&p, b.fh,
                                        hipIpcMemLazyEnablePeerAccess);
                if (e != hipSuccess) {
                    ucc_warn("IPC open flags of rank %u failed: %s", r,
                             hipGetErrorString(e));
                    return UCC_ERR_NO_RESOURCE;
                }
                peers_[r].flags = (uint64_t *)p;
                peers_[r].ipc_f = true;
            }
        }
        return UCC_OK;
    }

    ucc_status_t create_test() override
    {
        if (init_st_ != UCC_OK) {
            return init_st_;
        }
        if (team_->rank == 0) {
            return UCC_OK;
        }
        if (!attached_) {
            ucc_status_t st =
                seg_.attach(seg_name_, team_->size, cfg_.nslots, 0);
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

    void get_scores(Team *team, ScoreMap &map) override;

    uint8_t *area(uint32_t rank, uint32_t slot, uint32_t parity, int which)
    {
        return peers_[rank].scratch +
               (((size_t)slot * 2 + parity) * 2 + which) * cfg_.chunk;
    }

    /* Persistent-slot allocator. Deterministic across ranks as long as
     * persistent triggered requests are created/finalized in the same
     * order on every rank (collective semantics): lowest matching index.
     * A slot's device-side graph counters are NEVER reset, so a slot may
     * only be reused by a request with the SAME kernel pattern (kind):
     * counters then continue monotonically and stay consistent. */
    enum PslotKind { PK_NONE = 0, PK_FUSED, PK_G_AR, PK_G_RS, PK_G_AG,
                     PK_G_A2A };
    int alloc_pslot(int kind)
    {
        /* first pass: same-kind slot; second: never-used slot */
        for (uint32_t i = 0; i < cfg_.npers; i++) {
            if (!pslot_used_[i] && pslot_kind_[i] == kind) {
                pslot_used_[i] = true;
                return (int)(cfg_.nslots + i);
            }
        }
        for (uint32_t i = 0; i < cfg_.npers; i++) {
            if (!pslot_used_[i] && pslot_kind_[i] == PK_NONE) {
                pslot_used_[i] = true;
                pslot_kind_[i] = kind;
                return (int)(cfg_.nslots + i);
            }
        }
        return -1;
    }
    void free_pslot(int slot)
    {
        if (slot >= (int)cfg_.nslots) {
            pslot_used_[slot - cfg_.nslots] = false;
        }
    }
    std::vector<int> pslot_kind_;

    Cdna4Cfg    cfg_;
    ShmSeg      seg_;
    std::string seg_name_;
    std::vector<bool> pslot_used_;
    /* device-gated pipeline: cumulative kernel-launch counts per
     * (phase, slot, parity) — identical on every rank because
     * collectives post in the same order (targets = count x 32 blocks) */
    uint64_t gated_launch_[3][ec_hip::kGatedSlots][2] = {};
    std::vector<int> dev_map_;  /* rank -> hip device (topology gate) */
    uint32_t         max_hops_ = 1;
    /* spin bound scaled by the farthest peer's hop count (user
     * override wins) */
    uint64_t spin_limit() const
    {
        if (cfg_.spin_limit) {
            return cfg_.spin_limit;
        }
        return max_hops_ > 1
                   ? ec_hip::kDefaultSpinLimit * (uint64_t)max_hops_
                   : 0; /* 0 = kernel default */
    }
    uint64_t    seq_ = 0;
    std::vector<uint64_t> stage_cum_; /* per-slot fused block arrivals */
    uint8_t    *scratch_ = nullptr;
    size_t      scratch_bytes_ = 0;
    uint64_t   *flags_ = nullptr;  /* device fine-grained               */
    uint64_t   *err_host_  = nullptr; /* host-pinned, device-writable   */
    uint64_t   *done_host_ = nullptr; /* per-slot fused completion flags */
    uint64_t   *gdone_host_ = nullptr; /* per-slot gated completion flags */
    std::vector<uint64_t> done_cum_;  /* cumulative completion arrivals  */
    std::vector<uint64_t> gdone_seq_; /* per-slot gated completion seqs  */
    std::vector<PeerRes> peers_;
    ucc_status_t init_st_ = UCC_OK;
    bool         attached_ = false;
};

/* -------------------------------------------------------------- tasks  */
class Cdna4Task : public Task {
  public:
    Cdna4Task(Context *ctx, Cdna4TlTeam *tt, const ucc_coll_args_t &args)
        : Task(ctx), tt_(tt), a_(args)
    {
    }
    ~Cdna4Task() override
    {
        for (auto ev : evs_) {
            if (ev) {
                HIPWARN(hipEventDestroy(ev));
            }
        }
    }

  protected:
    hipEvent_t ev(int i)
    {
        while ((int)evs_.size() <= i) {
            evs_.push_back(nullptr);
        }
        if (!evs_[i]) {
            HIPWARN(hipEventCreateWithFlags(&evs_[i], hipEventDisableTiming));
        }
        return evs_[i];
    }

    void publish(uint64_t k)
    {
        tt_->seg_.step(slot_, me_)->store(base_ + k,
                                          std::memory_order_release);
    }
    void close_slot()
    {
        tt_->seg_.step(slot_, me_)->store(base_ + kCap,
                                          std::memory_order_release);
    }
    bool all_ge(uint64_t k)
    {
        for (uint32_t r = 0; r < n_; r++) {
            if (tt_->seg_.step(slot_, r)->load(std::memory_order_acquire) <
                base_ + k) {
                return false;
            }
        }
        return true;
    }

    void begin_use()
    {
        me_          = tt_->team_->rank;
        n_           = tt_->team_->size;
        uint64_t use = tt_->seq_++;
        slot_        = (uint32_t)(use % tt_->cfg_.nslots);
        base_        = (use / tt_->cfg_.nslots) * kCap;
        fseq_        = use / tt_->cfg_.nslots + 1;
        phase_       = 0;
        frag_        = 0;
    }

    hipStream_t comp() { return ((Cdna4TlContext *)tt_->tlc_)->comp_; }
    hipStream_t copy_s() { return ((Cdna4TlContext *)tt_->tlc_)->copy_; }

    Cdna4TlTeam    *tt_;
    ucc_coll_args_t a_;
    std::vector<hipEvent_t> evs_;
    uint32_t me_ = 0, n_ = 1, slot_ = 0;
    uint64_t base_ = 0, fseq_ = 0;
    int      phase_ = 0;
    size_t   frag_  = 0;
};

/* Fused single-kernel allreduce (small messages). */
class FusedAllreduceTask final : public Cdna4Task {
  public:
    using Cdna4Task::Cdna4Task;

    ucc_status_t post() override
    {
        begin_use();
        bytes_ = a_.dst.info.count * ucc_dt_size(a_.dst.info.datatype);
        status = UCC_INPROGRESS;
        return progress();
    }

    ucc_status_t progress() override
    {
        if (phase_ == 0) {
            if (!all_ge(0)) {
                return UCC_INPROGRESS;
            }
            ec_hip::FusedArgs fa{};
            const bool inplace = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
            fa.dst     = a_.dst.info.buffer;
            fa.src     = inplace ? fa.dst : a_.src.info.buffer;
            fa.count   = a_.dst.info.count;
            fa.dt      = a_.dst.info.datatype;
            fa.op      = a_.op == UCC_OP_AVG ? UCC_OP_SUM : a_.op;
            fa.alpha   = a_.op == UCC_OP_AVG ? 1.0f / (float)n_ : 1.0f;
            if (a_.op == UCC_OP_AVG) {
                fa.op = (ucc_reduction_op_t)12; /* SUM+alpha path */
            }
            fa.my_scratch = tt_->area(me_, slot_, 0, 0);
            for (uint32_t r = 0; r < n_; r++) {
                fa.peer_scratch[r] = tt_->area(r, slot_, 0, 0);
                fa.peer_flags[r]   = tt_->peers_[r].flags;
            }
            fa.local_flags = tt_->flags_;
            fa.rank        = (int)me_;
            fa.nranks      = (int)n_;
            fa.slot        = (int)slot_;
            fa.seq         = fseq_;
            fa.error_word  = tt_->err_host_;
            fa.spin_limit  = tt_->spin_limit();
            size_t blocks  = (bytes_ + 128 * 1024 - 1) / (128 * 1024);
            fa.nblocks     = (int)(blocks < 1 ? 1 : blocks > 16 ? 16 : blocks);
            if (tt_->stage_cum_.size() < tt_->cfg_.nslots) {
                tt_->stage_cum_.assign(tt_->cfg_.nslots, 0);
                tt_->done_cum_.assign(tt_->cfg_.nslots, 0);
            }
            tt_->stage_cum_[slot_] += (uint64_t)fa.nblocks;
            fa.stage_target = tt_->stage_cum_[slot_];
            /* host-pinned completion flag instead of event
             * record+query (µs-class small-message latency saving) */
            tt_->done_cum_[slot_] += (uint64_t)fa.nblocks;
            fa.done_target = tt_->done_cum_[slot_];
            fa.done_seq    = fseq_;
            fa.done_host   = tt_->done_host_ + slot_;
            ucc_status_t st = ec_hip::fused_allreduce(fa, comp());
            if (st != UCC_OK) {
                return st;
            }
            phase_ = 1;
        }
        if (phase_ == 1) {
            if (*tt_->err_host_ != 0) {
                ucc_error("fused allreduce timed out waiting for peers");
                close_slot();
                return UCC_ERR_TIMED_OUT;
            }
            if (__atomic_load_n(tt_->done_host_ + slot_,
                                __ATOMIC_ACQUIRE) < fseq_) {
                return UCC_INPROGRESS;
            }
            close_slot();
            return UCC_OK;
        }
        return UCC_INPROGRESS;
    }

    /* Stream-triggered (and hipGraph-capturable) post: launch ONE fused
     * kernel with a device-derived iteration number onto the EE stream.
     * Completion is stream-ordered (request status = OK on return); the
     * same capture replays as a fresh collective iteration each time.
     * Requires a per-team dedicated slot (all ranks replay in lockstep).
     * Reference parity: ucc_triggered_post (core/ucc_coll.c:510-659) +
     * persistent executor-in-stream design, collapsed to one kernel. */
    ucc_status_t triggered_post(void *ee_stream) override
    {
        size_t bytes =
            a_.dst.info.count * ucc_dt_size(a_.dst.info.datatype);
        if (bytes > tt_->cfg_.chunk) {
            return UCC_ERR_NOT_SUPPORTED;
        }
        if (pslot_ < 0) {
            pslot_ = tt_->alloc_pslot(Cdna4TlTeam::PK_FUSED);
            if (pslot_ < 0) {
                ucc_error("no free persistent slot (TL_CDNA4 "
                          "PERSISTENT_SLOTS exhausted or pattern "
                          "mismatch)");
                return UCC_ERR_NO_RESOURCE;
            }
            me_ = tt_->team_->rank;
            n_  = tt_->team_->size;
        }
        ec_hip::GraphFusedArgs ga{};
        const bool inplace = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        ga.dst     = a_.dst.info.buffer;
        ga.src     = inplace ? ga.dst : a_.src.info.buffer;
        ga.count   = a_.dst.info.count;
        ga.dt      = a_.dst.info.datatype;
        ga.op      = a_.op == UCC_OP_AVG ? (ucc_reduction_op_t)12 : a_.op;
        ga.alpha   = a_.op == UCC_OP_AVG ? 1.0f / (float)n_ : 1.0f;
        ga.my_scratch = tt_->area(me_, (uint32_t)pslot_, 0, 0);
        for (uint32_t r = 0; r < n_; r++) {
            ga.peer_scratch[r] = tt_->area(r, (uint32_t)pslot_, 0, 0);
            ga.peer_flags[r]   = tt_->peers_[r].flags;
        }
        ga.local_flags   = tt_->flags_;
        ga.rank          = (int)me_;
        ga.nranks        = (int)n_;
        ga.slot          = pslot_;
        ga.error_word    = tt_->err_host_;
        ga.parity_stride = 2 * tt_->cfg_.chunk;
        /* FIXED grid: per-block launch counters must advance uniformly
         * across every request that ever reuses this slot */
        ga.nblocks = 8;
        ucc_status_t st =
            ec_hip::fused_allreduce_graph(ga, (hipStream_t)ee_stream);
        if (st != UCC_OK) {
            return st;
        }
        status = UCC_OK;
        return UCC_OK;
    }

    ~FusedAllreduceTask() override
    {
        if (pslot_ >= 0) {
            tt_->free_pslot(pslot_);
        }
    }

  private:
    size_t bytes_ = 0;
    int    pslot_ = -1;
};

/* Staged linear collectives: allreduce / allgather(v) / reduce_scatter(v) /
 * bcast / reduce / alltoall(v) / gather(v) / scatter(v). Per fragment
 * (parity p), phases:
 *   A-issue (gated) -> A-wait (event) -> publish kA
 *   B-gate all>=kA  -> issue copy(f+1), launch compute -> B-wait -> kB
 *   [allreduce only] C-gate all>=kB -> gather -> C-wait -> kC
 * Block colls fragment with granularity `chunk`; per-peer colls (alltoall,
 * scatter) partition the staging area into n cells of `cell_` bytes and
 * fragment per-peer blocks with granularity cell_. Ranks may have different
 * nfrags (v-colls): close_slot() publishes base+kCap so early finishers
 * never stall peers' all_ge() gates (slot_seg.h protocol).               */
class StagedTask final : public Cdna4Task {
  public:
    using Cdna4Task::Cdna4Task;

    ucc_status_t post() override
    {
        begin_use();
        const bool inplace = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        ct_     = a_.coll_type;
        steps_  = ct_ == UCC_COLL_TYPE_ALLREDUCE ? 3 : 2;
        alpha_  = 1.0f;
        op_     = a_.op;
        if (op_ == UCC_OP_AVG) {
            op_    = (ucc_reduction_op_t)12;
            alpha_ = 1.0f / (float)n_;
        }
        cnt_.assign(n_, 0);
        dsp_.assign(n_, 0);
        rcnt_.assign(n_, 0);
        rdsp_.assign(n_, 0);
        gran_ = tt_->cfg_.chunk;
        cell_ = (tt_->cfg_.chunk / n_) & ~(size_t)255;
        switch (ct_) {
        case UCC_COLL_TYPE_ALLREDUCE:
            dt_    = a_.dst.info.datatype;
            dtsz_  = ucc_dt_size(dt_);
            total_ = a_.dst.info.count * dtsz_;
            dbuf_  = (uint8_t *)a_.dst.info.buffer;
            sbuf_  = inplace ? dbuf_ : (const uint8_t *)a_.src.info.buffer;
            break;
        case UCC_COLL_TYPE_REDUCE:
            dt_    = a_.src.info.datatype;
            dtsz_  = ucc_dt_size(dt_);
            total_ = a_.src.info.count * dtsz_;
            dbuf_  = (uint8_t *)a_.dst.info.buffer;
            sbuf_  = (inplace && me_ == a_.root)
                         ? dbuf_
                         : (const uint8_t *)a_.src.info.buffer;
            break;
        case UCC_COLL_TYPE_BCAST:
            dt_    = a_.src.info.datatype;
            dtsz_  = ucc_dt_size(dt_);
            total_ = a_.src.info.count * dtsz_;
            dbuf_  = (uint8_t *)a_.src.info.buffer;
            sbuf_  = dbuf_;
            break;
        case UCC_COLL_TYPE_REDUCE_SCATTER: {
            dt_   = a_.dst.info.datatype;
            dtsz_ = ucc_dt_size(dt_);
            size_t out_b;
            if (inplace) {
                total_ = a_.dst.info.count * dtsz_;
                out_b  = total_ / n_;
                sbuf_  = (uint8_t *)a_.dst.info.buffer;
                dbuf_  = (uint8_t *)a_.dst.info.buffer + me_ * out_b;
            } else {
                out_b  = a_.dst.info.count * dtsz_;
                total_ = out_b * n_;
                sbuf_  = (const uint8_t *)a_.src.info.buffer;
                dbuf_  = (uint8_t *)a_.dst.info.buffer;
            }
            for (uint32_t r = 0; r < n_; r++) {
                cnt_[r] = out_b;
                dsp_[r] = (size_t)r * out_b;
            }
            break;
        }
        case UCC_COLL_TYPE_REDUCE_SCATTERV: {
            dt_   = a_.dst.info_v.datatype;
            dtsz_ = ucc_dt_size(dt_);
            size_t off = 0;
            for (uint32_t r = 0; r < n_; r++) {
                cnt_[r] = coll_count_at(a_, a_.dst.info_v.counts, r) * dtsz_;
                dsp_[r] = off;
                off += cnt_[r];
            }
            total_ = off;
            sbuf_  = inplace ? (const uint8_t *)a_.dst.info_v.buffer
                             : (const uint8_t *)a_.src.info.buffer;
            dbuf_  = inplace
                         ? (uint8_t *)a_.dst.info_v.buffer + dsp_[me_]
                         : (uint8_t *)a_.dst.info_v.buffer;
            ct_ = UCC_COLL_TYPE_REDUCE_SCATTER; /* same B compute */
            break;
        }
        case UCC_COLL_TYPE_ALLGATHER: {
            dt_          = a_.dst.info.datatype;
            dtsz_        = ucc_dt_size(dt_);
            size_t block = a_.dst.info.count * dtsz_ / n_;
            for (uint32_t r = 0; r < n_; r++) {
                cnt_[r] = block;
                dsp_[r] = (size_t)r * block;
            }
            dbuf_ = (uint8_t *)a_.dst.info.buffer;
            sbuf_ = inplace ? dbuf_ + dsp_[me_]
                            : (const uint8_t *)a_.src.info.buffer;
            total_  = block;
            break;
        }
        case UCC_COLL_TYPE_ALLGATHERV: {
            dt_   = a_.dst.info_v.datatype;
            dtsz_ = ucc_dt_size(dt_);
            size_t maxb = 0;
            for (uint32_t r = 0; r < n_; r++) {
                cnt_[r] = coll_count_at(a_, a_.dst.info_v.counts, r) * dtsz_;
                dsp_[r] =
                    coll_disp_at(a_, a_.dst.info_v.displacements, r) * dtsz_;
                maxb = cnt_[r] > maxb ? cnt_[r] : maxb;
            }
            dbuf_ = (uint8_t *)a_.dst.info_v.buffer;
            sbuf_ = inplace ? dbuf_ + dsp_[me_]
                            : (const uint8_t *)a_.src.info.buffer;
            total_ = maxb;
            ct_    = UCC_COLL_TYPE_ALLGATHER; /* same A/B machinery */
            break;
        }
        case UCC_COLL_TYPE_ALLTOALL: {
            if (inplace || cell_ == 0) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            dt_          = a_.dst.info.datatype;
            dtsz_        = ucc_dt_size(dt_);
            size_t block = a_.dst.info.count * dtsz_ / n_;
            for (uint32_t r = 0; r < n_; r++) {
                cnt_[r] = rcnt_[r] = block;
                dsp_[r] = rdsp_[r] = (size_t)r * block;
            }
            sbuf_  = (const uint8_t *)a_.src.info.buffer;
            dbuf_  = (uint8_t *)a_.dst.info.buffer;
            total_ = block;
            gran_  = cell_;
            break;
        }
        case UCC_COLL_TYPE_ALLTOALLV: {
            if (inplace || cell_ == 0) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            dt_   = a_.src.info_v.datatype;
            dtsz_ = ucc_dt_size(dt_);
            size_t rdtsz = ucc_dt_size(a_.dst.info_v.datatype);
            size_t maxb  = 0;
            for (uint32_t r = 0; r < n_; r++) {
                cnt_[r] = coll_count_at(a_, a_.src.info_v.counts, r) * dtsz_;
                dsp_[r] =
                    coll_disp_at(a_, a_.src.info_v.displacements, r) * dtsz_;
                rcnt_[r] =
                    coll_count_at(a_, a_.dst.info_v.counts, r) * rdtsz;
                rdsp_[r] =
                    coll_disp_at(a_, a_.dst.info_v.displacements, r) * rdtsz;
                maxb = cnt_[r] > maxb ? cnt_[r] : maxb;
                maxb = rcnt_[r] > maxb ? rcnt_[r] : maxb;
            }
            sbuf_  = (const uint8_t *)a_.src.info_v.buffer;
            dbuf_  = (uint8_t *)a_.dst.info_v.buffer;
            total_ = maxb;
            gran_  = cell_;
            ct_    = UCC_COLL_TYPE_ALLTOALL;
            break;
        }
        case UCC_COLL_TYPE_GATHER:
        case UCC_COLL_TYPE_GATHERV: {
            const bool is_v = ct_ == UCC_COLL_TYPE_GATHERV;
            if (me_ == a_.root) {
                dt_   = is_v ? a_.dst.info_v.datatype : a_.dst.info.datatype;
                dtsz_ = ucc_dt_size(dt_);
                size_t maxb = 0;
                if (is_v) {
                    for (uint32_t r = 0; r < n_; r++) {
                        rcnt_[r] =
                            coll_count_at(a_, a_.dst.info_v.counts, r) *
                            dtsz_;
                        rdsp_[r] =
                            coll_disp_at(a_, a_.dst.info_v.displacements,
                                         r) *
                            dtsz_;
                        maxb = rcnt_[r] > maxb ? rcnt_[r] : maxb;
                    }
                    dbuf_ = (uint8_t *)a_.dst.info_v.buffer;
                } else {
                    size_t block = a_.dst.info.count * dtsz_ / n_;
                    for (uint32_t r = 0; r < n_; r++) {
                        rcnt_[r] = block;
                        rdsp_[r] = (size_t)r * block;
                    }
                    maxb  = block;
                    dbuf_ = (uint8_t *)a_.dst.info.buffer;
                }
                sbuf_ = inplace ? dbuf_ + rdsp_[me_]
                                : (const uint8_t *)a_.src.info.buffer;
                total_ = maxb;
            } else {
                dt_    = a_.src.info.datatype;
                dtsz_  = ucc_dt_size(dt_);
                sbuf_  = (const uint8_t *)a_.src.info.buffer;
                dbuf_  = nullptr;
                total_ = a_.src.info.count * dtsz_;
                cnt_[me_] = total_;
            }
            ct_ = UCC_COLL_TYPE_GATHER;
            break;
        }
        case UCC_COLL_TYPE_SCATTER:
        case UCC_COLL_TYPE_SCATTERV: {
            const bool is_v = ct_ == UCC_COLL_TYPE_SCATTERV;
            if (cell_ == 0) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            if (me_ == a_.root) {
                dt_   = is_v ? a_.src.info_v.datatype : a_.src.info.datatype;
                dtsz_ = ucc_dt_size(dt_);
                size_t maxb = 0;
                if (is_v) {
                    for (uint32_t r = 0; r < n_; r++) {
                        cnt_[r] =
                            coll_count_at(a_, a_.src.info_v.counts, r) *
                            dtsz_;
                        dsp_[r] =
                            coll_disp_at(a_, a_.src.info_v.displacements,
                                         r) *
                            dtsz_;
                        maxb = cnt_[r] > maxb ? cnt_[r] : maxb;
                    }
                    sbuf_ = (const uint8_t *)a_.src.info_v.buffer;
                } else {
                    size_t block = a_.src.info.count * dtsz_ / n_;
                    for (uint32_t r = 0; r < n_; r++) {
                        cnt_[r] = block;
                        dsp_[r] = (size_t)r * block;
                    }
                    maxb  = block;
                    sbuf_ = (const uint8_t *)a_.src.info.buffer;
                }
                dbuf_ = inplace ? nullptr : (uint8_t *)a_.dst.info.buffer;
                rcnt_[me_] = inplace ? 0 : cnt_[me_];
                rdsp_[me_] = 0;
                total_     = maxb;
            } else {
                dt_    = a_.dst.info.datatype;
                dtsz_  = ucc_dt_size(dt_);
                dbuf_  = (uint8_t *)a_.dst.info.buffer;
                sbuf_  = nullptr;
                total_ = a_.dst.info.count * dtsz_;
                rcnt_[a_.root] = total_;
            }
            gran_ = cell_;
            ct_   = UCC_COLL_TYPE_SCATTER;
            break;
        }
        default: return UCC_ERR_NOT_SUPPORTED;
        }
        nfrags_ = (total_ + gran_ - 1) / gran_;
        if (nfrags_ == 0) {
            nfrags_ = 1;
        }
        status = UCC_INPROGRESS;
        return progress();
    }

    ucc_status_t progress() override
    {
        while (frag_ < nfrags_) {
            const size_t   f   = frag_;
            const uint32_t p   = (uint32_t)(f & 1);
            const uint64_t kA  = steps_ * f + 1;
            const uint64_t kB  = steps_ * f + 2;
            const uint64_t kC  = steps_ * f + 3;
            const size_t   off = f * gran_;
            const size_t   len =
                total_ - off < gran_ ? total_ - off : gran_;
            switch (phase_) {
            case 0: /* entry: first copy */
                if (!all_ge(0)) {
                    return UCC_INPROGRESS;
                }
                issue_copy(0);
                phase_ = 1;
                break;
            case 1: { /* A-wait */
                if (copy_pending_) {
                    hipError_t e = hipEventQuery(ev(0 + (int)p));
                    if (e == hipErrorNotReady) {
                        return UCC_INPROGRESS;
                    }
                    if (e != hipSuccess) {
                        return UCC_ERR_NO_RESOURCE;
                    }
                }
                publish(kA);
                phase_ = 2;
                break;
            }
            case 2: { /* B-gate */
                if (!all_ge(kA)) {
                    return UCC_INPROGRESS;
                }
                if (f + 1 < nfrags_) {
                    issue_copy(f + 1);
                }
                bool launched = launch_compute(f, p, off, len);
                if (launched) {
                    HIPWARN(hipEventRecord(ev(2), comp()));
                }
                b_launched_ = launched;
                phase_      = 3;
                break;
            }
            case 3: { /* B-wait */
                if (b_launched_) {
                    hipError_t e = hipEventQuery(ev(2));
                    if (e == hipErrorNotReady) {
                        return UCC_INPROGRESS;
                    }
                    if (e != hipSuccess) {
                        return UCC_ERR_NO_RESOURCE;
                    }
                }
                publish(kB);
                if (steps_ == 2) {
                    phase_ = 1;
                    frag_++;
                    break;
                }
                phase_ = 4;
                break;
            }
            case 4: { /* C-gate (allreduce gather) */
                if (!all_ge(kB)) {
                    return UCC_INPROGRESS;
                }
                launch_gather(f, p, off, len);
                HIPWARN(hipEventRecord(ev(3), comp()));
                phase_ = 5;
                break;
            }
            case 5: { /* C-wait */
                hipError_t e = hipEventQuery(ev(3));
                if (e == hipErrorNotReady) {
                    return UCC_INPROGRESS;
                }
                if (e != hipSuccess) {
                    return UCC_ERR_NO_RESOURCE;
                }
                publish(kC);
                phase_ = 1;
                frag_++;
                break;
            }
            }
        }
        close_slot();
        return UCC_OK;
    }

  private:
    /* stage my contribution of fragment f into in[slot][f%2] */
    void issue_copy(size_t f)
    {
        const uint32_t p   = (uint32_t)(f & 1);
        const size_t   off = f * gran_;
        size_t         len = 0;
        const uint8_t *src = nullptr;
        copy_pending_ = false;
        switch (ct_) {
        case UCC_COLL_TYPE_ALLTOALL: {
            /* per-dest cells: in[r*cell_] <- sbuf[dsp_[r]+off ..] */
            ec_hip::GatherArgs ga{};
            ga.dst_base = tt_->area(me_, slot_, p, 0);
            int k       = 0;
            for (uint32_t r = 0; r < n_; r++) {
                if (off >= cnt_[r]) {
                    continue;
                }
                size_t l = cnt_[r] - off < cell_ ? cnt_[r] - off : cell_;
                ga.srcs[k] = sbuf_ + dsp_[r] + off;
                ga.offs[k] = (uint64_t)r * cell_;
                ga.lens[k] = l;
                k++;
            }
            if (k > 0) {
                ga.n = k;
                ec_hip::gather_copy(ga, copy_s());
                copy_pending_ = true;
            }
            HIPWARN(hipEventRecord(ev(0 + (int)p), copy_s()));
            return;
        }
        case UCC_COLL_TYPE_SCATTER: {
            if (me_ == a_.root) {
                ec_hip::GatherArgs ga{};
                ga.dst_base = tt_->area(me_, slot_, p, 0);
                int k       = 0;
                for (uint32_t r = 0; r < n_; r++) {
                    if (r == me_ || off >= cnt_[r]) {
                        continue;
                    }
                    size_t l = cnt_[r] - off < cell_ ? cnt_[r] - off
                                                     : cell_;
                    ga.srcs[k] = sbuf_ + dsp_[r] + off;
                    ga.offs[k] = (uint64_t)r * cell_;
                    ga.lens[k] = l;
                    k++;
                }
                if (k > 0) {
                    ga.n = k;
                    ec_hip::gather_copy(ga, copy_s());
                    copy_pending_ = true;
                }
            }
            HIPWARN(hipEventRecord(ev(0 + (int)p), copy_s()));
            return;
        }
        case UCC_COLL_TYPE_ALLGATHER:
        case UCC_COLL_TYPE_GATHER:
            if (off < cnt_[me_]) {
                len = cnt_[me_] - off < gran_ ? cnt_[me_] - off : gran_;
                src = sbuf_ + off;
            }
            break;
        case UCC_COLL_TYPE_BCAST:
            if (me_ == a_.root && off < total_) {
                len = total_ - off < gran_ ? total_ - off : gran_;
                src = sbuf_ + off;
            }
            break;
        default:
            if (off < total_) {
                len = total_ - off < gran_ ? total_ - off : gran_;
                src = sbuf_ + off;
            }
            break;
        }
        copy_pending_ = len > 0;
        if (len > 0) {
            /* stage with a COPY KERNEL by default: kernel writes are
             * L2-coherent with the peer processes' reader kernels on
             * this device by construction, while hipMemcpyAsync may
             * ride an SDMA engine (one unreproduced data-mismatch on
             * this path is on record — NEXT_STEPS.md). SDMA staging
             * stays available via UCC_TL_CDNA4_STAGE_SDMA=1. */
            static const bool sdma = Config::instance().get_bool(
                "TL_CDNA4", "STAGE_SDMA", false);
            if (sdma) {
                HIPWARN(hipMemcpyAsync(tt_->area(me_, slot_, p, 0), src,
                                       len, hipMemcpyDeviceToDevice,
                                       copy_s()));
            } else {
                ec_hip::GatherArgs ga{};
                ga.dst_base = tt_->area(me_, slot_, p, 0);
                ga.srcs[0]  = src;
                ga.offs[0]  = 0;
                ga.lens[0]  = len;
                ga.n        = 1;
                ec_hip::gather_copy(ga, copy_s());
            }
        }
        HIPWARN(hipEventRecord(ev(0 + (int)p), copy_s()));
    }

    /* B compute for fragment f; returns whether a kernel was launched */
    bool launch_compute(size_t f, uint32_t p, size_t off, size_t len)
    {
        switch (ct_) {
        case UCC_COLL_TYPE_ALLREDUCE: {
            /* reduce my 1/n slice of the fragment into out area */
            size_t nelem = len / dtsz_;
            size_t gran  = 256 / dtsz_;
            size_t per   = (nelem / n_) / gran * gran;
            size_t b     = me_ * per;
            size_t e     = me_ == n_ - 1 ? nelem : b + per;
            if (e <= b) {
                return false;
            }
            ec_hip::ReduceArgs ra{};
            ra.dst    = tt_->area(me_, slot_, p, 1);
            ra.n_srcs = (int)n_;
            ra.count  = e - b;
            ra.dt     = dt_;
            ra.op     = op_;
            ra.alpha  = alpha_;
            for (uint32_t r = 0; r < n_; r++) {
                ra.srcs[r] = tt_->area(r, slot_, p, 0) + b * dtsz_;
            }
            ec_hip::reduce(ra, comp());
            return true;
        }
        case UCC_COLL_TYPE_REDUCE: {
            if (me_ != a_.root) {
                return false;
            }
            ec_hip::ReduceArgs ra{};
            ra.dst    = dbuf_ + off;
            ra.n_srcs = (int)n_;
            ra.count  = len / dtsz_;
            ra.dt     = dt_;
            ra.op     = op_;
            ra.alpha  = alpha_;
            for (uint32_t r = 0; r < n_; r++) {
                ra.srcs[r] = tt_->area(r, slot_, p, 0);
            }
            ec_hip::reduce(ra, comp());
            return true;
        }
        case UCC_COLL_TYPE_REDUCE_SCATTER: {
            /* my dst slice ∩ fragment */
            size_t s0 = dsp_[me_], s1 = dsp_[me_] + cnt_[me_];
            size_t f0 = off, f1 = off + len;
            size_t b  = s0 > f0 ? s0 : f0;
            size_t e  = s1 < f1 ? s1 : f1;
            if (e <= b) {
                return false;
            }
            ec_hip::ReduceArgs ra{};
            ra.dst    = dbuf_ + (b - s0);
            ra.n_srcs = (int)n_;
            ra.count  = (e - b) / dtsz_;
            ra.dt     = dt_;
            ra.op     = op_;
            ra.alpha  = alpha_;
            for (uint32_t r = 0; r < n_; r++) {
                ra.srcs[r] = tt_->area(r, slot_, p, 0) + (b - off);
            }
            ec_hip::reduce(ra, comp());
            return true;
        }
        case UCC_COLL_TYPE_ALLGATHER: {
            ec_hip::GatherArgs ga{};
            int k = 0;
            for (uint32_t r = 0; r < n_; r++) {
                if (off >= cnt_[r]) {
                    continue;
                }
                size_t l = cnt_[r] - off < tt_->cfg_.chunk
                               ? cnt_[r] - off
                               : tt_->cfg_.chunk;
                if (r == me_ && (a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE)) {
                    continue;
                }
                ga.srcs[k] = tt_->area(r, slot_, p, 0);
                ga.offs[k] = dsp_[r] + off;
                ga.lens[k] = l;
                k++;
            }
            if (k == 0) {
                return false;
            }
            ga.dst_base = dbuf_;
            ga.n        = k;
            ec_hip::gather_copy(ga, comp());
            return true;
        }
        case UCC_COLL_TYPE_BCAST: {
            if (me_ == a_.root) {
                return false;
            }
            ec_hip::GatherArgs ga{};
            ga.dst_base = dbuf_;
            ga.srcs[0]  = tt_->area((uint32_t)a_.root, slot_, p, 0);
            ga.offs[0]  = off;
            ga.lens[0]  = len;
            ga.n        = 1;
            ec_hip::gather_copy(ga, comp());
            return true;
        }
        case UCC_COLL_TYPE_ALLTOALL: {
            /* read my cell from every peer's staged fragment */
            ec_hip::GatherArgs ga{};
            ga.dst_base = dbuf_;
            int k       = 0;
            for (uint32_t r = 0; r < n_; r++) {
                if (off >= rcnt_[r]) {
                    continue;
                }
                size_t l = rcnt_[r] - off < cell_ ? rcnt_[r] - off : cell_;
                ga.srcs[k] = tt_->area(r, slot_, p, 0) + me_ * cell_;
                ga.offs[k] = rdsp_[r] + off;
                ga.lens[k] = l;
                k++;
            }
            if (k == 0) {
                return false;
            }
            ga.n = k;
            ec_hip::gather_copy(ga, comp());
            return true;
        }
        case UCC_COLL_TYPE_GATHER: {
            if (me_ != a_.root) {
                return false;
            }
            ec_hip::GatherArgs ga{};
            ga.dst_base = dbuf_;
            int k       = 0;
            for (uint32_t r = 0; r < n_; r++) {
                if (off >= rcnt_[r]) {
                    continue;
                }
                size_t l = rcnt_[r] - off < gran_ ? rcnt_[r] - off : gran_;
                if (r == me_) {
                    if (a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE) {
                        continue; /* already in place */
                    }
                    ga.srcs[k] = sbuf_ + off;
                } else {
                    ga.srcs[k] = tt_->area(r, slot_, p, 0);
                }
                ga.offs[k] = rdsp_[r] + off;
                ga.lens[k] = l;
                k++;
            }
            if (k == 0) {
                return false;
            }
            ga.n = k;
            ec_hip::gather_copy(ga, comp());
            return true;
        }
        case UCC_COLL_TYPE_SCATTER: {
            if (me_ == a_.root) {
                /* my own block: direct local copy (skipped if inplace) */
                if (!dbuf_ || off >= cnt_[me_]) {
                    return false;
                }
                size_t l = cnt_[me_] - off < cell_ ? cnt_[me_] - off
                                                   : cell_;
                ec_hip::GatherArgs ga{};
                ga.dst_base = dbuf_;
                ga.srcs[0]  = sbuf_ + dsp_[me_] + off;
                ga.offs[0]  = off;
                ga.lens[0]  = l;
                ga.n        = 1;
                ec_hip::gather_copy(ga, comp());
                return true;
            }
            if (off >= rcnt_[a_.root]) {
                return false;
            }
            size_t l = rcnt_[a_.root] - off < cell_ ? rcnt_[a_.root] - off
                                                    : cell_;
            ec_hip::GatherArgs ga{};
            ga.dst_base = dbuf_;
            ga.srcs[0] =
                tt_->area((uint32_t)a_.root, slot_, p, 0) + me_ * cell_;
            ga.offs[0] = off;
            ga.lens[0] = l;
            ga.n       = 1;
            ec_hip::gather_copy(ga, comp());
            return true;
        }
        default: return false;
        }
    }

    /* allreduce stage C: gather every rank's reduced slice */
    void launch_gather(size_t f, uint32_t p, size_t off, size_t len)
    {
        (void)f;
        size_t nelem = len / dtsz_;
        size_t gran  = 256 / dtsz_;
        size_t per   = (nelem / n_) / gran * gran;
        ec_hip::GatherArgs ga{};
        ga.dst_base = dbuf_ + off;
        int k       = 0;
        for (uint32_t r = 0; r < n_; r++) {
            size_t b = r * per;
            size_t e = r == n_ - 1 ? nelem : b + per;
            if (e <= b) {
                continue;
            }
            ga.srcs[k] = tt_->area(r, slot_, p, 1);
            ga.offs[k] = b * dtsz_;
            ga.lens[k] = (e - b) * dtsz_;
            k++;
        }
        ga.n = k;
        ec_hip::gather_copy(ga, comp());
    }

    ucc_coll_type_t ct_ = UCC_COLL_TYPE_ALLREDUCE;
    const uint8_t  *sbuf_ = nullptr;
    uint8_t        *dbuf_ = nullptr;
    ucc_datatype_t  dt_   = UCC_DT_INT8;
    size_t          dtsz_ = 1;
    ucc_reduction_op_t op_ = UCC_OP_SUM;
    float           alpha_ = 1.0f;
    size_t          total_ = 0, nfrags_ = 0;
    size_t          gran_ = 0, cell_ = 0;
    std::vector<size_t> cnt_, dsp_, rcnt_, rdsp_;
    int             steps_ = 3;
    bool            copy_pending_ = false, b_launched_ = false;
};

/* Device-gated staged allreduce: the large-message fast path. All
 * fragments' stage/reduce/gather kernels are enqueued up front (stage on
 * the copy stream, reduce+gather on the compute stream); cross-rank and
 * cross-fragment ordering is enforced INSIDE the kernels by prologue
 * spins on cumulative per-(slot,parity,phase) block counters in the
 * peers' fine-grained flag buffers (ec_hip::GatedArgs) — zero host
 * round-trips inside the collective, one trailing event. Same xGMI
 * traffic as the host-gated staged path (1.75 x S per rank), minus all
 * host gating latency. */
class GatedCollTask final : public Cdna4Task {
  public:
    using Cdna4Task::Cdna4Task;

    ucc_status_t post() override
    {
        begin_use();
        if (slot_ >= (uint32_t)ec_hip::kGatedSlots) {
            return UCC_ERR_NOT_SUPPORTED;
        }
        ucc_status_t st = setup_coll();
        if (st != UCC_OK) {
            return st;
        }
        if (!zc_ready_ &&
            (ct_ == UCC_COLL_TYPE_ALLREDUCE ||
             ct_ == UCC_COLL_TYPE_REDUCE_SCATTER ||
             ct_ == UCC_COLL_TYPE_ALLGATHER ||
             ct_ == UCC_COLL_TYPE_ALLTOALL) &&
            Config::instance().get_bool("TL_CDNA4", "ZCOPY", true)) {
            if (a_.flags & UCC_COLL_ARGS_FLAG_PERSISTENT) {
                zc_ = true; /* exchange amortized over re-posts */
            } else if (a_.flags & UCC_COLL_ARGS_FLAG_MEM_MAPPED_BUFFERS) {
                /* caller promises src/dst sit inside ucc_mem_map'd
                 * regions: imports come from the memh/IPC caches, so
                 * zero-copy pays at any size (reference MEM_MAPPED
                 * flag role, ucc.h onesided semantics) */
                zc_ = true;
            } else {
                /* one-shot: the 4 host-gated exchange rounds (team
                 * IPC-import cache makes the opens free after first
                 * contact) pay off once the staging copy they remove
                 * is big enough */
                size_t min_b = Config::instance().get_size(
                    "TL_CDNA4", "ZCOPY_ONESHOT_MIN", 16 * 1024 * 1024);
                if (min_b > 0 && total_ >= min_b) {
                    zc_ = true;
                }
            }
        }
        phase_ = zc_ && !zc_ready_ ? 10 : 0;
        if (ct_ == UCC_COLL_TYPE_ALLTOALLV && !a2av_ready_) {
            phase_ = 20; /* global-max exchange first */
        }
        status = UCC_INPROGRESS;
        return progress();
    }

    /* Stream-triggered / hipGraph-capturable post: kernels derive their
     * iteration on device (GatedArgs.derive) on a DEDICATED slot, so one
     * capture replays as a fresh collective each time — the large-message
     * counterpart of the fused graph kernel (docs/GATED_PIPELINE.md). */
    ucc_status_t triggered_post(void *ee_stream) override
    {
        if (a_.coll_type == UCC_COLL_TYPE_ALLTOALLV) {
            /* needs a host-side global-max exchange before enqueue:
             * not expressible as a pure stream capture */
            return UCC_ERR_NOT_SUPPORTED;
        }
        if (pslot_ < 0) {
            ucc_coll_type_t k_ct = a_.coll_type;
            int kind = k_ct == UCC_COLL_TYPE_ALLREDUCE
                           ? Cdna4TlTeam::PK_G_AR
                       : (k_ct == UCC_COLL_TYPE_REDUCE_SCATTER ||
                          k_ct == UCC_COLL_TYPE_REDUCE_SCATTERV)
                           ? Cdna4TlTeam::PK_G_RS
                       : (k_ct == UCC_COLL_TYPE_ALLGATHER ||
                          k_ct == UCC_COLL_TYPE_ALLGATHERV)
                           ? Cdna4TlTeam::PK_G_AG
                           : Cdna4TlTeam::PK_G_A2A;
            pslot_ = tt_->alloc_pslot(kind);
            if (pslot_ < 0 || pslot_ >= ec_hip::kGatedSlots) {
                ucc_error("no persistent slot for triggered gated coll "
                          "(PERSISTENT_SLOTS exhausted or pattern "
                          "mismatch)");
                return UCC_ERR_NO_RESOURCE;
            }
            me_ = tt_->team_->rank;
            n_  = tt_->team_->size;
            ucc_status_t st = setup_coll();
            if (st != UCC_OK) {
                return st;
            }
        }
        slot_ = (uint32_t)pslot_;
        ucc_status_t st = enqueue_frags((hipStream_t)ee_stream,
                                        (hipStream_t)ee_stream, true);
        if (st != UCC_OK) {
            return st;
        }
        status = UCC_OK; /* stream-ordered completion */
        return UCC_OK;
    }

  private:
    ucc_status_t setup_coll()
    {
        const bool inplace = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        ct_     = a_.coll_type;
        op_     = a_.op;
        alpha_  = 1.0f;
        gran_   = tt_->cfg_.chunk;
        cell_   = (tt_->cfg_.chunk / n_) & ~(size_t)255;
        if (op_ == UCC_OP_AVG) {
            op_    = (ucc_reduction_op_t)12;
            alpha_ = 1.0f / (float)n_;
        }
        switch (ct_) {
        case UCC_COLL_TYPE_ALLREDUCE:
            dt_    = a_.dst.info.datatype;
            dtsz_  = ucc_dt_size(dt_);
            total_ = a_.dst.info.count * dtsz_;
            dbuf_  = (uint8_t *)a_.dst.info.buffer;
            sbuf_  = inplace ? dbuf_ : (const uint8_t *)a_.src.info.buffer;
            break;
        case UCC_COLL_TYPE_REDUCE_SCATTER: {
            dt_   = a_.dst.info.datatype;
            dtsz_ = ucc_dt_size(dt_);
            size_t out_b;
            if (inplace) {
                total_ = a_.dst.info.count * dtsz_;
                out_b  = total_ / n_;
                sbuf_  = (uint8_t *)a_.dst.info.buffer;
                dbuf_  = (uint8_t *)a_.dst.info.buffer + me_ * out_b;
            } else {
                out_b  = a_.dst.info.count * dtsz_;
                total_ = out_b * n_;
                sbuf_  = (const uint8_t *)a_.src.info.buffer;
                dbuf_  = (uint8_t *)a_.dst.info.buffer;
            }
            out_b_ = out_b;
            break;
        }
        case UCC_COLL_TYPE_ALLGATHER: {
            dt_    = a_.dst.info.datatype;
            dtsz_  = ucc_dt_size(dt_);
            size_t block = a_.dst.info.count * dtsz_ / n_;
            out_b_ = block;
            cnt_.assign(n_, block);
            dsp_.resize(n_);
            for (uint32_t r = 0; r < n_; r++) {
                dsp_[r] = (size_t)r * block;
            }
            dbuf_  = (uint8_t *)a_.dst.info.buffer;
            sbuf_  = inplace ? dbuf_ + me_ * block
                             : (const uint8_t *)a_.src.info.buffer;
            total_ = block; /* fragment over the per-rank block */
            break;
        }
        case UCC_COLL_TYPE_ALLGATHERV: {
            /* per-rank counts are global knowledge (dst.info_v), so the
             * symmetric-launch invariant holds: every rank launches
             * ceil(max_block / chunk) fragments (empty tails still
             * launch and signal) */
            dt_   = a_.dst.info_v.datatype;
            dtsz_ = ucc_dt_size(dt_);
            cnt_.resize(n_);
            dsp_.resize(n_);
            size_t maxb = 0;
            for (uint32_t r = 0; r < n_; r++) {
                cnt_[r] = coll_count_at(a_, a_.dst.info_v.counts, r) *
                          dtsz_;
                dsp_[r] =
                    coll_disp_at(a_, a_.dst.info_v.displacements, r) *
                    dtsz_;
                maxb = cnt_[r] > maxb ? cnt_[r] : maxb;
            }
            dbuf_  = (uint8_t *)a_.dst.info_v.buffer;
            sbuf_  = inplace ? dbuf_ + dsp_[me_]
                             : (const uint8_t *)a_.src.info.buffer;
            total_ = maxb;
            ct_    = UCC_COLL_TYPE_ALLGATHERV;
            break;
        }
        case UCC_COLL_TYPE_REDUCE_SCATTERV: {
            dt_   = a_.dst.info_v.datatype;
            dtsz_ = ucc_dt_size(dt_);
            cnt_.resize(n_);
            dsp_.resize(n_);
            size_t off = 0;
            for (uint32_t r = 0; r < n_; r++) {
                cnt_[r] = coll_count_at(a_, a_.dst.info_v.counts, r) *
                          dtsz_;
                dsp_[r] = off;
                off += cnt_[r];
            }
            total_ = off;
            sbuf_  = inplace ? (const uint8_t *)a_.dst.info_v.buffer
                             : (const uint8_t *)a_.src.info.buffer;
            dbuf_  = inplace
                         ? (uint8_t *)a_.dst.info_v.buffer + dsp_[me_]
                         : (uint8_t *)a_.dst.info_v.buffer;
            ct_ = UCC_COLL_TYPE_REDUCE_SCATTERV;
            break;
        }
        case UCC_COLL_TYPE_ALLTOALL: {
            if (inplace || cell_ == 0) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            dt_    = a_.dst.info.datatype;
            dtsz_  = ucc_dt_size(dt_);
            out_b_ = a_.dst.info.count * dtsz_ / n_; /* per-peer block */
            sbuf_  = (const uint8_t *)a_.src.info.buffer;
            dbuf_  = (uint8_t *)a_.dst.info.buffer;
            total_ = out_b_;
            gran_  = cell_;
            break;
        }
        case UCC_COLL_TYPE_ALLTOALLV: {
            /* Per-peer send row (src.info_v) and recv column
             * (dst.info_v) are LOCAL knowledge, but the symmetric-launch
             * invariant needs the GLOBAL max pair length — exchanged
             * host-side through the scratch channel (phases 20/21)
             * before any fragment is enqueued. Rank i's send cell to j
             * and its gather of j's cell both clip to the locally-known
             * lengths; empty tails still launch and signal. */
            if (inplace || cell_ == 0) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            dt_          = a_.src.info_v.datatype;
            dtsz_        = ucc_dt_size(dt_);
            size_t rdtsz = ucc_dt_size(a_.dst.info_v.datatype);
            cnt_.resize(n_);
            dsp_.resize(n_);
            rcnt_.resize(n_);
            rdsp_.resize(n_);
            size_t lmax = 0;
            for (uint32_t r = 0; r < n_; r++) {
                cnt_[r] = coll_count_at(a_, a_.src.info_v.counts, r) *
                          dtsz_;
                dsp_[r] =
                    coll_disp_at(a_, a_.src.info_v.displacements, r) *
                    dtsz_;
                rcnt_[r] = coll_count_at(a_, a_.dst.info_v.counts, r) *
                           rdtsz;
                rdsp_[r] =
                    coll_disp_at(a_, a_.dst.info_v.displacements, r) *
                    rdtsz;
                lmax = cnt_[r] > lmax ? cnt_[r] : lmax;
                lmax = rcnt_[r] > lmax ? rcnt_[r] : lmax;
            }
            sbuf_  = (const uint8_t *)a_.src.info_v.buffer;
            dbuf_  = (uint8_t *)a_.dst.info_v.buffer;
            total_ = lmax; /* provisional; replaced by the global max */
            gran_  = cell_;
            break;
        }
        default:
            return UCC_ERR_NOT_SUPPORTED;
        }
        nfrags_ = (total_ + gran_ - 1) / gran_;
        if (nfrags_ == 0) {
            nfrags_ = 1;
        }
        return UCC_OK;
    }

  public:
    /* zero-copy handle exchange: my src's HIP-IPC handle travels through
     * my scratch in-area (peers read it over the already-mapped IPC
     * scratch), gated by the shm slot steps. One slot use, then the
     * collective itself takes a fresh use. */
    struct ZcBlob {
        uint64_t          magic;
        hipIpcMemHandle_t h;      /* src allocation  */
        uint64_t          base_off;
        uint64_t          raw_ptr;
        hipIpcMemHandle_t hd;     /* dst allocation  */
        uint64_t          d_base_off;
        uint64_t          d_raw_ptr;
        int32_t           pid;
        int32_t           pad;
    };

    ucc_status_t progress() override
    {
        if (phase_ == 20) { /* a2av: publish my local max pair length */
            if (!all_ge(0)) {
                return UCC_INPROGRESS;
            }
            uint64_t mm[2] = {(uint64_t)total_, 0};
            for (uint32_t r = 0; r < n_; r++) {
                mm[1] += cnt_[r]; /* my total send bytes */
            }
            if (hipMemcpy(tt_->area(me_, slot_, 0, 0), mm, sizeof(mm),
                          hipMemcpyHostToDevice) != hipSuccess) {
                return UCC_ERR_NO_RESOURCE;
            }
            publish(1);
            phase_ = 21;
        }
        if (phase_ == 21) { /* a2av: global max + global byte sum */
            if (!all_ge(1)) {
                return UCC_INPROGRESS;
            }
            uint64_t gmax = 0;
            a2av_gsum_    = 0;
            for (uint32_t r = 0; r < n_; r++) {
                uint64_t v[2] = {0, 0};
                if (hipMemcpy(v, tt_->area(r, slot_, 0, 0), sizeof(v),
                              hipMemcpyDeviceToHost) != hipSuccess) {
                    return UCC_ERR_NO_RESOURCE;
                }
                gmax = v[0] > gmax ? v[0] : gmax;
                a2av_gsum_ += v[1];
            }
            total_  = gmax;
            nfrags_ = (total_ + gran_ - 1) / gran_;
            if (nfrags_ == 0) {
                nfrags_ = 1;
            }
            a2av_ready_ = true;
            close_slot();
            begin_use(); /* fresh slot use for the collective itself */
            phase_ = 0;
            /* CE a2av (reference alltoallv_ce.c role): symmetric
             * decision from the GLOBAL byte sum (exchanged with the
             * max), so skewed MoE matrices — whose per-rank totals
             * differ — still get one team-wide verdict. The zc
             * exchange then maps peers' src buffers and their send
             * tables so each rank PULLS its column with SDMA memcpys.
             * SDMA also sidesteps the gated path's worst skew cost:
             * fragment count there scales with the MAX pair length. */
            if (!zc_ready_ && tt_->cfg_.ce_alltoall &&
                Config::instance().get_bool("TL_CDNA4", "ZCOPY", true) &&
                a2av_gsum_ / (n_ ? n_ : 1) >=
                    tt_->cfg_.ce_alltoall_min / 4) {
                zc_    = true;
                phase_ = 10;
            }
        }
        if (phase_ == 10) { /* publish my src handle */
            if (!all_ge(0)) {
                return UCC_INPROGRESS;
            }
            ZcBlob b{};
            b.magic = 0x5a43;
            /* registered-buffer fast path (reference
             * alltoall_onesided.c src_memh/dst_memh role): handles
             * come from the user's ucc_mem_map export instead of a
             * per-post hipIpcGetMemHandle */
            uint64_t moff = 0;
            bool     src_reg =
                (a_.mask & UCC_COLL_ARGS_FIELD_MEM_MAP_SRC_MEMH) &&
                a_.src_memh &&
                ucc_memh_lookup(a_.src_memh, sbuf_, total_, &b.h,
                                &moff);
            if (src_reg) {
                g_memh_uses.fetch_add(1, std::memory_order_relaxed);
                b.base_off = moff;
                b.raw_ptr  = (uint64_t)(uintptr_t)sbuf_;
                b.pid      = (int32_t)getpid();
                uint64_t doff = 0;
                if ((a_.mask &
                     UCC_COLL_ARGS_FIELD_MEM_MAP_DST_MEMH) &&
                    a_.dst_memh &&
                    ucc_memh_lookup(a_.dst_memh, dbuf_, total_, &b.hd,
                                    &doff)) {
                    b.d_base_off = doff;
                    b.d_raw_ptr  = (uint64_t)(uintptr_t)dbuf_;
                    b.pad        = 0;
                } else {
                    /* dst not registered: fall through to the runtime
                     * export for the dst side only */
                    hipDeviceptr_t dbase  = nullptr;
                    size_t         dbsize = 0;
                    if (hipMemGetAddressRange(&dbase, &dbsize,
                                              (hipDeviceptr_t)dbuf_) ==
                            hipSuccess &&
                        hipIpcGetMemHandle(&b.hd, (void *)dbase) ==
                            hipSuccess) {
                        b.d_base_off = (uint64_t)((uintptr_t)dbuf_ -
                                                  (uintptr_t)dbase);
                        b.d_raw_ptr  = (uint64_t)(uintptr_t)dbuf_;
                        b.pad        = 0;
                    } else {
                        b.magic = 0;
                        zc_     = false;
                    }
                }
            }
            hipDeviceptr_t base  = nullptr;
            size_t         bsize = 0;
            if (src_reg) {
                /* blob already filled from the registered handles */
            } else if (hipMemGetAddressRange(&base, &bsize,
                                      (hipDeviceptr_t)sbuf_) !=
                    hipSuccess ||
                hipIpcGetMemHandle(&b.h, (void *)base) != hipSuccess) {
                /* MUST stay in the consensus rounds: a unilateral jump
                 * to staging would satisfy peers' all_ge() gates via
                 * close_slot() while they read a stale blob from a
                 * prior use of this rotating slot — silent wrong mode.
                 * Publish a magic=0 blob so the AND-consensus in
                 * phases 11-13 forces every rank to staging. */
                ucc_warn("zero-copy src export failed, using staging");
                zc_ = false;
                b   = ZcBlob{}; /* magic = 0 */
            } else {
                b.base_off =
                    (uint64_t)((uintptr_t)sbuf_ - (uintptr_t)base);
                b.raw_ptr = (uint64_t)(uintptr_t)sbuf_;
                b.pid     = (int32_t)getpid();
                hipDeviceptr_t dbase  = nullptr;
                size_t         dbsize = 0;
                if (ct_ == UCC_COLL_TYPE_ALLTOALLV) {
                    /* CE a2av pulls from peers' src into LOCAL dst:
                     * no dst export needed */
                } else if (hipMemGetAddressRange(&dbase, &dbsize,
                                          (hipDeviceptr_t)dbuf_) ==
                        hipSuccess &&
                    hipIpcGetMemHandle(&b.hd, (void *)dbase) ==
                        hipSuccess) {
                    b.d_base_off =
                        (uint64_t)((uintptr_t)dbuf_ - (uintptr_t)dbase);
                    b.d_raw_ptr = (uint64_t)(uintptr_t)dbuf_;
                    b.pad = (dbase == base) ? 1 : 0; /* same alloc */
                } else {
                    b.magic = 0; /* dst export failed: fall back */
                }
            }
            if (hipMemcpy(tt_->area(me_, slot_, 0, 0), &b, sizeof(b),
                          hipMemcpyHostToDevice) != hipSuccess) {
                return UCC_ERR_NO_RESOURCE;
            }
            if (ct_ == UCC_COLL_TYPE_ALLTOALLV) {
                /* publish my send table (byte counts + displs) after
                 * the blob so peers can pull their columns */
                uint64_t tbl[2 * ec_hip::kMaxRanks] = {};
                for (uint32_t r = 0; r < n_; r++) {
                    tbl[r]                      = cnt_[r];
                    tbl[ec_hip::kMaxRanks + r] = dsp_[r];
                }
                if (hipMemcpy(tt_->area(me_, slot_, 0, 0) + 256, tbl,
                              sizeof(tbl),
                              hipMemcpyHostToDevice) != hipSuccess) {
                    return UCC_ERR_NO_RESOURCE;
                }
            }
            publish(1);
            phase_ = 11;
        }
        if (phase_ == 11) { /* open peers' handles */
            if (!all_ge(1)) {
                return UCC_INPROGRESS;
            }
            const bool a2av = ct_ == UCC_COLL_TYPE_ALLTOALLV;
            for (uint32_t r = 0; r < n_ && zc_; r++) {
                if (a2av) { /* read peer r's send table */
                    uint64_t tbl[2 * ec_hip::kMaxRanks];
                    if (hipMemcpy(tbl, tt_->area(r, slot_, 0, 0) + 256,
                                  sizeof(tbl),
                                  hipMemcpyDeviceToHost) != hipSuccess) {
                        zc_ = false;
                        break;
                    }
                    zc_scol_[r] = tbl[me_];
                    zc_sdsp_[r] = tbl[ec_hip::kMaxRanks + me_];
                }
                if (r == me_) {
                    zc_peer_src_[r] = sbuf_;
                    zc_peer_dst_[r] = dbuf_;
                    continue;
                }
                ZcBlob b{};
                if (hipMemcpy(&b, tt_->area(r, slot_, 0, 0), sizeof(b),
                              hipMemcpyDeviceToHost) != hipSuccess ||
                    b.magic != 0x5a43) {
                    ucc_warn("zero-copy import failed, using staging");
                    zc_ = false;
                    break;
                }
                if (b.pid == tt_->team_->ctx->proc.pid) {
                    zc_peer_src_[r] =
                        (const uint8_t *)(uintptr_t)b.raw_ptr;
                    zc_peer_dst_[r] = (uint8_t *)(uintptr_t)b.d_raw_ptr;
                    continue;
                }
                void *m = tt_->ipc_open_cached(
                    b.pid, b.raw_ptr - b.base_off, b.h);
                void *md = nullptr;
                if (!m) {
                    ucc_warn("zero-copy open failed, using staging");
                    zc_ = false;
                    break;
                }
                zc_peer_src_[r] = (const uint8_t *)m + b.base_off;
                if (a2av) {
                    continue; /* pulls land in LOCAL dst */
                }
                md              = tt_->ipc_open_cached(
                    b.pid, b.d_raw_ptr - b.d_base_off, b.hd);
                if (md) {
                    zc_peer_dst_[r] = (uint8_t *)md + b.d_base_off;
                } else if (b.pad == 1) {
                    /* dst lives in the src allocation (e.g. inplace) */
                    zc_peer_dst_[r] =
                        (uint8_t *)m +
                        ((uint64_t)b.d_raw_ptr -
                         ((uint64_t)b.raw_ptr - b.base_off));
                } else {
                    ucc_warn("zero-copy dst open failed, using staging");
                    zc_ = false;
                    break;
                }
            }
            /* direct vector loads/stores need 16B-aligned user
             * buffers on every rank (CE a2av moves bytes with
             * memcpy: exempt) */
            if (zc_ && !a2av) {
                for (uint32_t r = 0; r < n_; r++) {
                    if (((uintptr_t)zc_peer_src_[r] & 15) ||
                        ((uintptr_t)zc_peer_dst_[r] & 15)) {
                        ucc_warn("zero-copy needs 16B-aligned buffers, "
                                 "using staging");
                        zc_ = false;
                        break;
                    }
                }
            }
            publish(2);
            phase_ = 12;
        }
        if (phase_ == 12) { /* blobs consumed -> publish my ok verdict */
            if (!all_ge(2)) {
                return UCC_INPROGRESS;
            }
            /* zero-copy must be ALL-or-nothing: a rank that fell back
             * (export/import/alignment failure) would stage while
             * zero-copy peers read its never-staged areas. Consensus
             * round: everyone posts ok/fail, mode = AND of all. */
            uint64_t ok = zc_ ? 0x5a43u : 0;
            if (hipMemcpy(tt_->area(me_, slot_, 1, 0), &ok, sizeof(ok),
                          hipMemcpyHostToDevice) != hipSuccess) {
                return UCC_ERR_NO_RESOURCE;
            }
            publish(3);
            phase_ = 13;
        }
        if (phase_ == 13) { /* read all verdicts */
            if (!all_ge(3)) {
                return UCC_INPROGRESS;
            }
            for (uint32_t r = 0; r < n_ && zc_; r++) {
                uint64_t ok = 0;
                if (hipMemcpy(&ok, tt_->area(r, slot_, 1, 0), sizeof(ok),
                              hipMemcpyDeviceToHost) != hipSuccess ||
                    ok != 0x5a43u) {
                    zc_ = false;
                }
            }
            if (!zc_) {
                ucc_warn("zero-copy disabled by team consensus, "
                         "using the staged pipeline");
            }
            zc_ready_ = true;
            close_slot();
            begin_use(); /* fresh slot use for the collective itself */
            phase_ = 0;
        }
        if (phase_ == 0) { /* entry: previous use of this slot closed */
            if (!all_ge(0)) {
                return UCC_INPROGRESS;
            }
            done_seq_       = ++tt_->gdone_seq_[slot_];
            ucc_status_t st = enqueue_frags(copy_s(), comp(), false);
            if (st != UCC_OK) {
                return st;
            }
            phase_ = 1;
        }
        /* host-pinned completion flag written by the final kernel once
         * the TEAM's final phase is confirmed (every peer has stopped
         * reading this rank's buffers) — no trailing hipEvent */
        if (*tt_->err_host_ != 0) {
            ucc_error("gated %s timed out waiting for peers",
                      coll_type_name(ct_));
            close_slot();
            return UCC_ERR_TIMED_OUT;
        }
        if (__atomic_load_n(tt_->gdone_host_ + slot_,
                            __ATOMIC_ACQUIRE) < done_seq_) {
            return UCC_INPROGRESS;
        }
        close_slot();
        return UCC_OK;
    }

  private:
    /* SDMA copy-engine alltoall (reference alltoallv_ce.c role): data
     * moves entirely by hipMemcpyAsync pulls over the zero-copy-mapped
     * peer src buffers — the copy engines carry the bytes while the CUs
     * stay free for overlapped compute. Kernels only gate:
     *   [entry signal+reuse waits] -> [all-entered wait] ->
     *   n SDMA pulls on kNumCe streams -> [done signal + team wait]. */
    ucc_status_t enqueue_ce_a2a(hipStream_t comp_s)
    {
        auto *ctx        = (Cdna4TlContext *)tt_->tlc_;
        auto &L          = tt_->gated_launch_;
        const int nblk   = tt_->cfg_.gated_blocks
                               ? tt_->cfg_.gated_blocks
                               : ec_hip::kGatedBlocks;
        const uint64_t B = (uint64_t)nblk;
        const uint32_t p = 0;
        ec_hip::GatedArgs ga{};
        ga.local_flags = tt_->flags_;
        ga.error_word  = tt_->err_host_;
        ga.spin_limit  = tt_->spin_limit();
        ga.nblocks     = nblk;
        ga.pull_wait =
            !Config::instance().get_bool("TL_CDNA4", "PUSH", true);
        ga.rank        = (int)me_;
        ga.nranks      = (int)n_;
        ga.slot        = (int)slot_;
        ga.parity      = (int)p;
        for (uint32_t r = 0; r < n_; r++) {
            ga.peer_flags[r] = tt_->peers_[r].flags;
        }
        ga.len     = 0; /* entry stage is a pure signal */
        ga.n_cells = 0;
        ga.t_sw_reduce  = L[1][slot_][p];
        ga.t_sw_gather  = L[2][slot_][p];
        ga.t_sig_stage  = L[0][slot_][p] + B;
        ga.t_sig_gather = L[2][slot_][p] + B;
        ucc_status_t st = ec_hip::staged_stage(ga, comp_s);
        L[0][slot_][p] += B;
        if (st != UCC_OK) {
            return st;
        }
        /* no peer buffer is written until ITS owner posted (entered) */
        ga.gw_phase      = 0;
        ga.t_gather_wait = L[0][slot_][p];
        st = ec_hip::gated_wait_only(ga, comp_s);
        if (st != UCC_OK) {
            return st;
        }
        HIPCHK(hipEventRecord(ev(8), comp_s));
        bool used[Cdna4TlContext::kNumCe] = {};
        for (uint32_t r = 0; r < n_; r++) {
            /* stagger start peer per rank so SDMA queues don't all hit
             * the same xGMI link first */
            uint32_t rr = (r + me_ + 1) % n_;
            int      si = (int)(r % Cdna4TlContext::kNumCe);
            hipStream_t cs = ctx->ce(si);
            if (!cs) {
                return UCC_ERR_NO_RESOURCE;
            }
            if (!used[si]) {
                HIPCHK(hipStreamWaitEvent(cs, ev(8), 0));
                used[si] = true;
            }
            HIPCHK(hipMemcpyAsync(dbuf_ + (size_t)rr * out_b_,
                                  zc_peer_src_[rr] +
                                      (size_t)me_ * out_b_,
                                  out_b_, hipMemcpyDeviceToDevice, cs));
        }
        for (int si = 0; si < Cdna4TlContext::kNumCe; si++) {
            if (used[si]) {
                HIPCHK(hipEventRecord(ev(9 + si), ctx->ce(si)));
                HIPCHK(hipStreamWaitEvent(comp_s, ev(9 + si), 0));
            }
        }
        /* my pulls done -> publish; complete when every rank's pulls
         * (including reads of MY src) are done */
        ga.t_gather_wait = L[2][slot_][p] + B;
        ga.done_host     = tt_->gdone_host_ + slot_;
        ga.done_seq      = done_seq_;
        st = ec_hip::gated_done(ga, comp_s);
        L[2][slot_][p] += B;
        return st;
    }

    /* CE alltoallv: identical gating to enqueue_ce_a2a; per-peer pull
     * lengths come from the exchanged send tables (len clipped to my
     * recv column, matching the gated a2av semantics). */
    ucc_status_t enqueue_ce_a2av(hipStream_t comp_s)
    {
        auto *ctx        = (Cdna4TlContext *)tt_->tlc_;
        auto &L          = tt_->gated_launch_;
        const int nblk   = tt_->cfg_.gated_blocks
                               ? tt_->cfg_.gated_blocks
                               : ec_hip::kGatedBlocks;
        const uint64_t B = (uint64_t)nblk;
        const uint32_t p = 0;
        ec_hip::GatedArgs ga{};
        ga.local_flags = tt_->flags_;
        ga.error_word  = tt_->err_host_;
        ga.spin_limit  = tt_->spin_limit();
        ga.nblocks     = nblk;
        ga.pull_wait =
            !Config::instance().get_bool("TL_CDNA4", "PUSH", true);
        ga.rank   = (int)me_;
        ga.nranks = (int)n_;
        ga.slot   = (int)slot_;
        ga.parity = (int)p;
        for (uint32_t r = 0; r < n_; r++) {
            ga.peer_flags[r] = tt_->peers_[r].flags;
        }
        ga.len         = 0;
        ga.n_cells     = 0;
        ga.t_sw_reduce  = L[1][slot_][p];
        ga.t_sw_gather  = L[2][slot_][p];
        ga.t_sig_stage  = L[0][slot_][p] + B;
        ga.t_sig_gather = L[2][slot_][p] + B;
        ucc_status_t st = ec_hip::staged_stage(ga, comp_s);
        L[0][slot_][p] += B;
        if (st != UCC_OK) {
            return st;
        }
        ga.gw_phase      = 0;
        ga.t_gather_wait = L[0][slot_][p];
        st = ec_hip::gated_wait_only(ga, comp_s);
        if (st != UCC_OK) {
            return st;
        }
        HIPCHK(hipEventRecord(ev(8), comp_s));
        bool used[Cdna4TlContext::kNumCe] = {};
        for (uint32_t r = 0; r < n_; r++) {
            uint32_t rr  = (r + me_ + 1) % n_;
            uint64_t len = zc_scol_[rr] < rcnt_[rr] ? zc_scol_[rr]
                                                    : rcnt_[rr];
            if (len == 0) {
                continue;
            }
            int         si = (int)(r % Cdna4TlContext::kNumCe);
            hipStream_t cs = ctx->ce(si);
            if (!cs) {
                return UCC_ERR_NO_RESOURCE;
            }
            if (!used[si]) {
                HIPCHK(hipStreamWaitEvent(cs, ev(8), 0));
                used[si] = true;
            }
            HIPCHK(hipMemcpyAsync(dbuf_ + rdsp_[rr],
                                  zc_peer_src_[rr] + zc_sdsp_[rr], len,
                                  hipMemcpyDeviceToDevice, cs));
        }
        for (int si = 0; si < Cdna4TlContext::kNumCe; si++) {
            if (used[si]) {
                HIPCHK(hipEventRecord(ev(9 + si), ctx->ce(si)));
                HIPCHK(hipStreamWaitEvent(comp_s, ev(9 + si), 0));
            }
        }
        ga.t_gather_wait = L[2][slot_][p] + B;
        ga.done_host     = tt_->gdone_host_ + slot_;
        ga.done_seq      = done_seq_;
        st = ec_hip::gated_done(ga, comp_s);
        L[2][slot_][p] += B;
        return st;
    }

    ucc_status_t enqueue_frags(hipStream_t stage_s, hipStream_t comp_s,
                               bool derive)
    {
        if (ct_ == UCC_COLL_TYPE_ALLTOALL && zc_ && zc_ready_ &&
            !derive && tt_->cfg_.ce_alltoall &&
            out_b_ * n_ >= tt_->cfg_.ce_alltoall_min) {
            return enqueue_ce_a2a(comp_s);
        }
        if (ct_ == UCC_COLL_TYPE_ALLTOALLV && zc_ && zc_ready_ &&
            !derive) {
            return enqueue_ce_a2av(comp_s);
        }
        if (zc_ && zc_ready_ && nfrags_ > 1 &&
            Config::instance().get_bool("TL_CDNA4", "ZC_DEFRAG", true)) {
            /* zero-copy reads/writes USER buffers only — no staging
             * area is touched, so the chunk-size bound (the only reason
             * to fragment) does not apply: one fragment = one
             * stage-signal + reduce + gather per iteration instead of
             * 3 x nfrags launches and their gate waits. Symmetric on
             * every rank (zc is team-consensus). */
            gran_   = total_;
            nfrags_ = 1;
        }
        auto &L          = tt_->gated_launch_;
        /* grid size: measured crossover on the 2-proc rig
         * (profiles/rocprof_kernels_r02.md): 64 blocks win below
         * ~160 MiB, 96 above (single-frag zc reduce fills more CUs on
         * big slices). The launch ledger accumulates BLOCK counts, so
         * per-collective grids compose on shared (slot,parity)
         * counters. */
        const int nblk =
            tt_->cfg_.gated_blocks
                ? tt_->cfg_.gated_blocks
                : (nfrags_ == 1 && total_ >= 160u * 1024 * 1024
                       ? 128 /* re-measured after 1-block gates:
                                profiles/rocprof_kernels_r02.md */
                       : ec_hip::kGatedBlocks);
        const uint64_t B = (uint64_t)nblk;
        for (size_t f = 0; f < nfrags_; f++) {
            const uint32_t p   = (uint32_t)(f & 1);
            const size_t   off = f * gran_;
            const size_t   len =
                total_ - off < gran_ ? total_ - off : gran_;
            ec_hip::GatedArgs ga{};
            ga.my_in  = tt_->area(me_, slot_, p, 0);
            ga.my_out = tt_->area(me_, slot_, p, 1);
            for (uint32_t r = 0; r < n_; r++) {
                ga.peer_in[r]    = tt_->area(r, slot_, p, 0);
                ga.peer_out[r]   = tt_->area(r, slot_, p, 1);
                ga.peer_flags[r] = tt_->peers_[r].flags;
            }
            ga.local_flags = tt_->flags_;
            ga.error_word  = tt_->err_host_;
            ga.spin_limit  = tt_->spin_limit();
            ga.nblocks     = nblk;
            ga.pull_wait   = !Config::instance().get_bool(
                "TL_CDNA4", "PUSH", true);
            ga.len         = len;
            ga.rank        = (int)me_;
            ga.nranks      = (int)n_;
            ga.slot        = (int)slot_;
            ga.parity      = (int)p;
            ga.dt          = dt_;
            ga.op          = op_;
            ga.alpha       = alpha_;
            if (derive) {
                const bool is_rs =
                    ct_ == UCC_COLL_TYPE_REDUCE_SCATTER ||
                    ct_ == UCC_COLL_TYPE_REDUCE_SCATTERV;
                ga.derive     = 1;
                ga.pp         = ct_ == UCC_COLL_TYPE_ALLREDUCE ? 3 : 2;
                ga.has_reduce = ct_ == UCC_COLL_TYPE_ALLREDUCE || is_rs;
                ga.has_gather = !is_rs;
            }
            ucc_status_t st = UCC_OK;
            switch (ct_) {
            case UCC_COLL_TYPE_ALLREDUCE: {
                /* stage frag -> reduce my slice -> gather all slices.
                 * zero-copy: stage is a pure signal (len 0) and reduce
                 * reads peers' USER src directly over xGMI */
                ga.src = sbuf_ + off;
                ga.dst = dbuf_ + off;
                size_t per = (len / n_) & ~(size_t)255;
                for (uint32_t r = 0; r < n_; r++) {
                    ga.slice_b[r] = (uint64_t)r * per;
                    ga.slice_e[r] =
                        r == n_ - 1 ? len : (uint64_t)(r + 1) * per;
                }
                ga.sl_b = ga.slice_b[me_];
                ga.sl_e = ga.slice_e[me_];
                if (zc_ && zc_ready_) {
                    /* reduce reads peers' src AND writes every rank's
                     * dst directly; gather becomes a pure wait */
                    ga.len      = 0;
                    ga.zc_write = 1;
                    for (uint32_t r = 0; r < n_; r++) {
                        ga.peer_in[r] = zc_peer_src_[r] + off;
                        ga.peer_out[r] = zc_peer_dst_[r] + off + ga.sl_b;
                    }
                }
                /* zero-copy: stage and gather are pure signal/wait —
                 * a full grid of spinning blocks would only steal CUs
                 * from the reduce; the cumulative-block ledger lets
                 * them launch with ONE block (derive mode keeps the
                 * uniform grid its device-side arithmetic needs). */
                const uint64_t Bs =
                    (!derive && ga.zc_write) ? 1 : B;
                const uint64_t Bg = Bs;
                ga.t_sw_reduce   = L[1][slot_][p];
                ga.t_sw_gather   = L[2][slot_][p];
                ga.t_prev_gather = L[2][slot_][p];
                ga.t_stage       = L[0][slot_][p] + Bs;
                ga.gw_phase      = 1;
                ga.t_gather_wait = L[1][slot_][p] + B;
                ga.t_sig_stage   = L[0][slot_][p] + Bs;
                ga.t_sig_reduce  = L[1][slot_][p] + B;
                ga.t_sig_gather  = L[2][slot_][p] + Bg;
                {
                    ec_hip::GatedArgs gs = ga;
                    gs.nblocks           = (int)Bs;
                    st = ec_hip::staged_stage(gs, stage_s);
                }
                if (!derive) { L[0][slot_][p] += Bs; }
                if (st == UCC_OK) {
                    st = ec_hip::staged_reduce(ga, comp_s);
                    if (!derive) { L[1][slot_][p] += B; }
                }
                if (st == UCC_OK) {
                    if (ga.zc_write) {
                        /* dst already written by every rank's reduce:
                         * the gather launch is a pure completion wait */
                        for (uint32_t r = 0; r < n_; r++) {
                            ga.slice_e[r] = ga.slice_b[r];
                        }
                    }
                    ec_hip::GatedArgs gg = ga;
                    gg.nblocks           = (int)Bg;
                    if (!derive && f == nfrags_ - 1) {
                        gg.done_host = tt_->gdone_host_ + slot_;
                        gg.done_seq  = done_seq_;
                    }
                    st = ec_hip::staged_gather(gg, comp_s);
                    if (!derive) { L[2][slot_][p] += Bg; }
                }
                break;
            }
            case UCC_COLL_TYPE_REDUCE_SCATTERV:
            case UCC_COLL_TYPE_REDUCE_SCATTER: {
                /* stage frag of the whole source; reduce ONLY the
                 * intersection with my output slice, directly into my
                 * user dst (no gather phase) */
                ga.src = sbuf_ + off;
                size_t s0, s1;
                if (ct_ == UCC_COLL_TYPE_REDUCE_SCATTERV) {
                    s0 = dsp_[me_];
                    s1 = s0 + cnt_[me_];
                } else {
                    s0 = (size_t)me_ * out_b_;
                    s1 = s0 + out_b_;
                }
                size_t b = s0 > off ? s0 : off;
                size_t e = s1 < off + len ? s1 : off + len;
                ga.sl_b   = b > e ? 0 : b - off;
                ga.sl_e   = b > e ? 0 : e - off;
                ga.my_out = dbuf_ + (b >= s0 && b <= s1 ? b - s0 : 0);
                if (zc_ && zc_ready_) {
                    /* reduce reads peers' user src directly */
                    ga.len = 0;
                    for (uint32_t r = 0; r < n_; r++) {
                        ga.peer_in[r] = zc_peer_src_[r] + off;
                    }
                }
                const uint64_t Bs =
                    (!derive && zc_ && zc_ready_) ? 1 : B;
                ga.t_sw_reduce   = L[1][slot_][p];
                ga.t_sw_gather   = L[2][slot_][p];
                ga.t_prev_gather = 0;
                ga.t_stage       = L[0][slot_][p] + Bs;
                ga.t_sig_stage   = L[0][slot_][p] + Bs;
                ga.t_sig_reduce  = L[1][slot_][p] + B;
                {
                    ec_hip::GatedArgs gs = ga;
                    gs.nblocks           = (int)Bs;
                    st = ec_hip::staged_stage(gs, stage_s);
                }
                if (!derive) { L[0][slot_][p] += Bs; }
                if (st == UCC_OK) {
                    st = ec_hip::staged_reduce(ga, comp_s);
                    if (!derive) { L[1][slot_][p] += B; }
                }
                if (st == UCC_OK && !derive && f == nfrags_ - 1) {
                    /* completion = the TEAM's reduces done: a peer's
                     * zero-copy reduce may still be reading MY src
                     * when my own reduce retires */
                    ec_hip::GatedArgs gw = ga;
                    gw.gw_phase          = 1;
                    gw.t_gather_wait     = L[1][slot_][p];
                    gw.done_host         = tt_->gdone_host_ + slot_;
                    gw.done_seq          = done_seq_;
                    st = ec_hip::gated_wait_only(gw, comp_s);
                }
                break;
            }
            case UCC_COLL_TYPE_ALLGATHERV:
            case UCC_COLL_TYPE_ALLGATHER: {
                /* stage my block frag; gather reads peers' STAGED data
                 * (in areas) into dst block positions. v: per-rank
                 * lengths clipped to each block's count */
                size_t mylen =
                    off >= cnt_[me_] ? 0
                    : cnt_[me_] - off < gran_ ? cnt_[me_] - off
                                              : gran_;
                ga.src = sbuf_ + off;
                ga.len = mylen;
                ga.dst = dbuf_;
                const bool agzc = zc_ && zc_ready_;
                if (agzc) {
                    ga.len = 0; /* stage = pure signal */
                }
                for (uint32_t r = 0; r < n_; r++) {
                    ga.peer_out[r] =
                        agzc ? (const void *)(zc_peer_src_[r] + off)
                             : (const void *)tt_->area(r, slot_, p, 0);
                    size_t l = off >= cnt_[r] ? 0
                               : cnt_[r] - off < gran_ ? cnt_[r] - off
                                                       : gran_;
                    ga.slice_b[r] = dsp_[r] + off;
                    ga.slice_e[r] = ga.slice_b[r] + l;
                }
                const uint64_t Bs =
                    (!derive && agzc) ? 1 : B;
                ga.t_sw_reduce   = L[1][slot_][p];
                ga.t_sw_gather   = L[2][slot_][p];
                ga.gw_phase      = 0;
                ga.t_gather_wait = L[0][slot_][p] + Bs;
                ga.t_sig_stage   = L[0][slot_][p] + Bs;
                ga.t_sig_gather  = L[2][slot_][p] + B;
                {
                    ec_hip::GatedArgs gs = ga;
                    gs.nblocks           = (int)Bs;
                    st = ec_hip::staged_stage(gs, stage_s);
                }
                if (!derive) { L[0][slot_][p] += Bs; }
                if (st == UCC_OK) {
                    st = ec_hip::staged_gather(ga, comp_s);
                    if (!derive) { L[2][slot_][p] += B; }
                }
                if (st == UCC_OK && !derive && f == nfrags_ - 1) {
                    ec_hip::GatedArgs gw = ga;
                    gw.gw_phase          = 2;
                    gw.t_gather_wait     = L[2][slot_][p];
                    gw.done_host         = tt_->gdone_host_ + slot_;
                    gw.done_seq          = done_seq_;
                    st = ec_hip::gated_wait_only(gw, comp_s);
                }
                break;
            }
            case UCC_COLL_TYPE_ALLTOALLV: {
                /* per-dest cells with per-pair lengths: send cells clip
                 * to my send row, gather slices clip to my recv column.
                 * Fragment count is the exchanged GLOBAL max, so every
                 * rank launches identically (ledger stays in sync). */
                ga.src     = sbuf_;
                ga.dst     = dbuf_;
                ga.n_cells = (int)n_;
                for (uint32_t r = 0; r < n_; r++) {
                    size_t sl = off >= cnt_[r] ? 0
                                : cnt_[r] - off < gran_ ? cnt_[r] - off
                                                        : gran_;
                    size_t rl = off >= rcnt_[r] ? 0
                                : rcnt_[r] - off < gran_
                                    ? rcnt_[r] - off
                                    : gran_;
                    ga.c_src_off[r] = dsp_[r] + off;
                    ga.c_dst_off[r] = (uint64_t)r * cell_;
                    ga.c_len[r]     = sl;
                    ga.peer_out[r] =
                        (const void *)(tt_->area(r, slot_, p, 0) +
                                       me_ * cell_);
                    ga.slice_b[r] = rdsp_[r] + off;
                    ga.slice_e[r] = ga.slice_b[r] + rl;
                }
                ga.t_sw_reduce   = L[1][slot_][p];
                ga.t_sw_gather   = L[2][slot_][p];
                ga.gw_phase      = 0;
                ga.t_gather_wait = L[0][slot_][p] + B;
                ga.t_sig_stage   = L[0][slot_][p] + B;
                ga.t_sig_gather  = L[2][slot_][p] + B;
                st = ec_hip::staged_stage(ga, stage_s);
                if (!derive) { L[0][slot_][p] += B; }
                if (st == UCC_OK) {
                    st = ec_hip::staged_gather(ga, comp_s);
                    if (!derive) { L[2][slot_][p] += B; }
                }
                if (st == UCC_OK && !derive && f == nfrags_ - 1) {
                    ec_hip::GatedArgs gw = ga;
                    gw.gw_phase          = 2;
                    gw.t_gather_wait     = L[2][slot_][p];
                    gw.done_host         = tt_->gdone_host_ + slot_;
                    gw.done_seq          = done_seq_;
                    st = ec_hip::gated_wait_only(gw, comp_s);
                }
                break;
            }
            case UCC_COLL_TYPE_ALLTOALL: {
                /* stage per-dest cells; gather my cell from each peer.
                 * zero-copy: gather reads peers' USER src at my block's
                 * offset directly; stage = pure signal */
                ga.src     = sbuf_;
                ga.dst     = dbuf_;
                const bool a2zc = zc_ && zc_ready_;
                ga.n_cells = a2zc ? 0 : (int)n_;
                if (a2zc) {
                    ga.len = 0; /* stage = pure signal */
                }
                for (uint32_t r = 0; r < n_; r++) {
                    ga.c_src_off[r] = (uint64_t)r * out_b_ + off;
                    ga.c_dst_off[r] = (uint64_t)r * cell_;
                    ga.c_len[r]     = a2zc ? 0 : len;
                    ga.peer_out[r] =
                        a2zc ? (const void *)(zc_peer_src_[r] +
                                              me_ * out_b_ + off)
                             : (const void *)(tt_->area(r, slot_, p, 0) +
                                              me_ * cell_);
                    ga.slice_b[r] = (uint64_t)r * out_b_ + off;
                    ga.slice_e[r] = ga.slice_b[r] + len;
                }
                const uint64_t Bs =
                    (!derive && a2zc) ? 1 : B;
                ga.t_sw_reduce   = L[1][slot_][p];
                ga.t_sw_gather   = L[2][slot_][p];
                ga.gw_phase      = 0;
                ga.t_gather_wait = L[0][slot_][p] + Bs;
                ga.t_sig_stage   = L[0][slot_][p] + Bs;
                ga.t_sig_gather  = L[2][slot_][p] + B;
                {
                    ec_hip::GatedArgs gs = ga;
                    gs.nblocks           = (int)Bs;
                    st = ec_hip::staged_stage(gs, stage_s);
                }
                if (!derive) { L[0][slot_][p] += Bs; }
                if (st == UCC_OK) {
                    st = ec_hip::staged_gather(ga, comp_s);
                    if (!derive) { L[2][slot_][p] += B; }
                }
                if (st == UCC_OK && !derive && f == nfrags_ - 1) {
                    ec_hip::GatedArgs gw = ga;
                    gw.gw_phase          = 2;
                    gw.t_gather_wait     = L[2][slot_][p];
                    gw.done_host         = tt_->gdone_host_ + slot_;
                    gw.done_seq          = done_seq_;
                    st = ec_hip::gated_wait_only(gw, comp_s);
                }
                break;
            }
            default:
                return UCC_ERR_NOT_SUPPORTED;
            }
            if (st != UCC_OK) {
                return st;
            }
        }
        return UCC_OK;
    }

    ucc_coll_type_t    ct_    = UCC_COLL_TYPE_ALLREDUCE;
    ucc_datatype_t     dt_    = UCC_DT_FLOAT32;
    ucc_reduction_op_t op_    = UCC_OP_SUM;
    float              alpha_ = 1.0f;
    size_t             dtsz_ = 4, total_ = 0, nfrags_ = 0;
    size_t             gran_ = 0, cell_ = 0, out_b_ = 0;
    const uint8_t     *sbuf_ = nullptr;
    uint8_t           *dbuf_ = nullptr;
    std::vector<size_t> cnt_, dsp_;   /* send row bytes (ag/rs/a2av)  */
    std::vector<size_t> rcnt_, rdsp_; /* recv column bytes (a2av)     */
    bool                a2av_ready_ = false; /* global max exchanged  */
    uint64_t            a2av_gsum_  = 0;     /* global send-byte sum  */
    uint64_t            done_seq_   = 0;     /* host completion seq   */
    int                pslot_ = -1;
    /* zero-copy persistent allreduce: peers' USER src buffers mapped
     * via HIP-IPC (handles exchanged through the scratch channel at
     * first post) — the reduce phase reads them directly, the stage
     * kernel degenerates to a pure signal (no staging copy). */
    bool               zc_ = false, zc_ready_ = false;
    const uint8_t     *zc_peer_src_[ec_hip::kMaxRanks] = {};
    uint8_t           *zc_peer_dst_[ec_hip::kMaxRanks] = {};
    /* CE a2av: peer r's send-to-me byte count / displacement */
    uint64_t           zc_scol_[ec_hip::kMaxRanks] = {};
    uint64_t           zc_sdsp_[ec_hip::kMaxRanks] = {};

  public:
    ~GatedCollTask() override
    {
        if (pslot_ >= 0) {
            tt_->free_pslot(pslot_);
        }
        /* zero-copy imports stay in the team cache (ipc_open_cached) */
    }
};

/* ------------------------------------------------------------ scoring  */
class Cdna4Tl final : public Tl {
  public:
    const char *name() const override { return "cdna4"; }
    int         default_score() const override { return 80; }

    TlContext *context_create(Context *ctx) override
    {
        auto &cfg = Config::instance();
        cfg.declare("TL_CDNA4", "ENABLE", "1",
                    "enable the native xGMI device transport");
        cfg.declare("TL_CDNA4", "MAX_CONCURRENT", "4",
                    "rotating in-flight collective slots");
        cfg.declare("TL_CDNA4", "PERSISTENT_SLOTS", "2",
                    "dedicated slots for persistent triggered colls");
        cfg.declare("TL_CDNA4", "CHUNK_SIZE", "32m",
                    "staging fragment bytes per slot area");
        cfg.declare("TL_CDNA4", "FUSED_MAX", "4m",
                    "max msg bytes for the fused single-kernel allreduce "
                    "(measured crossover vs the gated pipeline, "
                    "profiles/rocprof_kernels_r02.md)");
        cfg.declare("TL_CDNA4", "GATED", "1",
                    "device-gated pipeline for large colls");
        cfg.declare("TL_CDNA4", "SPIN_LIMIT", "0",
                    "device spin bound override (0 = default ~seconds)");
        cfg.declare("TL_CDNA4", "GATED_BLOCKS", "0",
                    "gated-pipeline kernel grid size (workgroups of 256; "
                    "0 = built-in default; must match on every rank)");
        cfg.declare("TL_CDNA4", "STAGE_SDMA", "0",
                    "stage fragments with hipMemcpyAsync (SDMA) "
                    "instead of a copy kernel");
        cfg.declare("TL_CDNA4", "CE_ALLTOALL", "1",
                    "move zero-copy alltoall data on SDMA copy engines "
                    "(hipMemcpyAsync) instead of gather kernels");
        cfg.declare("TL_CDNA4", "ZC_DEFRAG", "1",
                    "single-fragment zero-copy collectives (debug off)");
        cfg.declare("TL_CDNA4", "PUSH", "1",
                    "push-model gated flags; 0 = remote-poll fallback "
                    "(debug)");
        cfg.declare("TL_CDNA4", "CE_ALLTOALL_MIN", "192m",
                    "min total message bytes for the SDMA alltoall "
                    "(measured crossover vs the gather kernel, "
                    "profiles/rocprof_kernels_r02.md)");
        if (!cfg.get_bool("TL_CDNA4", "ENABLE", true) ||
            !mc::hip_available()) {
            return nullptr;
        }
        int dev = 0;
        if (hipGetDevice(&dev) != hipSuccess) {
            return nullptr;
        }
        ctx->proc.device = dev;
        return new Cdna4TlContext(ctx, this, dev);
    }

    TlTeam *team_create(TlContext *tlc, Team *team) override
    {
        if (team->size < 2 || team->size > (uint32_t)kMaxRanks ||
            !team->all_same_node() || !team->all_have_device()) {
            return nullptr;
        }
        auto    &cfg = Config::instance();
        Cdna4Cfg c;
        c.nslots    = (uint32_t)cfg.get_int("TL_CDNA4", "MAX_CONCURRENT", 4);
        c.npers  = (uint32_t)cfg.get_int("TL_CDNA4", "PERSISTENT_SLOTS", 2);
        if (c.nslots + c.npers > 8) { /* flags layout: 8 slots max */
            c.npers = c.nslots < 8 ? 8 - c.nslots : 0;
        }
        c.chunk     = cfg.get_size("TL_CDNA4", "CHUNK_SIZE", 32 * 1024 * 1024);
        c.spin_limit =
            (uint64_t)cfg.get_int("TL_CDNA4", "SPIN_LIMIT", 0);
        c.fused_max = cfg.get_size("TL_CDNA4", "FUSED_MAX",
                                   4 * 1024 * 1024);
        c.gated_blocks =
            (int)cfg.get_int("TL_CDNA4", "GATED_BLOCKS", 0);
        if (c.gated_blocks < 0) {
            c.gated_blocks = 0;
        }
        if (c.gated_blocks > ec_hip::kGatedMaxBlocks) {
            c.gated_blocks = ec_hip::kGatedMaxBlocks;
        }
        c.ce_alltoall = cfg.get_bool("TL_CDNA4", "CE_ALLTOALL", true);
        c.ce_alltoall_min =
            cfg.get_size("TL_CDNA4", "CE_ALLTOALL_MIN",
                         192 * 1024 * 1024);
        return new Cdna4TlTeam(tlc, team, c);
    }
};

static Cdna4Tl g_cdna4_tl;

Tl *Cdna4TlContext::iface() { return &g_cdna4_tl; }

void Cdna4TlTeam::get_scores(Team *team, ScoreMap &map)
{
    (void)team;
    Cdna4TlTeam *self = this;
    auto add_gated = [&](ucc_coll_type_t ct, size_t lo, size_t hi,
                         int score) {
        ScoreRange r;
        r.start    = lo;
        r.end      = hi;
        r.score    = score;
        r.tl_name  = "cdna4";
        r.alg_name = "gated_pipeline";
        r.init     = [self](const ucc_coll_args_t &args, Team *t,
                        Task **task) -> ucc_status_t {
            if (args.coll_type == UCC_COLL_TYPE_ALLREDUCE ||
                args.coll_type == UCC_COLL_TYPE_REDUCE_SCATTER ||
                args.coll_type == UCC_COLL_TYPE_REDUCE_SCATTERV) {
                ucc_datatype_t gdt =
                    args.coll_type == UCC_COLL_TYPE_REDUCE_SCATTERV
                        ? args.dst.info_v.datatype
                        : args.dst.info.datatype;
                if (!ec_hip::op_supported(gdt, args.op)) {
                    return UCC_ERR_NOT_SUPPORTED;
                }
            }
            /* in-process multi-rank jigs can serialize spinning kernels
             * on one HW queue: host-gated staged path is the safe
             * fallback there (same reasoning as the fused kernel). */
            int same_proc = 0;
            for (auto &p : t->procs) {
                same_proc += (p.pid == t->ctx->proc.pid);
            }
            if (same_proc > 1) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            *task = new GatedCollTask(t->ctx, self, args);
            return UCC_OK;
        };
        for (auto mt : {UCC_MEMORY_TYPE_CUDA, UCC_MEMORY_TYPE_CUDA_MANAGED}) {
            map.add(ct, mt, r);
        }
    };
    auto add = [&](ucc_coll_type_t ct, size_t lo, size_t hi, int score,
                   const char *alg, bool fused) {
        ScoreRange r;
        r.start    = lo;
        r.end      = hi;
        r.score    = score;
        r.tl_name  = "cdna4";
        r.alg_name = alg;
        r.init     = [self, fused](const ucc_coll_args_t &args, Team *t,
                               Task **task) -> ucc_status_t {
            /* reductions need a supported dtype x op on device */
            if (args.coll_type == UCC_COLL_TYPE_ALLREDUCE ||
                args.coll_type == UCC_COLL_TYPE_REDUCE ||
                args.coll_type == UCC_COLL_TYPE_REDUCE_SCATTER ||
                args.coll_type == UCC_COLL_TYPE_REDUCE_SCATTERV) {
                ucc_datatype_t dt =
                    args.coll_type == UCC_COLL_TYPE_REDUCE_SCATTERV
                        ? args.dst.info_v.datatype
                        : args.dst.info.datatype;
                if (args.coll_type == UCC_COLL_TYPE_REDUCE) {
                    dt = args.src.info.datatype;
                }
                if (!ec_hip::op_supported(dt, args.op)) {
                    return UCC_ERR_NOT_SUPPORTED;
                }
            }
            if (fused) {
                /* The fused kernel spin-waits for every peer's kernel.
                 * With several ranks of one process on one GPU (the
                 * in-process jig) their streams can share a HW queue and
                 * event barrier packets serialize the dispatches ->
                 * deadlock. One process per GPU (production) is safe;
                 * same-process multi-rank falls back to the host-gated
                 * staged path. */
                int same_proc = 0;
                for (auto &p : t->procs) {
                    same_proc += (p.pid == t->ctx->proc.pid);
                }
                if (same_proc > 1) {
                    return UCC_ERR_NOT_SUPPORTED;
                }
                *task = new FusedAllreduceTask(t->ctx, self, args);
            } else {
                *task = new StagedTask(t->ctx, self, args);
            }
            return UCC_OK;
        };
        for (auto mt : {UCC_MEMORY_TYPE_CUDA, UCC_MEMORY_TYPE_CUDA_MANAGED}) {
            map.add(ct, mt, r);
        }
    };
    add(UCC_COLL_TYPE_ALLREDUCE, 0, cfg_.fused_max, 100, "fused", true);
    if (Config::instance().get_bool("TL_CDNA4", "GATED", true)) {
        add_gated(UCC_COLL_TYPE_ALLREDUCE, cfg_.fused_max + 1, SIZE_MAX,
                  90);
        add_gated(UCC_COLL_TYPE_REDUCE_SCATTER, 0, SIZE_MAX, 90);
        add_gated(UCC_COLL_TYPE_REDUCE_SCATTERV, 0, SIZE_MAX, 90);
        add_gated(UCC_COLL_TYPE_ALLGATHER, 0, SIZE_MAX, 90);
        add_gated(UCC_COLL_TYPE_ALLGATHERV, 0, SIZE_MAX, 90);
        add_gated(UCC_COLL_TYPE_ALLTOALL, 0, SIZE_MAX, 90);
        add_gated(UCC_COLL_TYPE_ALLTOALLV, 0, SIZE_MAX, 90);
    }
    add(UCC_COLL_TYPE_ALLREDUCE, 0, SIZE_MAX, 80, "staged_linear", false);
    add(UCC_COLL_TYPE_ALLGATHER, 0, SIZE_MAX, 80, "staged_linear", false);
    add(UCC_COLL_TYPE_ALLGATHERV, 0, SIZE_MAX, 80, "staged_linear", false);
    add(UCC_COLL_TYPE_REDUCE_SCATTER, 0, SIZE_MAX, 80, "staged_linear",
        false);
    add(UCC_COLL_TYPE_REDUCE_SCATTERV, 0, SIZE_MAX, 80, "staged_linear",
        false);
    add(UCC_COLL_TYPE_BCAST, 0, SIZE_MAX, 80, "staged_linear", false);
    add(UCC_COLL_TYPE_REDUCE, 0, SIZE_MAX, 80, "staged_linear", false);
    add(UCC_COLL_TYPE_ALLTOALL, 0, SIZE_MAX, 80, "staged_linear", false);
    add(UCC_COLL_TYPE_ALLTOALLV, 0, SIZE_MAX, 80, "staged_linear", false);
    add(UCC_COLL_TYPE_GATHER, 0, SIZE_MAX, 80, "staged_linear", false);
    add(UCC_COLL_TYPE_GATHERV, 0, SIZE_MAX, 80, "staged_linear", false);
    add(UCC_COLL_TYPE_SCATTER, 0, SIZE_MAX, 80, "staged_linear", false);
    add(UCC_COLL_TYPE_SCATTERV, 0, SIZE_MAX, 80, "staged_linear", false);
}

} // namespace

Tl *tl_cdna4_iface() { return &g_cdna4_tl; }

} // namespace ucc
