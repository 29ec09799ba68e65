/* TL "tcp": host-memory inter-node transport — the tl/ucp role
 *  without UCX: tagged p2p over nonblocking full-mesh TCP
 * (p2p.h) with re-derived pattern algorithms:
 *   allreduce       recursive doubling w/ non-power-of-2 fold
 *                   (tl/ucp allreduce knomial family role)
 *   bcast/reduce    binomial tree
 *   (all)gather(v)  ring / linear-to-root
 *   alltoall(v)     pairwise exchange (throttle-free small-n)
 *   reduce_scatter(v) reduce + scatter schedule
 *   barrier/fanin/fanout  binomial fanin-fanout
 * Scores below tl/shm (5 vs 50): same-node teams stay on shared memory;
 * TCP serves teams spanning nodes (team create declines nothing — any
 * host-memory team works, it is the universal fallback).
 *
 * Team bootstrap: listen socket per context; {ip, port} published through
 * the team's combined OOB exchange; mesh built lazily in create_test
 * (lower rank accepts, higher connects, 4-byte rank hello). */
#include <arpa/inet.h>
#include <errno.h>
#include <fcntl.h>
#include <ifaddrs.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <algorithm>

#include "../../core/core.h"
#include "../../ec/ec_cpu.h"
#include "p2p.h"

namespace ucc {
namespace tcp {

/* ------------------------------------------------------- Conn progress */
static bool io_fatal(ssize_t r)
{
    return r == 0 || (r < 0 && errno != EAGAIN && errno != EWOULDBLOCK &&
                      errno != EINTR);
}

void Conn::match_header()
{
    /* find first posted recv with this tag */
    for (auto it = recvq.begin(); it != recvq.end(); ++it) {
        if ((*it)->tag == hdr.tag) {
            cur_recv = *it;
            recvq.erase(it);
            return;
        }
    }
    unexp.push_back(UnexpMsg{hdr.tag, std::vector<uint8_t>(hdr.len)});
    cur_un = &unexp.back().data;
}

void Conn::progress()
{
    if (fd < 0) {
        return;
    }
    /* ---- sends */
    while (!sendq.empty()) {
        SendOp *op = sendq.front();
        if (op->off < sizeof(WireHdr)) {
            WireHdr h{op->tag, op->len};
            ssize_t r = ::send(fd, (uint8_t *)&h + op->off,
                               sizeof(h) - op->off, MSG_NOSIGNAL);
            if (r <= 0) {
                if (io_fatal(r)) {
                    ucc_error("tcp send hdr failed: %s", strerror(errno));
                }
                break;
            }
            op->off += (size_t)r;
            if (op->off < sizeof(WireHdr)) {
                break;
            }
        }
        size_t poff = op->off - sizeof(WireHdr);
        while (poff < op->len) {
            ssize_t r = ::send(fd, op->buf + poff, op->len - poff,
                               MSG_NOSIGNAL);
            if (r <= 0) {
                break;
            }
            poff += (size_t)r;
            op->off += (size_t)r;
        }
        if (poff < op->len) {
            break;
        }
        op->done = true;
        sendq.pop_front();
    }
    /* ---- recvs */
    while (true) {
        if (hdr_got < sizeof(WireHdr)) {
            ssize_t r = ::recv(fd, (uint8_t *)&hdr + hdr_got,
                               sizeof(hdr) - hdr_got, 0);
            if (r <= 0) {
                if (io_fatal(r)) {
                    /* peer gone: fatal only if mid-message */
                }
                return;
            }
            hdr_got += (size_t)r;
            if (hdr_got < sizeof(WireHdr)) {
                return;
            }
            pay_got = 0;
            match_header();
        }
        uint8_t *dst =
            cur_recv ? cur_recv->buf : (cur_un ? cur_un->data() : nullptr);
        size_t want = hdr.len;
        if (cur_recv && cur_recv->len < want) {
            want = cur_recv->len; /* truncate (caller sized exactly) */
        }
        while (pay_got < hdr.len) {
            uint8_t  sink[4096];
            uint8_t *p;
            size_t   room;
            if (dst && pay_got < want) {
                p    = dst + pay_got;
                room = want - pay_got;
            } else {
                p    = sink;
                room = std::min(hdr.len - pay_got, sizeof(sink));
            }
            ssize_t r = ::recv(fd, p, room, 0);
            if (r <= 0) {
                return;
            }
            pay_got += (size_t)r;
        }
        if (cur_recv) {
            cur_recv->done = true;
        }
        if (cur_un) {
            /* mark the queue entry complete (take_unexp eligibility) */
            for (auto &u : unexp) {
                if (&u.data == cur_un) {
                    u.done = true;
                    break;
                }
            }
        }
        cur_recv = nullptr;
        cur_un   = nullptr;
        hdr_got  = 0;
    }
}

/* --------------------------------------------------------- context/team */
struct TcpAddr {
    uint32_t ip;
    uint16_t port;
    uint16_t pad;
};

static uint32_t local_ip()
{
    uint32_t        ip = htonl(INADDR_LOOPBACK);
    struct ifaddrs *ifs = nullptr;
    if (getifaddrs(&ifs) == 0) {
        for (auto *i = ifs; i; i = i->ifa_next) {
            if (i->ifa_addr && i->ifa_addr->sa_family == AF_INET) {
                auto *sin = (struct sockaddr_in *)i->ifa_addr;
                if (ntohl(sin->sin_addr.s_addr) != INADDR_LOOPBACK) {
                    ip = sin->sin_addr.s_addr;
                    break;
                }
            }
        }
        freeifaddrs(ifs);
    }
    return ip;
}

static void set_nonblock(int fd)
{
    fcntl(fd, F_SETFL, fcntl(fd, F_GETFL, 0) | O_NONBLOCK);
    int one = 1;
    setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
    /* large-message throughput: the ~208KB kernel default stalls the
     * single-pass progress loop once per buffer-full (reference UCX
     * sockcm sizes buffers up the same way); knob in bytes, 0 keeps
     * the kernel default */
    int sb = (int)Config::instance().get_size("TL_TCP", "SOCKBUF",
                                              4 * 1024 * 1024);
    if (sb > 0) {
        setsockopt(fd, SOL_SOCKET, SO_SNDBUF, &sb, sizeof(sb));
        setsockopt(fd, SOL_SOCKET, SO_RCVBUF, &sb, sizeof(sb));
    }
}

class TcpTl;

class TcpTlContext final : public TlContext {
  public:
    TcpTlContext(Context *ctx, Tl *tl) : TlContext(ctx), tl_(tl)
    {
        lfd_ = socket(AF_INET, SOCK_STREAM, 0);
        if (lfd_ < 0) {
            return;
        }
        struct sockaddr_in a{};
        a.sin_family      = AF_INET;
        a.sin_addr.s_addr = htonl(INADDR_ANY);
        a.sin_port        = 0;
        if (bind(lfd_, (struct sockaddr *)&a, sizeof(a)) != 0 ||
            listen(lfd_, 64) != 0) {
            close(lfd_);
            lfd_ = -1;
            return;
        }
        socklen_t al = sizeof(a);
        getsockname(lfd_, (struct sockaddr *)&a, &al);
        port_ = ntohs(a.sin_port);
        ip_   = local_ip();
        set_nonblock(lfd_);
    }
    ~TcpTlContext() override
    {
        if (lfd_ >= 0) {
            close(lfd_);
        }
    }
    Tl *iface() override;

    Tl      *tl_;
    int      lfd_ = -1;
    uint32_t ip_  = 0;
    uint16_t port_ = 0;
    /* accepted-but-unclaimed connections: the ONE listener serves every
     * team of this context (parent + hier sub-teams), so accepts are
     * pooled here and claimed by uid from each team's create_test —
     * never closed on mismatch (they belong to another team). */
    struct Pending {
        int      fd;
        bool     have_hello = false;
        uint32_t uid = 0, rank = 0;
    };
    std::vector<Pending> pending_;

    void poll_accepts()
    {
        while (true) {
            int fd = accept(lfd_, nullptr, nullptr);
            if (fd < 0) {
                break;
            }
            set_nonblock(fd);
            pending_.push_back(Pending{fd});
        }
        for (auto &p : pending_) {
            if (p.have_hello || p.fd < 0) {
                continue;
            }
            uint32_t hello[2];
            ssize_t  r = ::recv(p.fd, hello, sizeof(hello), MSG_PEEK);
            if (r == (ssize_t)sizeof(hello)) {
                ::recv(p.fd, hello, sizeof(hello), 0);
                p.have_hello = true;
                p.uid        = hello[0];
                p.rank       = hello[1];
            } else if (r == 0 || (r < 0 && errno != EAGAIN &&
                                  errno != EWOULDBLOCK)) {
                close(p.fd);
                p.fd = -1;
            }
        }
        pending_.erase(std::remove_if(pending_.begin(), pending_.end(),
                                      [](const Pending &p) {
                                          return p.fd < 0;
                                      }),
                       pending_.end());
    }

    /* claim a pending connection for this team uid; fills fd+rank */
    bool claim(uint32_t uid, uint32_t n, int *fd_out, uint32_t *rank_out)
    {
        for (auto &p : pending_) {
            if (p.fd >= 0 && p.have_hello && p.uid == uid && p.rank < n) {
                *fd_out   = p.fd;
                *rank_out = p.rank;
                p.fd      = -1;
                return true;
            }
        }
        return false;
    }
};

class TcpTlTeam final : public TlTeam {
  public:
    TcpTlTeam(TlContext *tlc, Team *team) : TlTeam(tlc, team)
    {
        conns_.resize(team->size);
    }
    ~TcpTlTeam() override
    {
        for (auto &c : conns_) {
            if (c.fd >= 0) {
                close(c.fd);
            }
        }
    }

    size_t exchg_size() override { return sizeof(TcpAddr); }

    void exchg_pack(void *buf) override
    {
        auto   *c = (TcpTlContext *)tlc_;
        TcpAddr a{c->ip_, c->port_, 0};
        memcpy(buf, &a, sizeof(a));
    }

    ucc_status_t exchg_unpack(const void *all, size_t stride) override
    {
        addrs_.resize(team_->size);
        for (uint32_t r = 0; r < team_->size; r++) {
            memcpy(&addrs_[r], (const uint8_t *)all + r * stride,
                   sizeof(TcpAddr));
        }
        return UCC_OK;
    }

    ucc_status_t create_test() override
    {
        auto          *c  = (TcpTlContext *)tlc_;
        const uint32_t me = team_->rank;
        const uint32_t n  = team_->size;
        /* accept incoming (peers with rank > me connect to me); the
         * context pools accepts for all teams, claimed by uid */
        c->poll_accepts();
        int      cfd;
        uint32_t crank;
        while (c->claim((uint32_t)team_->team_uid, n, &cfd, &crank)) {
            conns_[crank].fd = cfd;
        }
        /* connect to lower ranks */
        for (uint32_t r = 0; r < me; r++) {
            if (conns_[r].fd >= 0 || connecting_[r] >= 0) {
                continue;
            }
            int fd = socket(AF_INET, SOCK_STREAM, 0);
            struct sockaddr_in a{};
            a.sin_family      = AF_INET;
            a.sin_addr.s_addr = addrs_[r].ip;
            a.sin_port        = htons(addrs_[r].port);
            if (connect(fd, (struct sockaddr *)&a, sizeof(a)) == 0) {
                uint32_t hello[2] = {(uint32_t)team_->team_uid, me};
                if (::send(fd, hello, sizeof(hello), MSG_NOSIGNAL) !=
                    (ssize_t)sizeof(hello)) {
                    close(fd);
                    return UCC_ERR_NO_RESOURCE;
                }
                set_nonblock(fd);
                conns_[r].fd = fd;
            } else {
                close(fd);
                /* retry next create_test call */
            }
        }
        for (uint32_t r = 0; r < n; r++) {
            if (r != me && conns_[r].fd < 0) {
                return UCC_INPROGRESS;
            }
        }
        return UCC_OK;
    }

    void get_scores(Team *team, ScoreMap &map) override;

    void progress()
    {
        for (auto &c : conns_) {
            c.progress();
        }
    }

    uint64_t mktag(uint64_t seq, uint32_t step)
    {
        return ((team_->team_uid & 0xffffull) << 48) | (seq << 16) | step;
    }

    std::vector<Conn>    conns_;
    std::vector<TcpAddr> addrs_;
    std::vector<int>     pending_;
    std::map<uint32_t, int> connecting_init_; /* unused placeholder */
    int                  connecting_[64] = {
        -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1,
        -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1,
        -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1,
        -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1, -1};
    uint64_t seq_ = 1;
};

/* ------------------------------------------------------------ TcpTask  */
/* Base: owns send/recv op storage, helpers over the team mesh. */
class TcpTask : public Task {
  public:
    TcpTask(Context *ctx, TcpTlTeam *tt, const ucc_coll_args_t &args)
        : Task(ctx), tt_(tt), a_(args)
    {
    }
    ~TcpTask() override
    {
        /* error-path teardown: ops may still sit in connection queues —
         * unlink them before freeing or the next Conn::progress would
         * touch freed memory */
        for (auto &c : tt_->conns_) {
            for (auto *s : sends_) {
                auto it = std::find(c.sendq.begin(), c.sendq.end(), s);
                if (it != c.sendq.end()) {
                    c.sendq.erase(it);
                }
            }
            for (auto *r : recvs_) {
                auto it = std::find(c.recvq.begin(), c.recvq.end(), r);
                if (it != c.recvq.end()) {
                    c.recvq.erase(it);
                }
                if (c.cur_recv == r) {
                    c.cur_recv = nullptr; /* drain remainder to sink */
                }
            }
        }
        for (auto *s : sends_) {
            delete s;
        }
        for (auto *r : recvs_) {
            delete r;
        }
    }

  protected:
    SendOp *send_to(uint32_t peer, uint32_t step, const void *buf,
                    size_t len)
    {
        auto *op = new SendOp{tt_->mktag(seq_, step), (const uint8_t *)buf,
                              len};
        sends_.push_back(op);
        tt_->conns_[peer].sendq.push_back(op);
        return op;
    }
    RecvOp *recv_from(uint32_t peer, uint32_t step, void *buf, size_t len)
    {
        auto *op = new RecvOp{tt_->mktag(seq_, step), (uint8_t *)buf, len};
        recvs_.push_back(op);
        Conn &c = tt_->conns_[peer];
        /* check unexpected queue first */
        for (auto it = c.unexp.begin(); it != c.unexp.end(); ++it) {
            if (it->tag == op->tag) {
                memcpy(op->buf, it->data.data(),
                       std::min(op->len, it->data.size()));
                op->done = true;
                c.unexp.erase(it);
                return op;
            }
        }
        c.recvq.push_back(op);
        return op;
    }
    bool ops_done()
    {
        tt_->progress();
        for (auto *s : sends_) {
            if (!s->done) {
                return false;
            }
        }
        for (auto *r : recvs_) {
            if (!r->done) {
                return false;
            }
        }
        return true;
    }
    void clear_ops()
    {
        /* all done => no queue references remain */
        for (auto *s : sends_) {
            delete s;
        }
        for (auto *r : recvs_) {
            delete r;
        }
        sends_.clear();
        recvs_.clear();
    }

    void begin()
    {
        me_  = tt_->team_->rank;
        n_   = tt_->team_->size;
        seq_ = tt_->seq_++;
    }

    TcpTlTeam            *tt_;
    ucc_coll_args_t       a_;
    std::vector<SendOp *> sends_;
    std::vector<RecvOp *> recvs_;
    uint32_t              me_ = 0, n_ = 1;
    uint64_t              seq_ = 0;
    int                   phase_ = 0, round_ = 0;
};

/* ---- allreduce: recursive doubling with non-power-of-2 fold
 * (reference tl/ucp allreduce recursive knomial role, radix 2) */
class TcpAllreduceTask final : public TcpTask {
  public:
    using TcpTask::TcpTask;

    ucc_status_t post() override
    {
        begin();
        dt_    = a_.dst.info.datatype;
        op_    = a_.op;
        dtsz_  = ucc_dt_size(dt_);
        count_ = a_.dst.info.count;
        bytes_ = count_ * dtsz_;
        dst_   = (uint8_t *)a_.dst.info.buffer;
        const bool inplace = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        if (!inplace) {
            ec_cpu::copy(dst_, a_.src.info.buffer, bytes_);
        }
        tmp_.resize(bytes_);
        n2_ = 1;
        while (n2_ * 2 <= n_) {
            n2_ *= 2;
        }
        extras_ = n_ - n2_;
        phase_  = 0;
        mask_   = 1;
        status  = UCC_INPROGRESS;
        return progress_();
    }

    ucc_status_t progress() override { return progress_(); }

  private:
    /* relabeled id among the n2 participants, or -1 */
    int relabel() const
    {
        if (me_ < 2 * extras_) {
            return (me_ & 1) ? (int)(me_ / 2) : -1;
        }
        return (int)(me_ - extras_);
    }
    uint32_t unrelabel(int id) const
    {
        return id < (int)extras_ ? (uint32_t)(2 * id + 1)
                                 : (uint32_t)(id + extras_);
    }

    ucc_status_t progress_()
    {
        switch (phase_) {
        case 0: /* fold: even of first 2*extras sends to odd */
            if (extras_ > 0 && me_ < 2 * extras_) {
                if ((me_ & 1) == 0) {
                    send_to(me_ + 1, 0, dst_, bytes_);
                } else {
                    recv_from(me_ - 1, 0, tmp_.data(), bytes_);
                }
            }
            phase_ = 1;
            [[fallthrough]];
        case 1:
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            if (extras_ > 0 && me_ < 2 * extras_ && (me_ & 1)) {
                const void *srcs[2] = {dst_, tmp_.data()};
                ec_cpu::reduce(dst_, srcs, 2, count_, dt_, op_);
            }
            clear_ops();
            phase_ = 2;
            [[fallthrough]];
        case 2: { /* recursive doubling rounds */
            int id = relabel();
            if (id < 0) {
                phase_ = 4; /* folded-out rank waits for result */
                return progress_();
            }
            if (mask_ >= n2_) {
                phase_ = 4;
                return progress_();
            }
            uint32_t peer = unrelabel(id ^ (int)mask_);
            send_to(peer, 16 + (uint32_t)mask_, dst_, bytes_);
            recv_from(peer, 16 + (uint32_t)mask_, tmp_.data(), bytes_);
            phase_ = 3;
            [[fallthrough]];
        }
        case 3: {
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            const void *srcs[2] = {dst_, tmp_.data()};
            bool last = (mask_ * 2 >= n2_);
            ec_cpu::reduce(dst_, srcs, 2, count_, dt_, op_,
                           last && op_ == UCC_OP_AVG ? 1.0 / n_ : 1.0);
            clear_ops();
            mask_ *= 2;
            phase_ = 2;
            return progress_();
        }
        case 4: /* unfold: odd of first 2*extras sends result to even */
            if (extras_ > 0 && me_ < 2 * extras_) {
                if (me_ & 1) {
                    send_to(me_ - 1, 1, dst_, bytes_);
                } else {
                    recv_from(me_ + 1, 1, dst_, bytes_);
                }
            }
            phase_ = 5;
            [[fallthrough]];
        case 5:
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            return UCC_OK;
        }
        return UCC_INPROGRESS;
    }

    ucc_datatype_t     dt_ = UCC_DT_FLOAT32;
    ucc_reduction_op_t op_ = UCC_OP_SUM;
    size_t             dtsz_ = 4, count_ = 0, bytes_ = 0;
    uint8_t           *dst_ = nullptr;
    std::vector<uint8_t> tmp_;
    uint32_t           n2_ = 1, extras_ = 0;
    size_t             mask_ = 1;
};


/* ---- allreduce SRA: ring reduce-scatter then ring allgather — the
 * bandwidth-optimal large-message algorithm (reference tl/ucp
 * allreduce_sra_knomial.c role, ring-flavoured): total wire traffic
 * 2·S·(n-1)/n per rank vs recursive doubling's S·log2(n). Selected for
 * messages >= UCC_TL_TCP_SRA_MIN (64 KiB default). */
class TcpAllreduceSraTask final : public TcpTask {
  public:
    using TcpTask::TcpTask;

    ucc_status_t post() override
    {
        begin();
        dt_    = a_.dst.info.datatype;
        op_    = a_.op;
        dtsz_  = ucc_dt_size(dt_);
        count_ = a_.dst.info.count;
        dst_   = (uint8_t *)a_.dst.info.buffer;
        const bool inplace = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        if (!inplace) {
            ec_cpu::copy(dst_, a_.src.info.buffer, count_ * dtsz_);
        }
        /* n contiguous element-blocks (last takes the tail) */
        per_ = count_ / n_;
        if (per_ == 0) {
            /* degenerate: fall back semantics = everyone owns block 0 */
            per_ = 0;
        }
        tmp_.resize((per_ + count_ % n_) * dtsz_);
        round_  = 0;
        phase_  = 0;
        in_rs_  = true;
        status  = UCC_INPROGRESS;
        return progress_();
    }

    ucc_status_t progress() override { return progress_(); }

  private:
    uint64_t blk_off(uint32_t b) const { return (uint64_t)b * per_; }
    uint64_t blk_cnt(uint32_t b) const
    {
        return b == n_ - 1 ? count_ - (uint64_t)(n_ - 1) * per_ : per_;
    }

    ucc_status_t progress_()
    {
        const uint32_t right = (me_ + 1) % n_;
        const uint32_t left  = (me_ + n_ - 1) % n_;
        /* phase A: ring reduce-scatter. Step k: send block
         * (me - k), receive block (me - k - 1) and reduce into dst. */
        while (in_rs_ && round_ < (int)n_ - 1) {
            if (phase_ == 0) {
                uint32_t sb = (me_ + n_ - round_) % n_;
                uint32_t rb = (me_ + n_ - round_ - 1) % n_;
                if (blk_cnt(sb)) {
                    send_to(right, (uint32_t)round_,
                            dst_ + blk_off(sb) * dtsz_,
                            blk_cnt(sb) * dtsz_);
                }
                if (blk_cnt(rb)) {
                    recv_from(left, (uint32_t)round_, tmp_.data(),
                              blk_cnt(rb) * dtsz_);
                }
                phase_ = 1;
            }
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            uint32_t rb = (me_ + n_ - round_ - 1) % n_;
            if (blk_cnt(rb)) {
                const bool last = round_ == (int)n_ - 2;
                const void *srcs[2] = {dst_ + blk_off(rb) * dtsz_,
                                       tmp_.data()};
                ec_cpu::reduce(dst_ + blk_off(rb) * dtsz_, srcs, 2,
                               blk_cnt(rb), dt_, 
                               op_ == UCC_OP_AVG ? UCC_OP_SUM : op_,
                               (last && op_ == UCC_OP_AVG) ? 1.0 / n_
                                                           : 1.0);
            }
            phase_ = 0;
            round_++;
        }
        if (in_rs_) {
            in_rs_ = false;
            round_ = 0;
        }
        /* phase B: ring allgather of the reduced blocks */
        while (round_ < (int)n_ - 1) {
            if (phase_ == 0) {
                uint32_t sb = (me_ + 1 + n_ - round_) % n_;
                uint32_t rb = (me_ + n_ - round_) % n_;
                if (blk_cnt(sb)) {
                    send_to(right, 64 + (uint32_t)round_,
                            dst_ + blk_off(sb) * dtsz_,
                            blk_cnt(sb) * dtsz_);
                }
                if (blk_cnt(rb)) {
                    recv_from(left, 64 + (uint32_t)round_,
                              dst_ + blk_off(rb) * dtsz_,
                              blk_cnt(rb) * dtsz_);
                }
                phase_ = 1;
            }
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            phase_ = 0;
            round_++;
        }
        return UCC_OK;
    }

    ucc_datatype_t     dt_ = UCC_DT_FLOAT32;
    ucc_reduction_op_t op_ = UCC_OP_SUM;
    size_t             dtsz_ = 4;
    uint64_t           count_ = 0, per_ = 0;
    uint8_t           *dst_ = nullptr;
    std::vector<uint8_t> tmp_;
    bool               in_rs_ = true;
};

/* ---- k-nomial (radix-k) allreduce (reference tl/ucp
 * allreduce_knomial + recursive_knomial.h PROXY/EXTRA role,
 * re-derived): n2 = largest power of k <= n; EXTRA ranks (vr >= n2)
 * fold their vector into a PROXY (vr - n2) first; proxies run
 * ceil(log_k n2) rounds where each k-group exchanges full vectors and
 * reduces; proxies return the result to their extras. Radix k trades
 * (k-1) messages per round for fewer rounds — the small-message
 * latency knob (UCC_TL_TCP_KN_RADIX). */
class TcpAllreduceKnomialTask final : public TcpTask {
  public:
    TcpAllreduceKnomialTask(Context *ctx, TcpTlTeam *tt,
                            const ucc_coll_args_t &args, uint32_t radix)
        : TcpTask(ctx, tt, args), k_(radix < 2 ? 2 : radix)
    {
    }

    ucc_status_t post() override
    {
        begin();
        dt_    = a_.dst.info.datatype;
        op_    = a_.op == UCC_OP_AVG ? UCC_OP_SUM : a_.op;
        dtsz_  = ucc_dt_size(dt_);
        count_ = a_.dst.info.count;
        bytes_ = count_ * dtsz_;
        dst_   = (uint8_t *)a_.dst.info.buffer;
        if (!(a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE)) {
            ec_cpu::copy(dst_, a_.src.info.buffer, bytes_);
        }
        vr_ = me_; /* allreduce is symmetric: no root shift needed */
        n2_ = 1;
        while (n2_ * k_ <= n_) {
            n2_ *= k_;
        }
        extra_ = vr_ >= n2_;
        /* a proxy may serve SEVERAL extras when n > 2*n2 (extras fold
         * modulo n2 — reference recursive_knomial PROXY role) */
        nex_ = 0;
        if (!extra_) {
            for (uint64_t e = vr_ + n2_; e < n_; e += n2_) {
                nex_++;
            }
        }
        proxy_       = nex_ > 0;
        size_t slots = k_ - 1 > nex_ ? k_ - 1 : nex_;
        tmp_.resize(slots * bytes_);
        phase_ = 0;
        q_     = 1;
        if (extra_) { /* fold into my proxy first */
            send_to((uint32_t)(vr_ % n2_), 63, dst_, bytes_);
        } else {
            uint32_t slot = 0;
            for (uint64_t e = vr_ + n2_; e < n_; e += n2_) {
                recv_from((uint32_t)e, 63,
                          tmp_.data() + (size_t)slot * bytes_, bytes_);
                slot++;
            }
        }
        status = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    ucc_status_t progress_()
    {
        if (phase_ == 0) { /* extra fold */
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            for (uint32_t j = 0; j < nex_; j++) {
                const void *srcs[2] = {dst_,
                                       tmp_.data() + (size_t)j * bytes_};
                ec_cpu::reduce(dst_, srcs, 2, count_, dt_, op_);
            }
            if (extra_) { /* wait for the final result */
                recv_from((uint32_t)(vr_ % n2_), 62, dst_, bytes_);
                phase_ = 3;
                return progress_();
            }
            phase_ = 1;
        }
        while (q_ < n2_) {
            if (phase_ == 1) { /* post this round's k-group exchange */
                uint32_t digit = (uint32_t)((vr_ / q_) % k_);
                uint32_t base  = vr_ - digit * (uint32_t)q_;
                int      slot  = 0;
                for (uint32_t j = 0; j < k_; j++) {
                    if (j == digit) {
                        continue;
                    }
                    uint32_t peer = base + j * (uint32_t)q_;
                    send_to(peer, (uint32_t)round_, dst_, bytes_);
                    recv_from(peer, (uint32_t)round_,
                              tmp_.data() + (size_t)slot * bytes_,
                              bytes_);
                    slot++;
                }
                phase_ = 2;
            }
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            for (uint32_t j = 0; j + 1 < k_; j++) {
                const void *srcs[2] = {dst_,
                                       tmp_.data() + (size_t)j * bytes_};
                ec_cpu::reduce(dst_, srcs, 2, count_, dt_, op_);
            }
            q_ *= k_;
            round_++;
            phase_ = 1;
        }
        if (phase_ != 3 && proxy_) { /* return the result to my extras */
            for (uint64_t e = vr_ + n2_; e < n_; e += n2_) {
                send_to((uint32_t)e, 62, dst_, bytes_);
            }
            phase_ = 3;
        }
        if (!ops_done()) {
            return UCC_INPROGRESS;
        }
        clear_ops();
        if (a_.op == UCC_OP_AVG) {
            const void *srcs[1] = {dst_};
            ec_cpu::reduce(dst_, srcs, 1, count_, dt_, UCC_OP_SUM,
                           1.0 / (double)n_);
        }
        return UCC_OK;
    }

    uint32_t k_ = 2, vr_ = 0;
    uint64_t n2_ = 1, q_ = 1;
    uint32_t nex_ = 0;
    bool     extra_ = false, proxy_ = false;
    ucc_datatype_t     dt_ = UCC_DT_FLOAT32;
    ucc_reduction_op_t op_ = UCC_OP_SUM;
    size_t   dtsz_ = 4, bytes_ = 0;
    uint64_t count_ = 0;
    uint8_t *dst_ = nullptr;
    std::vector<uint8_t> tmp_;
};

/* ---- sliding-window allreduce (reference tl/ucp
 * allreduce_sliding_window.c:23-45 role, re-derived for the socket
 * p2p layer): the message is cut into windows; each window runs the
 * SRA ring independently and up to DEPTH windows are in flight, so
 * the reduce-scatter rounds of window w overlap the allgather rounds
 * of window w-1 — bounded-buffer overlap for huge host messages
 * without the O(message) idle time of a monolithic ring. Window/round
 * pairs are disambiguated by the 16-bit tag step (w*128 + round), so
 * no cross-rank posting-order constraint exists. */
class TcpAllreduceSlidingTask final : public TcpTask {
  public:
    TcpAllreduceSlidingTask(Context *ctx, TcpTlTeam *tt,
                            const ucc_coll_args_t &args,
                            size_t win_bytes, int depth)
        : TcpTask(ctx, tt, args), win_bytes_(win_bytes),
          depth_(depth < 1 ? 1 : depth)
    {
    }

    ucc_status_t post() override
    {
        begin();
        dt_    = a_.dst.info.datatype;
        op_    = a_.op;
        dtsz_  = ucc_dt_size(dt_);
        count_ = a_.dst.info.count;
        dst_   = (uint8_t *)a_.dst.info.buffer;
        if (!(a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE)) {
            ec_cpu::copy(dst_, a_.src.info.buffer, count_ * dtsz_);
        }
        welems_ = win_bytes_ / dtsz_;
        if (welems_ == 0) {
            welems_ = 1;
        }
        /* tag step field fits 512 windows of 128 steps */
        uint64_t maxw = 448;
        if ((count_ + welems_ - 1) / welems_ > maxw) {
            welems_ = (count_ + maxw - 1) / maxw;
        }
        nwin_ = (count_ + welems_ - 1) / welems_;
        size_t d = std::min<size_t>((size_t)depth_, nwin_);
        fl_.assign(d, Win{});
        next_ = 0;
        done_ = 0;
        for (auto &w : fl_) {
            start_window(w);
        }
        status = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    struct Win {
        uint64_t idx = 0, base = 0, cnt = 0, per = 0;
        int      round = 0, phase = 0;
        bool     in_rs = true, active = false;
        std::vector<uint8_t>  tmp;
        std::vector<SendOp *> s_ops;
        std::vector<RecvOp *> r_ops;
    };

    void start_window(Win &w)
    {
        if (next_ >= nwin_) {
            w.active = false;
            return;
        }
        w.idx    = next_++;
        w.base   = w.idx * welems_;
        w.cnt    = std::min<uint64_t>(welems_, count_ - w.base);
        w.per    = w.cnt / n_;
        w.round  = 0;
        w.phase  = 0;
        w.in_rs  = true;
        w.active = true;
        w.tmp.resize((w.per + w.cnt % n_) * dtsz_);
        w.s_ops.clear();
        w.r_ops.clear();
    }

    uint64_t boff(const Win &w, uint32_t b) const
    {
        return w.base + (uint64_t)b * w.per;
    }
    uint64_t bcnt(const Win &w, uint32_t b) const
    {
        return b == n_ - 1 ? w.cnt - (uint64_t)(n_ - 1) * w.per : w.per;
    }
    bool wops_done(const Win &w) const
    {
        for (auto *s : w.s_ops) {
            if (!s->done) {
                return false;
            }
        }
        for (auto *r : w.r_ops) {
            if (!r->done) {
                return false;
            }
        }
        return true;
    }

    ucc_status_t progress_()
    {
        const uint32_t right = (me_ + 1) % n_;
        const uint32_t left  = (me_ + n_ - 1) % n_;
        tt_->progress();
        bool moved = true;
        while (moved) {
            moved = false;
            for (auto &w : fl_) {
                if (!w.active) {
                    continue;
                }
                const uint32_t tag0 = (uint32_t)(w.idx * 128);
                if (w.in_rs && w.round >= (int)n_ - 1) {
                    w.in_rs = false;
                    w.round = 0;
                }
                if (w.in_rs) { /* reduce-scatter rounds */
                    if (w.phase == 0) {
                        uint32_t sb = (me_ + n_ - w.round) % n_;
                        uint32_t rb = (me_ + n_ - w.round - 1) % n_;
                        w.s_ops.clear();
                        w.r_ops.clear();
                        if (bcnt(w, sb)) {
                            w.s_ops.push_back(send_to(
                                right, tag0 + (uint32_t)w.round,
                                dst_ + boff(w, sb) * dtsz_,
                                bcnt(w, sb) * dtsz_));
                        }
                        if (bcnt(w, rb)) {
                            w.r_ops.push_back(recv_from(
                                left, tag0 + (uint32_t)w.round,
                                w.tmp.data(), bcnt(w, rb) * dtsz_));
                        }
                        w.phase = 1;
                        moved   = true;
                    }
                    if (w.phase == 1 && wops_done(w)) {
                        uint32_t rb = (me_ + n_ - w.round - 1) % n_;
                        if (bcnt(w, rb)) {
                            const bool last = w.round == (int)n_ - 2;
                            const void *srcs[2] = {
                                dst_ + boff(w, rb) * dtsz_,
                                w.tmp.data()};
                            ec_cpu::reduce(
                                dst_ + boff(w, rb) * dtsz_, srcs, 2,
                                bcnt(w, rb), dt_,
                                op_ == UCC_OP_AVG ? UCC_OP_SUM : op_,
                                (last && op_ == UCC_OP_AVG)
                                    ? 1.0 / n_
                                    : 1.0);
                        }
                        w.phase = 0;
                        w.round++;
                        moved = true;
                        if (w.round >= (int)n_ - 1) {
                            w.in_rs = false;
                            w.round = 0;
                        }
                    }
                } else if (w.round < (int)n_ - 1) { /* allgather */
                    if (w.phase == 0) {
                        uint32_t sb = (me_ + 1 + n_ - w.round) % n_;
                        uint32_t rb = (me_ + n_ - w.round) % n_;
                        w.s_ops.clear();
                        w.r_ops.clear();
                        if (bcnt(w, sb)) {
                            w.s_ops.push_back(send_to(
                                right, tag0 + 64 + (uint32_t)w.round,
                                dst_ + boff(w, sb) * dtsz_,
                                bcnt(w, sb) * dtsz_));
                        }
                        if (bcnt(w, rb)) {
                            w.r_ops.push_back(recv_from(
                                left, tag0 + 64 + (uint32_t)w.round,
                                dst_ + boff(w, rb) * dtsz_,
                                bcnt(w, rb) * dtsz_));
                        }
                        w.phase = 1;
                        moved   = true;
                    }
                    if (w.phase == 1 && wops_done(w)) {
                        w.phase = 0;
                        w.round++;
                        moved = true;
                    }
                } else { /* window complete */
                    done_++;
                    start_window(w);
                    moved = w.active;
                }
            }
        }
        return done_ == nwin_ ? UCC_OK : UCC_INPROGRESS;
    }

    size_t   win_bytes_;
    int      depth_;
    ucc_datatype_t     dt_ = UCC_DT_FLOAT32;
    ucc_reduction_op_t op_ = UCC_OP_SUM;
    size_t   dtsz_ = 4;
    uint64_t count_ = 0, welems_ = 0, nwin_ = 0, next_ = 0, done_ = 0;
    uint8_t *dst_ = nullptr;
    std::vector<Win> fl_;
};

/* ---- bcast: binomial tree from root. With an ACTIVE_SET
 * ({start, stride, size}, reference ucc.h active_set + tl/ucp active-set
 * bcast), the tree runs over the strided subset only; the wire tag comes
 * from args.tag (FIELD_TAG) so subset traffic cannot collide with the
 * team's sequential collectives (non-members post nothing). */
class TcpBcastTask final : public TcpTask {
  public:
    using TcpTask::TcpTask;

    ucc_status_t post() override
    {
        begin();
        bytes_ = a_.src.info.count * ucc_dt_size(a_.src.info.datatype);
        buf_   = (uint8_t *)a_.src.info.buffer;
        /* non-contiguous generic datatype: move PACKED bytes through the
         * tree; root packs once, every receiver unpacks and forwards the
         * packed image (reference generic-dt pack/unpack cb semantics) */
        gdt_ = ucc_dt_generic_ops(a_.src.info.datatype);
        if (gdt_ && (gdt_->flags & UCC_GENERIC_DT_OPS_FLAG_CONTIG)) {
            gdt_ = nullptr; /* contig generics act like plain bytes */
        }
        if (gdt_) {
            if (!gdt_->ops.start_pack || !gdt_->ops.start_unpack ||
                !gdt_->ops.packed_size || !gdt_->ops.pack ||
                !gdt_->ops.unpack) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            if (me_ == (uint32_t)a_.root) {
                void *obj = gdt_->ops.start_pack(
                    gdt_->cookie, a_.src.info.buffer, a_.src.info.count);
                size_t psz = gdt_->ops.packed_size(obj);
                packed_.resize(psz);
                size_t got = 0;
                while (got < psz) {
                    size_t len = psz - got;
                    if (gdt_->ops.pack(obj, got, packed_.data() + got,
                                       &len) != UCC_OK ||
                        len == 0) {
                        if (gdt_->ops.finish) {
                            gdt_->ops.finish(obj);
                        }
                        return UCC_ERR_NO_MESSAGE;
                    }
                    got += len;
                }
                if (gdt_->ops.finish) {
                    gdt_->ops.finish(obj);
                }
                buf_   = packed_.data();
                bytes_ = psz;
            }
            /* receivers learn the size from the wire header */
        }
        if (a_.mask & UCC_COLL_ARGS_FIELD_ACTIVE_SET) {
            /* subset colls must not consume the team-wide sequence:
             * non-members never post, so bumping it would desync tags */
            tt_->seq_--;
            set_.clear();
            int64_t sz = a_.active_set.size;
            for (int64_t i = 0; i < sz; i++) {
                uint32_t r = (uint32_t)(((a_.active_set.start +
                                          i * a_.active_set.stride) %
                                             (int64_t)n_ +
                                         (int64_t)n_) %
                                        (int64_t)n_);
                set_.push_back(r);
            }
            int me_i = -1, root_i = -1;
            for (size_t i = 0; i < set_.size(); i++) {
                if (set_[i] == me_) {
                    me_i = (int)i;
                }
                if (set_[i] == (uint32_t)a_.root) {
                    root_i = (int)i;
                }
            }
            if (me_i < 0 || root_i < 0) {
                return UCC_ERR_INVALID_PARAM; /* caller not in the set */
            }
            as_n_  = (uint32_t)set_.size();
            vr_    = ((uint32_t)me_i + as_n_ - (uint32_t)root_i) % as_n_;
            root_i_ = (uint32_t)root_i;
            /* dedicated sequence space keyed by the user tag */
            seq_ = 0x2000000000000ull |
                   ((a_.mask & UCC_COLL_ARGS_FIELD_TAG) ? a_.tag : 0);
        } else {
            set_.clear();
            as_n_ = n_;
            vr_   = (me_ + n_ - a_.root) % n_;
        }
        phase_ = 0;
        status = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    uint32_t to_team(uint32_t v) const
    {
        uint32_t idx = (v + root_i_) % as_n_;
        return set_.empty() ? (v + (uint32_t)a_.root) % n_ : set_[idx];
    }

    ucc_status_t progress_()
    {
        if (phase_ == 0) {
            /* receive from parent (highest set bit of vr) */
            if (vr_ != 0) {
                uint32_t hb = 1;
                while (hb * 2 <= vr_) {
                    hb *= 2;
                }
                if (gdt_) {
                    gparent_ = to_team(vr_ - hb);
                } else {
                    recv_from(to_team(vr_ - hb), 0, buf_, bytes_);
                }
            }
            phase_ = 1;
        }
        if (phase_ == 1) {
            if (gdt_ && vr_ != 0 && packed_.empty()) {
                /* dynamic-size packed recv: steal the completed
                 * unexpected message (sender sized it) */
                tt_->progress();
                if (!tt_->conns_[gparent_].take_unexp(
                        tt_->mktag(seq_, 0), &packed_)) {
                    return UCC_INPROGRESS;
                }
                /* unpack into my user buffer */
                void *obj = gdt_->ops.start_unpack(
                    gdt_->cookie, a_.src.info.buffer, a_.src.info.count);
                ucc_status_t us = gdt_->ops.unpack(
                    obj, 0, packed_.data(), packed_.size());
                if (gdt_->ops.finish) {
                    gdt_->ops.finish(obj);
                }
                if (us != UCC_OK) {
                    return us;
                }
                buf_   = packed_.data();
                bytes_ = packed_.size();
            }
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            /* send to children: vr + m for m > highest bit, m < n */
            uint32_t hb = 1;
            while (hb <= vr_) {
                hb *= 2;
            }
            for (uint32_t m = hb; vr_ + m < as_n_; m *= 2) {
                send_to(to_team(vr_ + m), 0, buf_, bytes_);
            }
            phase_ = 2;
        }
        if (!ops_done()) {
            return UCC_INPROGRESS;
        }
        clear_ops();
        return UCC_OK;
    }

    size_t   bytes_ = 0;
    uint8_t *buf_   = nullptr;
    uint32_t vr_    = 0, as_n_ = 0, root_i_ = 0;
    std::vector<uint32_t> set_;
    const ucc_generic_dt_ops_t *gdt_ = nullptr;
    std::vector<uint8_t>        packed_;
    uint32_t                    gparent_ = 0;
};


/* ---- bcast SAG: root scatters blocks around the ring position space,
 * then a ring allgather completes every rank — 2·S·(n-1)/n wire traffic
 * vs the binomial tree's S·log2(n) (reference bcast sag_knomial role).
 * Selected for messages >= UCC_TL_TCP_SAG_MIN (64 KiB default). */
class TcpBcastSagTask final : public TcpTask {
  public:
    using TcpTask::TcpTask;

    ucc_status_t post() override
    {
        begin();
        bytes_ = a_.src.info.count * ucc_dt_size(a_.src.info.datatype);
        buf_   = (uint8_t *)a_.src.info.buffer;
        root_  = (uint32_t)a_.root;
        per_   = bytes_ / n_;
        phase_ = 0;
        round_ = 0;
        status = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    uint64_t blk_off(uint32_t b) const { return (uint64_t)b * per_; }
    uint64_t blk_len(uint32_t b) const
    {
        return b == n_ - 1 ? bytes_ - (uint64_t)(n_ - 1) * per_ : per_;
    }

    ucc_status_t progress_()
    {
        /* virtual ranks: vr 0 = root */
        const uint32_t vr    = (me_ + n_ - root_) % n_;
        const uint32_t right = (me_ + 1) % n_;
        const uint32_t left  = (me_ + n_ - 1) % n_;
        if (phase_ == 0) { /* scatter: root sends block vr to rank vr */
            if (vr == 0) {
                for (uint32_t v = 1; v < n_; v++) {
                    if (blk_len(v)) {
                        send_to((v + root_) % n_, 0, buf_ + blk_off(v),
                                blk_len(v));
                    }
                }
            } else if (blk_len(vr)) {
                recv_from(root_, 0, buf_ + blk_off(vr), blk_len(vr));
            }
            phase_ = 1;
        }
        if (phase_ == 1) {
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            phase_ = 2;
        }
        /* ring allgather over virtual positions: step k, send the block
         * I received k steps ago to the right, take one from the left */
        while (round_ < (int)n_ - 1) {
            if (phase_ == 2) {
                uint32_t sb = (vr + n_ - round_) % n_;
                uint32_t rb = (vr + n_ - round_ - 1) % n_;
                if (blk_len(sb)) {
                    send_to(right, 16 + (uint32_t)round_,
                            buf_ + blk_off(sb), blk_len(sb));
                }
                if (blk_len(rb)) {
                    recv_from(left, 16 + (uint32_t)round_,
                              buf_ + blk_off(rb), blk_len(rb));
                }
                phase_ = 3;
            }
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            phase_ = 2;
            round_++;
        }
        return UCC_OK;
    }

    size_t   bytes_ = 0;
    uint64_t per_   = 0;
    uint8_t *buf_   = nullptr;
    uint32_t root_  = 0;
};

/* ---- bcast: recursive k-nomial tree with a radix knob
 * (reference coll_patterns/recursive_knomial.h:13-17 role, re-derived).
 * Virtual rank vr = (me - root) mod n; vr's parent clears its lowest
 * nonzero base-k digit; after receiving, vr sends to children
 * vr + i*q for every digit level q below its own, i = 1..k-1. Radix
 * k=2 degenerates to the binomial tree; larger k trades tree depth
 * (latency) for root fan-out. UCC_TL_TCP_KN_RADIX. */
class TcpBcastKnomialTask final : public TcpTask {
  public:
    TcpBcastKnomialTask(Context *ctx, TcpTlTeam *tt,
                        const ucc_coll_args_t &args, uint32_t radix)
        : TcpTask(ctx, tt, args), k_(radix < 2 ? 2 : radix)
    {
    }

    ucc_status_t post() override
    {
        begin();
        bytes_ = a_.src.info.count * ucc_dt_size(a_.src.info.datatype);
        buf_   = (uint8_t *)a_.src.info.buffer;
        vr_    = (me_ + n_ - (uint32_t)a_.root) % n_;
        phase_ = 0;
        status = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    uint32_t to_team(uint32_t v) const
    {
        return (v + (uint32_t)a_.root) % n_;
    }

    ucc_status_t progress_()
    {
        if (phase_ == 0) {
            /* my receive level = lowest nonzero base-k digit of vr */
            plevel_ = 1;
            if (vr_ != 0) {
                uint64_t q = 1;
                while ((vr_ / q) % k_ == 0) {
                    q *= k_;
                }
                uint32_t digit  = (uint32_t)((vr_ / q) % k_);
                uint32_t parent = vr_ - digit * (uint32_t)q;
                recv_from(to_team(parent), 0, buf_, bytes_);
                plevel_ = q; /* I fan out at levels < q */
            } else {
                uint64_t q = 1;
                while (q < n_) {
                    q *= k_;
                }
                plevel_ = q;
            }
            phase_ = 1;
        }
        if (phase_ == 1) { /* wait for the parent data */
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            /* fan out: highest level first (big subtrees start early) */
            for (uint64_t q = plevel_ / k_; q >= 1; q /= k_) {
                for (uint32_t i = 1; i < k_; i++) {
                    uint64_t child = vr_ + (uint64_t)i * q;
                    if (child < n_) {
                        send_to(to_team((uint32_t)child), 0, buf_,
                                bytes_);
                    }
                }
                if (q == 1) {
                    break;
                }
            }
            phase_ = 2;
        }
        if (!ops_done()) {
            return UCC_INPROGRESS;
        }
        clear_ops();
        return UCC_OK;
    }

    uint32_t k_ = 2, vr_ = 0;
    uint64_t plevel_ = 1;
    size_t   bytes_ = 0;
    uint8_t *buf_   = nullptr;
};

/* ---- k-nomial reduce (reference tl/ucp reduce knomial role,
 * re-derived as the exact reverse of the k-nomial bcast tree): every
 * rank receives each child subtree's partial sum (children of virtual
 * rank v are v + i*q for level q < fan bound, i in [1,k)), reduces
 * them with its own contribution, and forwards one message to the
 * parent given by clearing the lowest nonzero base-k digit —
 * log_k(n) depth vs the linear task's n-1 root fan. */
class TcpReduceKnomialTask final : public TcpTask {
  public:
    TcpReduceKnomialTask(Context *ctx, TcpTlTeam *tt,
                         const ucc_coll_args_t &args, uint32_t radix)
        : TcpTask(ctx, tt, args), k_(radix < 2 ? 2 : radix)
    {
    }

    ucc_status_t post() override
    {
        begin();
        dt_    = a_.src.info.datatype;
        op_    = a_.op == UCC_OP_AVG ? UCC_OP_SUM : a_.op;
        count_ = a_.src.info.count;
        dtsz_  = ucc_dt_size(dt_);
        root_  = (uint32_t)a_.root;
        vr_    = (me_ + n_ - root_) % n_;
        work_.resize(count_ * dtsz_);
        const void *src = a_.src.info.buffer;
        if (me_ == root_ && (a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE)) {
            src = a_.dst.info.buffer;
        }
        memcpy(work_.data(), src, count_ * dtsz_);
        /* fan bound: lowest nonzero base-k digit position (the root
         * fans at every level) — identical to the bcast tree */
        uint64_t q = 1;
        if (vr_ != 0) {
            while ((vr_ / q) % k_ == 0) {
                q *= k_;
            }
            uint32_t digit = (uint32_t)((vr_ / q) % k_);
            parent_        = (vr_ - digit * (uint32_t)q + root_) % n_;
        } else {
            while (q < n_) {
                q *= k_;
            }
            parent_ = UINT32_MAX;
        }
        plevel_ = q;
        nch_    = 0;
        for (uint64_t lv = 1; lv * k_ <= plevel_; lv *= k_) {
            for (uint32_t i = 1; i < k_; i++) {
                if (vr_ + i * lv < n_) {
                    nch_++;
                }
            }
        }
        chbuf_.resize((size_t)nch_ * count_ * dtsz_);
        phase_ = 0;
        status = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    ucc_status_t progress_()
    {
        if (phase_ == 0) { /* post all child receives */
            uint32_t c = 0;
            for (uint64_t lv = 1; lv * k_ <= plevel_; lv *= k_) {
                for (uint32_t i = 1; i < k_; i++) {
                    uint64_t child = vr_ + (uint64_t)i * lv;
                    if (child < n_) {
                        recv_from((uint32_t)((child + root_) % n_), 0,
                                  chbuf_.data() +
                                      (size_t)c * count_ * dtsz_,
                                  count_ * dtsz_);
                        c++;
                    }
                }
            }
            phase_ = 1;
        }
        if (phase_ == 1) { /* reduce children, forward to parent */
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            for (uint32_t c = 0; c < nch_; c++) {
                const void *srcs[2] = {
                    work_.data(),
                    chbuf_.data() + (size_t)c * count_ * dtsz_};
                ec_cpu::reduce(work_.data(), srcs, 2, count_, dt_, op_);
            }
            if (parent_ != UINT32_MAX) {
                send_to(parent_, 0, work_.data(), count_ * dtsz_);
            }
            phase_ = 2;
        }
        if (!ops_done()) {
            return UCC_INPROGRESS;
        }
        clear_ops();
        if (me_ == root_) {
            void *dst = a_.dst.info.buffer;
            memcpy(dst, work_.data(), count_ * dtsz_);
            if (a_.op == UCC_OP_AVG) {
                const void *s[1] = {dst};
                ec_cpu::reduce(dst, s, 1, count_, dt_, UCC_OP_SUM,
                               1.0 / (double)n_);
            }
        }
        return UCC_OK;
    }

    uint32_t k_ = 2, vr_ = 0, root_ = 0, parent_ = UINT32_MAX, nch_ = 0;
    uint64_t plevel_ = 1, count_ = 0;
    size_t   dtsz_ = 4;
    ucc_datatype_t     dt_ = UCC_DT_FLOAT32;
    ucc_reduction_op_t op_ = UCC_OP_SUM;
    std::vector<uint8_t> work_, chbuf_;
};

/* ---- double binary tree (DBT) bcast/reduce
 * (reference coll_patterns/double_binary_tree.h:51-224 role,
 * re-derived as the two-tree scheme): the message splits in halves;
 * half A flows through an in-order balanced BST over virtual ranks,
 * half B through the same tree shifted by one position, so (for even
 * n) a rank that is internal in one tree is a leaf in the other and
 * per-rank send load stays ~1x the message across both trees —
 * latency*bandwidth-balanced for medium messages. The collective root
 * bridges to/from each tree's BST root with one extra hop when they
 * differ. */
struct DbtNode {
    int parent = -1, left = -1, right = -1;
};

/* node of the in-order balanced BST over positions [0, n) */
static inline DbtNode dbt_node(uint32_t n, uint32_t v)
{
    DbtNode  r;
    uint32_t lo = 0, hi = n;
    while (true) {
        uint32_t mid = lo + (hi - lo) / 2;
        if (v == mid) {
            if (mid > lo) {
                r.left = (int)(lo + (mid - lo) / 2);
            }
            if (mid + 1 < hi) {
                r.right = (int)(mid + 1 + (hi - mid - 1) / 2);
            }
            return r;
        }
        r.parent = (int)mid;
        if (v < mid) {
            hi = mid;
        } else {
            lo = mid + 1;
        }
    }
}

static inline uint32_t dbt_root_pos(uint32_t n) { return n / 2; }

class TcpBcastDbtTask final : public TcpTask {
  public:
    using TcpTask::TcpTask;

    ucc_status_t post() override
    {
        begin();
        bytes_ = a_.src.info.count * ucc_dt_size(a_.src.info.datatype);
        buf_   = (uint8_t *)a_.src.info.buffer;
        halfa_ = bytes_ / 2;
        if (n_ < 2 || halfa_ == 0) {
            return UCC_ERR_NOT_SUPPORTED;
        }
        /* positions: p1 maps the collective root onto the BST root */
        const uint32_t M = dbt_root_pos(n_);
        p1_ = (me_ + n_ - (uint32_t)a_.root + M) % n_;
        p2_ = (p1_ + 1) % n_;
        t1_ = dbt_node(n_, p1_);
        t2_ = dbt_node(n_, p2_);
        /* rank holding tree-2's BST root */
        r2_ = from_p1((M + n_ - 1) % n_);
        phase_ = 0;
        status = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    /* team rank from a p1 position */
    uint32_t from_p1(uint32_t p) const
    {
        const uint32_t M = dbt_root_pos(n_);
        return (p + n_ - M + (uint32_t)a_.root) % n_;
    }
    uint32_t t1_rank(int pos) const { return from_p1((uint32_t)pos); }
    uint32_t t2_rank(int pos) const
    {
        /* p2 = (p1+1) mod n  =>  p1 = (p2-1) mod n */
        return from_p1(((uint32_t)pos + n_ - 1) % n_);
    }

    ucc_status_t progress_()
    {
        const bool     is_root = me_ == (uint32_t)a_.root;
        const uint32_t M       = dbt_root_pos(n_);
        uint8_t       *bufb    = buf_ + halfa_;
        const size_t   lenb    = bytes_ - halfa_;
        if (phase_ == 0) {
            /* half A through tree 1 (root IS tree-1's BST root) */
            if (p1_ != M) {
                recv_from(t1_rank(t1_.parent), 1, buf_, halfa_);
            }
            /* half B through tree 2; bridge root -> r2 if distinct */
            if (is_root && r2_ != me_) {
                send_to(r2_, 2, bufb, lenb);
            }
            if (p2_ == M) { /* tree-2 BST root */
                if (!is_root) {
                    recv_from((uint32_t)a_.root, 2, bufb, lenb);
                }
            } else if (is_root) {
                /* the root already has half B, but its T2 parent still
                 * sends one copy — drain it into scratch so the
                 * connection's tag space stays clean */
                discard_.resize(lenb);
                recv_from(t2_rank(t2_.parent), 3, discard_.data(),
                          lenb);
            } else {
                recv_from(t2_rank(t2_.parent), 3, bufb, lenb);
            }
            phase_   = 1;
            senta_   = false;
            sentb_   = false;
            have_a_  = is_root; /* root == T1's BST root by mapping */
            have_b_  = is_root;
        }
        /* forward each half as soon as it lands (the two trees make
         * independent progress — that is the DBT bandwidth trick) */
        tt_->progress();
        if (phase_ == 1) {
            if (!senta_ && (have_a_ || half_done(0))) {
                if (t1_.left >= 0) {
                    send_to(t1_rank(t1_.left), 1, buf_, halfa_);
                }
                if (t1_.right >= 0) {
                    send_to(t1_rank(t1_.right), 1, buf_, halfa_);
                }
                senta_ = true;
            }
            bool b_in = have_b_ || half_done(1);
            if (!sentb_ && p2_ == M && !is_root) {
                /* tree-2 root got the bridge copy */
                b_in = b_in || bridge_done();
            }
            if (!sentb_ && b_in) {
                if (t2_.left >= 0) {
                    send_to(t2_rank(t2_.left), 3, bufb, lenb);
                }
                if (t2_.right >= 0) {
                    send_to(t2_rank(t2_.right), 3, bufb, lenb);
                }
                sentb_ = true;
            }
            if (senta_ && sentb_ && ops_done()) {
                clear_ops();
                return UCC_OK;
            }
            return UCC_INPROGRESS;
        }
        return UCC_INPROGRESS;
    }

    /* recv completion helpers: recvs_ were posted in phase 0 in a
     * known order; scan for the op with the matching buffer */
    bool half_done(int which) const
    {
        const uint8_t *b = which == 0 ? buf_ : buf_ + halfa_;
        for (auto *r : recvs_) {
            if (r->buf == b && r->done) {
                return true;
            }
        }
        return false;
    }
    bool bridge_done() const { return half_done(1); }

    size_t               bytes_ = 0, halfa_ = 0;
    uint8_t             *buf_   = nullptr;
    uint32_t             p1_ = 0, p2_ = 0, r2_ = 0;
    DbtNode              t1_, t2_;
    std::vector<uint8_t> discard_;
    bool                 senta_ = false, sentb_ = false;
    bool                 have_a_ = false, have_b_ = false;
};

/* ---- DBT reduce: the bcast flow reversed — children's partial sums
 * of each half arrive, are combined with my contribution, and flow to
 * the tree parent; tree roots bridge to the collective root. */
class TcpReduceDbtTask final : public TcpTask {
  public:
    using TcpTask::TcpTask;

    ucc_status_t post() override
    {
        begin();
        dt_    = a_.src.info.datatype;
        op_    = a_.op == UCC_OP_AVG ? UCC_OP_SUM : a_.op;
        dtsz_  = ucc_dt_size(dt_);
        count_ = a_.src.info.count;
        bytes_ = count_ * dtsz_;
        cnta_  = count_ / 2;
        if (n_ < 2 || cnta_ == 0 || count_ == cnta_) {
            return UCC_ERR_NOT_SUPPORTED;
        }
        const bool inplace = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        const bool is_root = me_ == (uint32_t)a_.root;
        /* working copy of my contribution */
        work_.resize(bytes_);
        memcpy(work_.data(),
               is_root && inplace ? a_.dst.info.buffer
                                  : a_.src.info.buffer,
               bytes_);
        const uint32_t M = dbt_root_pos(n_);
        p1_ = (me_ + n_ - (uint32_t)a_.root + M) % n_;
        p2_ = (p1_ + 1) % n_;
        t1_ = dbt_node(n_, p1_);
        t2_ = dbt_node(n_, p2_);
        r2_ = from_p1((M + n_ - 1) % n_);
        /* post child recvs for both halves */
        na_ = nb_ = 0;
        if (t1_.left >= 0) {
            child_recv(0, t1_rank(t1_.left), 1, true);
        }
        if (t1_.right >= 0) {
            child_recv(1, t1_rank(t1_.right), 1, true);
        }
        if (t2_.left >= 0) {
            child_recv(2, t2_rank(t2_.left), 3, false);
        }
        if (t2_.right >= 0) {
            child_recv(3, t2_rank(t2_.right), 3, false);
        }
        phase_ = 0;
        reda_ = redb_ = false;
        status = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    uint32_t from_p1(uint32_t p) const
    {
        const uint32_t M = dbt_root_pos(n_);
        return (p + n_ - M + (uint32_t)a_.root) % n_;
    }
    uint32_t t1_rank(int pos) const { return from_p1((uint32_t)pos); }
    uint32_t t2_rank(int pos) const
    {
        return from_p1(((uint32_t)pos + n_ - 1) % n_);
    }

    void child_recv(int slot, uint32_t rank, uint32_t tag, bool half_a)
    {
        size_t len = half_a ? cnta_ * dtsz_ : bytes_ - cnta_ * dtsz_;
        tmp_[slot].resize(len);
        cr_[slot] = recv_from(rank, tag, tmp_[slot].data(), len);
        if (half_a) {
            na_++;
        } else {
            nb_++;
        }
    }

    bool kids_done(bool half_a) const
    {
        for (int i = half_a ? 0 : 2; i < (half_a ? 2 : 4); i++) {
            if (cr_[i] && !cr_[i]->done) {
                return false;
            }
        }
        return true;
    }

    ucc_status_t progress_()
    {
        tt_->progress();
        const bool     is_root = me_ == (uint32_t)a_.root;
        const uint32_t M       = dbt_root_pos(n_);
        uint8_t       *wa      = work_.data();
        uint8_t       *wb      = work_.data() + cnta_ * dtsz_;
        const uint64_t cb      = count_ - cnta_;
        if (!reda_ && kids_done(true)) {
            /* fold children's half-A partials into mine, pass up T1 */
            for (int i = 0; i < 2; i++) {
                if (cr_[i]) {
                    const void *srcs[2] = {wa, tmp_[i].data()};
                    ec_cpu::reduce(wa, srcs, 2, cnta_, dt_, op_);
                }
            }
            if (p1_ != M) {
                send_to(t1_rank(t1_.parent), 1, wa, cnta_ * dtsz_);
            }
            reda_ = true;
        }
        if (!redb_ && kids_done(false)) {
            for (int i = 2; i < 4; i++) {
                if (cr_[i]) {
                    const void *srcs[2] = {wb, tmp_[i].data()};
                    ec_cpu::reduce(wb, srcs, 2, cb, dt_, op_);
                }
            }
            if (p2_ != M) {
                send_to(t2_rank(t2_.parent), 3, wb,
                        bytes_ - cnta_ * dtsz_);
            } else if (!is_root) {
                /* tree-2 root bridges half B to the collective root */
                send_to((uint32_t)a_.root, 2, wb,
                        bytes_ - cnta_ * dtsz_);
            }
            redb_ = true;
        }
        if (!(reda_ && redb_)) {
            return UCC_INPROGRESS;
        }
        if (is_root && !rootrecv_) {
            /* collective root: half A arrives via T1 (I am its BST
             * root); half B via the bridge unless I am also T2 root */
            if (r2_ != me_) {
                br_.resize(bytes_ - cnta_ * dtsz_);
                brop_ = recv_from(r2_, 2, br_.data(), br_.size());
            }
            rootrecv_ = true;
        }
        if (!ops_done()) {
            return UCC_INPROGRESS;
        }
        if (is_root) {
            uint8_t *dst = (uint8_t *)a_.dst.info.buffer;
            if (brop_) {
                memcpy(wb, br_.data(), br_.size());
            }
            memcpy(dst, work_.data(), bytes_);
            if (a_.op == UCC_OP_AVG) {
                const void *srcs[1] = {dst};
                ec_cpu::reduce(dst, srcs, 1, count_, dt_, UCC_OP_SUM,
                               1.0 / (double)n_);
            }
        }
        clear_ops();
        return UCC_OK;
    }

    ucc_datatype_t       dt_ = UCC_DT_FLOAT32;
    ucc_reduction_op_t   op_ = UCC_OP_SUM;
    size_t               dtsz_ = 4, bytes_ = 0;
    uint64_t             count_ = 0, cnta_ = 0;
    uint32_t             p1_ = 0, p2_ = 0, r2_ = 0;
    DbtNode              t1_, t2_;
    std::vector<uint8_t> work_, br_;
    std::vector<uint8_t> tmp_[4];
    RecvOp              *cr_[4] = {};
    RecvOp              *brop_  = nullptr;
    int                  na_ = 0, nb_ = 0;
    bool reda_ = false, redb_ = false, rootrecv_ = false;
};

/* ---- Bruck allgather (reference tl/ucp allgather bruck +
 * coll_patterns/bruck role, re-derived): ceil(log2 n) rounds; after
 * round k each rank owns 2^k consecutive blocks starting from its own,
 * sends them to (me - 2^k) and receives the next run from (me + 2^k);
 * a final local rotation lands blocks at their team positions. The
 * latency-optimal allgather for small blocks (vs the ring's n-1
 * rounds); any n. */
class TcpAllgatherBruckTask final : public TcpTask {
  public:
    using TcpTask::TcpTask;

    ucc_status_t post() override
    {
        begin();
        const bool inplace = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        dtsz_  = ucc_dt_size(a_.dst.info.datatype);
        total_ = a_.dst.info.count * dtsz_;
        blk_   = total_ / n_;
        dst_   = (uint8_t *)a_.dst.info.buffer;
        if (blk_ == 0 || blk_ * n_ != total_) {
            return UCC_ERR_NOT_SUPPORTED; /* ragged: ring handles it */
        }
        work_.resize(total_);
        memcpy(work_.data(),
               inplace ? dst_ + me_ * blk_
                       : (const uint8_t *)a_.src.info.buffer,
               blk_);
        have_  = 1;
        round_ = 0;
        phase_ = 0;
        status = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    ucc_status_t progress_()
    {
        while (have_ < n_) {
            if (phase_ == 0) {
                uint64_t cnt = have_ < n_ - have_ ? have_ : n_ - have_;
                uint32_t to   = (me_ + n_ - (uint32_t)have_) % n_;
                uint32_t from = (me_ + (uint32_t)have_) % n_;
                send_to(to, (uint32_t)round_, work_.data(), cnt * blk_);
                recv_from(from, (uint32_t)round_,
                          work_.data() + have_ * blk_, cnt * blk_);
                phase_ = 1;
            }
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            uint64_t cnt = have_ < n_ - have_ ? have_ : n_ - have_;
            have_ += cnt;
            round_++;
            phase_ = 0;
        }
        /* local rotation: work[j] is block (me + j) mod n */
        for (uint32_t j = 0; j < n_; j++) {
            memcpy(dst_ + ((me_ + j) % n_) * blk_,
                   work_.data() + (uint64_t)j * blk_, blk_);
        }
        return UCC_OK;
    }

    size_t               dtsz_ = 4;
    uint64_t             total_ = 0, blk_ = 0, have_ = 1;
    uint8_t             *dst_ = nullptr;
    std::vector<uint8_t> work_;
};

/* ---- radix-k knomial allgather (reference tl/ucp allgather knomial
 * role, re-derived as radix-k dissemination at absolute offsets): the
 * sparbit task generalized — at round r each rank owns the circular
 * run [me, me+own) with own = k^r and exchanges with the k-1 peers at
 * distances i*own, receiving their runs which extend the ownership to
 * k*own contiguously; ceil(log_k n) rounds, data-ordered, any n. */
class TcpAllgatherKnomialTask final : public TcpTask {
  public:
    TcpAllgatherKnomialTask(Context *ctx, TcpTlTeam *tt,
                            const ucc_coll_args_t &args, uint32_t radix)
        : TcpTask(ctx, tt, args), k_(radix < 2 ? 2 : radix)
    {
    }

    ucc_status_t post() override
    {
        begin();
        const bool inplace = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        dtsz_ = ucc_dt_size(a_.dst.info.datatype);
        blk_  = a_.dst.info.count * dtsz_ / n_;
        dst_  = (uint8_t *)a_.dst.info.buffer;
        if (blk_ == 0 || blk_ * n_ != a_.dst.info.count * dtsz_) {
            return UCC_ERR_NOT_SUPPORTED; /* ragged: ring handles */
        }
        if (!inplace) {
            memcpy(dst_ + me_ * blk_, a_.src.info.buffer, blk_);
        }
        own_   = 1;
        round_ = 0;
        phase_ = 0;
        status = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    void post_run(uint32_t peer, uint32_t tag, uint64_t first,
                  uint64_t cnt, bool is_send)
    {
        uint64_t head = std::min(cnt, (uint64_t)n_ - first);
        if (is_send) {
            send_to(peer, tag, dst_ + first * blk_, head * blk_);
            if (cnt > head) {
                send_to(peer, tag + 1, dst_, (cnt - head) * blk_);
            }
        } else {
            recv_from(peer, tag, dst_ + first * blk_, head * blk_);
            if (cnt > head) {
                recv_from(peer, tag + 1, dst_, (cnt - head) * blk_);
            }
        }
    }

    ucc_status_t progress_()
    {
        while (own_ < n_) {
            if (phase_ == 0) {
                for (uint32_t i = 1; i < k_; i++) {
                    uint64_t d = (uint64_t)i * own_;
                    if (d >= n_) {
                        break;
                    }
                    uint64_t cnt = std::min((uint64_t)own_,
                                            (uint64_t)n_ - d);
                    uint32_t to   = (uint32_t)((me_ + n_ - d) % n_);
                    uint32_t from = (uint32_t)((me_ + d) % n_);
                    uint32_t tag  =
                        (uint32_t)round_ * 2 * k_ + i * 2;
                    post_run(to, tag, me_, cnt, true);
                    post_run(from, tag, from, cnt, false);
                }
                phase_ = 1;
            }
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            own_ = std::min<uint64_t>((uint64_t)own_ * k_, n_);
            round_++;
            phase_ = 0;
        }
        return UCC_OK;
    }

    uint32_t k_ = 2;
    uint64_t own_ = 1;
    size_t   dtsz_ = 4, blk_ = 0;
    uint8_t *dst_ = nullptr;
};

/* ---- neighbor-exchange allgather (reference tl/ucp allgather
 * neighbor role, Chan et al., re-derived): EVEN n only. Round 0 pairs
 * (2i, 2i+1) swap their own blocks; every later round alternates
 * direction and swaps the CONTIGUOUS 2-block group acquired in the
 * previous round — n/2 rounds total with 2-block messages, vs the
 * ring's n-1 single-block rounds. The group locations follow a
 * data-independent pattern, so every rank simulates all ranks' group
 * state g[p] locally (O(n) per round) and always knows what its
 * partner is sending. */
class TcpAllgatherNeighborTask final : public TcpTask {
  public:
    using TcpTask::TcpTask;

    ucc_status_t post() override
    {
        begin();
        if (n_ % 2 != 0 || n_ < 4) {
            return UCC_ERR_NOT_SUPPORTED; /* odd n: bruck/ring */
        }
        const bool inplace = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        dtsz_ = ucc_dt_size(a_.dst.info.datatype);
        blk_  = a_.dst.info.count * dtsz_ / n_;
        dst_  = (uint8_t *)a_.dst.info.buffer;
        if (blk_ == 0 ||
            blk_ * n_ != a_.dst.info.count * dtsz_) {
            return UCC_ERR_NOT_SUPPORTED;
        }
        if (!inplace) {
            memcpy(dst_ + me_ * blk_, a_.src.info.buffer, blk_);
        }
        g_.resize(n_);
        round_ = 0;
        phase_ = 0;
        status = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    uint32_t partner(uint32_t p, int r) const
    {
        const bool even = (p % 2) == 0;
        if (r == 0) {
            return even ? p + 1 : p - 1;
        }
        if (even == ((r % 2) == 1)) {
            return (p + n_ - 1) % n_;
        }
        return (p + 1) % n_;
    }

    ucc_status_t progress_()
    {
        const int nr = (int)(n_ / 2);
        while (round_ < nr) {
            if (phase_ == 0) {
                uint32_t peer = partner(me_, round_);
                if (round_ == 0) {
                    send_to(peer, 0, dst_ + me_ * blk_, blk_);
                    recv_from(peer, 0, dst_ + peer * blk_, blk_);
                } else {
                    uint32_t sb = g_[me_], rb = g_[peer];
                    for (uint32_t j = 0; j < 2; j++) {
                        send_to(peer, (uint32_t)round_ * 4 + j,
                                dst_ + ((sb + j) % n_) * blk_, blk_);
                        recv_from(peer, (uint32_t)round_ * 4 + j,
                                  dst_ + ((rb + j) % n_) * blk_,
                                  blk_);
                    }
                }
                phase_ = 1;
            }
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            /* advance every rank's group state */
            if (round_ == 0) {
                for (uint32_t q = 0; q < n_; q++) {
                    g_[q] = q & ~1u; /* the (2i,2i+1) pair each holds */
                }
            } else {
                std::vector<uint32_t> ng(n_);
                for (uint32_t q = 0; q < n_; q++) {
                    ng[q] = g_[partner(q, round_)];
                }
                g_ = std::move(ng);
            }
            phase_ = 0;
            round_++;
        }
        return UCC_OK;
    }

    size_t   dtsz_ = 4, blk_ = 0;
    uint8_t *dst_ = nullptr;
    std::vector<uint32_t> g_;
};

/* ---- DBT allreduce (reference tl/ucp allreduce dbt role): composed
 * as DBT reduce to rank 0 followed by DBT bcast of the result — both
 * message halves stream through the two shifted in-order BSTs, so
 * per-rank send load stays ~2x the message across the whole
 * composition (the latency*bandwidth-balanced medium band). */
class TcpAllreduceDbtTask final : public TcpTask {
  public:
    using TcpTask::TcpTask;
    ~TcpAllreduceDbtTask() override { delete sub_; }

    ucc_status_t post() override
    {
        begin(); /* consume one team seq uniformly (composite itself
                    sends nothing) */
        delete sub_; /* re-post of a persistent/reused request */
        sub_ = nullptr;
        const bool inplace = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        dtsz_ = ucc_dt_size(a_.dst.info.datatype);
        scr_.resize(a_.dst.info.count * dtsz_);
        ucc_coll_args_t ra{};
        ra.mask              = UCC_COLL_ARGS_FIELD_FLAGS;
        ra.flags             = a_.flags & UCC_COLL_ARGS_FLAG_TIMEOUT;
        ra.coll_type         = UCC_COLL_TYPE_REDUCE;
        ra.root              = 0;
        ra.op                = a_.op;
        ra.src.info          = a_.dst.info;
        ra.src.info.buffer   = inplace ? a_.dst.info.buffer
                                       : a_.src.info.buffer;
        ra.dst.info          = a_.dst.info;
        ra.dst.info.buffer   = scr_.data();
        sub_                 = new TcpReduceDbtTask(ctx_, tt_, ra);
        ucc_status_t st      = sub_->post();
        if (st == UCC_ERR_NOT_SUPPORTED) {
            return st; /* tiny vectors: recursive doubling handles */
        }
        phase_ = 0;
        status = UCC_INPROGRESS;
        return step(st);
    }
    ucc_status_t progress() override { return step(sub_->progress()); }

  private:
    ucc_status_t step(ucc_status_t st)
    {
        while (true) {
            if (st == UCC_INPROGRESS) {
                return UCC_INPROGRESS;
            }
            if (st != UCC_OK) {
                return st;
            }
            if (phase_ == 0) { /* reduce done: bcast the result */
                delete sub_;
                sub_ = nullptr;
                if (me_ == 0) {
                    memcpy(a_.dst.info.buffer, scr_.data(),
                           scr_.size());
                }
                ucc_coll_args_t ba{};
                ba.mask            = UCC_COLL_ARGS_FIELD_FLAGS;
                ba.flags           = a_.flags & UCC_COLL_ARGS_FLAG_TIMEOUT;
                ba.coll_type       = UCC_COLL_TYPE_BCAST;
                ba.root            = 0;
                ba.src.info        = a_.dst.info;
                sub_               = new TcpBcastDbtTask(ctx_, tt_, ba);
                phase_             = 1;
                st                 = sub_->post();
                continue;
            }
            return UCC_OK; /* bcast complete */
        }
    }

    size_t               dtsz_ = 4;
    std::vector<uint8_t> scr_;
    TcpTask             *sub_ = nullptr;
};

/* ---- sparbit-role allgather (reference tl/ucp allgather sparbit,
 * re-derived): ceil(log2 n) rounds, DATA-ORDERED — every block lands
 * directly at its absolute dst position, so there is no work buffer
 * and no final rotation (Bruck pays both). Round k: rank p owns the
 * circular run [p, p+2^k) of blocks, sends its first min(2^k, n-2^k)
 * to (p-2^k) and receives the run [p+2^k, ...) from (p+2^k); runs that
 * wrap the buffer end split into two ops. Any n. */
class TcpAllgatherSparbitTask final : public TcpTask {
  public:
    using TcpTask::TcpTask;

    ucc_status_t post() override
    {
        begin();
        const bool inplace = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        dtsz_ = ucc_dt_size(a_.dst.info.datatype);
        blk_  = a_.dst.info.count * dtsz_ / n_;
        dst_  = (uint8_t *)a_.dst.info.buffer;
        if (blk_ == 0 ||
            blk_ * n_ != a_.dst.info.count * dtsz_) {
            return UCC_ERR_NOT_SUPPORTED; /* ragged: ring handles it */
        }
        if (!inplace) {
            memcpy(dst_ + me_ * blk_, a_.src.info.buffer, blk_);
        }
        round_ = 0;
        phase_ = 0;
        status = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    /* post run [first, first+cnt) of blocks (mod n) as 1-2 wire ops */
    void post_run(uint32_t peer, uint32_t tag, uint64_t first,
                  uint64_t cnt, bool is_send)
    {
        uint64_t head = std::min(cnt, (uint64_t)n_ - first);
        if (is_send) {
            send_to(peer, tag, dst_ + first * blk_, head * blk_);
            if (cnt > head) {
                send_to(peer, tag + 1, dst_, (cnt - head) * blk_);
            }
        } else {
            recv_from(peer, tag, dst_ + first * blk_, head * blk_);
            if (cnt > head) {
                recv_from(peer, tag + 1, dst_, (cnt - head) * blk_);
            }
        }
    }

    ucc_status_t progress_()
    {
        while ((1ull << round_) < n_) {
            uint64_t d = 1ull << round_;
            if (phase_ == 0) {
                uint64_t cnt  = std::min(d, (uint64_t)n_ - d);
                uint32_t to   = (uint32_t)((me_ + n_ - d) % n_);
                uint32_t from = (uint32_t)((me_ + d) % n_);
                post_run(to, (uint32_t)round_ * 4, me_, cnt, true);
                post_run(from, (uint32_t)round_ * 4, from, cnt, false);
                phase_ = 1;
            }
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            phase_ = 0;
            round_++;
        }
        return UCC_OK;
    }

    size_t   dtsz_ = 4, blk_ = 0;
    uint8_t *dst_ = nullptr;
};

/* ---- barrier / fanin / fanout: binomial fanin to 0 then fanout */
class TcpBarrierTask final : public TcpTask {
  public:
    using TcpTask::TcpTask;

    ucc_status_t post() override
    {
        begin();
        fanin_  = a_.coll_type != UCC_COLL_TYPE_FANOUT;
        fanout_ = a_.coll_type != UCC_COLL_TYPE_FANIN;
        phase_  = fanin_ ? 0 : 2;
        status  = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    ucc_status_t progress_()
    {
        if (phase_ == 0) { /* fanin: recv children then send parent */
            for (uint32_t m = 1; me_ + m < n_; m *= 2) {
                if (me_ & m) {
                    break;
                }
                if ((me_ | m) < n_ && (me_ & (m - 1)) == 0) {
                    recv_from(me_ + m, 0, &token_, 1);
                }
            }
            phase_ = 1;
        }
        if (phase_ == 1) {
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            if (me_ != 0) {
                uint32_t m = 1;
                while ((me_ & m) == 0) {
                    m *= 2;
                }
                send_to(me_ - m, 0, &token_, 1);
            }
            phase_ = 2;
        }
        if (phase_ == 2) {
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            if (!fanout_) {
                return UCC_OK;
            }
            /* fanout = bcast of the token from 0 */
            if (me_ != 0) {
                uint32_t hb = 1;
                while (hb * 2 <= me_) {
                    hb *= 2;
                }
                recv_from(me_ - hb, 1, &token_, 1);
            }
            phase_ = 3;
        }
        if (phase_ == 3) {
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            uint32_t hb = 1;
            while (hb <= me_) {
                hb *= 2;
            }
            for (uint32_t m = hb; me_ + m < n_; m *= 2) {
                send_to(me_ + m, 1, &token_, 1);
            }
            phase_ = 4;
        }
        if (!ops_done()) {
            return UCC_INPROGRESS;
        }
        clear_ops();
        return UCC_OK;
    }

    uint8_t token_ = 1;
    bool    fanin_ = true, fanout_ = true;
};


/* ---- reduce SRG: ring reduce-scatter then linear gather to the root —
 * large-message reduce with 2·S·(n-1)/n wire traffic (reference
 * reduce srg_knomial role). Selected >= UCC_TL_TCP_SRG_MIN (64 KiB). */
class TcpReduceSrgTask final : public TcpTask {
  public:
    using TcpTask::TcpTask;

    ucc_status_t post() override
    {
        begin();
        dt_    = a_.src.info.datatype;
        op_    = a_.op;
        dtsz_  = ucc_dt_size(dt_);
        count_ = a_.src.info.count;
        root_  = (uint32_t)a_.root;
        const bool inplace = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        if (me_ == root_) {
            dst_ = (uint8_t *)a_.dst.info.buffer;
            if (!inplace) {
                ec_cpu::copy(dst_, a_.src.info.buffer, count_ * dtsz_);
            }
            work_ = dst_;
        } else {
            /* non-roots reduce in a private copy (src is read-only) */
            priv_.resize(count_ * dtsz_);
            ec_cpu::copy(priv_.data(), a_.src.info.buffer,
                         count_ * dtsz_);
            work_ = priv_.data();
        }
        per_ = count_ / n_;
        tmp_.resize((per_ + count_ % n_) * dtsz_);
        round_ = 0;
        phase_ = 0;
        in_rs_ = true;
        status = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    uint64_t blk_off(uint32_t b) const { return (uint64_t)b * per_; }
    uint64_t blk_cnt(uint32_t b) const
    {
        return b == n_ - 1 ? count_ - (uint64_t)(n_ - 1) * per_ : per_;
    }

    ucc_status_t progress_()
    {
        const uint32_t right = (me_ + 1) % n_;
        const uint32_t left  = (me_ + n_ - 1) % n_;
        while (in_rs_ && round_ < (int)n_ - 1) {
            if (phase_ == 0) {
                uint32_t sb = (me_ + n_ - round_) % n_;
                uint32_t rb = (me_ + n_ - round_ - 1) % n_;
                if (blk_cnt(sb)) {
                    send_to(right, (uint32_t)round_,
                            work_ + blk_off(sb) * dtsz_,
                            blk_cnt(sb) * dtsz_);
                }
                if (blk_cnt(rb)) {
                    recv_from(left, (uint32_t)round_, tmp_.data(),
                              blk_cnt(rb) * dtsz_);
                }
                phase_ = 1;
            }
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            uint32_t rb = (me_ + n_ - round_ - 1) % n_;
            if (blk_cnt(rb)) {
                const bool last = round_ == (int)n_ - 2;
                const void *srcs[2] = {work_ + blk_off(rb) * dtsz_,
                                       tmp_.data()};
                ec_cpu::reduce(work_ + blk_off(rb) * dtsz_, srcs, 2,
                               blk_cnt(rb), dt_,
                               op_ == UCC_OP_AVG ? UCC_OP_SUM : op_,
                               (last && op_ == UCC_OP_AVG) ? 1.0 / n_
                                                           : 1.0);
            }
            phase_ = 0;
            round_++;
        }
        if (in_rs_) {
            in_rs_ = false;
            phase_ = 0;
            /* gather: each rank owns block (me+1)%n; send it to root */
            uint32_t own = (me_ + 1) % n_;
            if (me_ == root_) {
                for (uint32_t r = 0; r < n_; r++) {
                    uint32_t b = (r + 1) % n_;
                    if (r != me_ && blk_cnt(b)) {
                        recv_from(r, 64, dst_ + blk_off(b) * dtsz_,
                                  blk_cnt(b) * dtsz_);
                    }
                }
            } else if (blk_cnt(own)) {
                send_to(root_, 64, work_ + blk_off(own) * dtsz_,
                        blk_cnt(own) * dtsz_);
            }
        }
        if (!ops_done()) {
            return UCC_INPROGRESS;
        }
        clear_ops();
        return UCC_OK;
    }

    ucc_datatype_t     dt_ = UCC_DT_FLOAT32;
    ucc_reduction_op_t op_ = UCC_OP_SUM;
    size_t             dtsz_ = 4;
    uint64_t           count_ = 0, per_ = 0;
    uint32_t           root_ = 0;
    uint8_t           *dst_ = nullptr, *work_ = nullptr;
    std::vector<uint8_t> priv_, tmp_;
    bool               in_rs_ = true;
};

/* ---- allgather(v): ring */
class TcpAllgatherTask final : public TcpTask {
  public:
    using TcpTask::TcpTask;

    ucc_status_t post() override
    {
        begin();
        const bool is_v    = a_.coll_type == UCC_COLL_TYPE_ALLGATHERV;
        const bool inplace = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        cnt_.resize(n_);
        dsp_.resize(n_);
        if (is_v) {
            size_t ds = ucc_dt_size(a_.dst.info_v.datatype);
            for (uint32_t r = 0; r < n_; r++) {
                cnt_[r] = ((a_.flags & UCC_COLL_ARGS_FLAG_COUNT_64BIT)
                               ? ((const uint64_t *)a_.dst.info_v.counts)[r]
                               : ((const uint32_t *)a_.dst.info_v.counts)[r]) *
                          ds;
                dsp_[r] =
                    ((a_.flags & UCC_COLL_ARGS_FLAG_DISPLACEMENTS_64BIT)
                         ? ((const uint64_t *)a_.dst.info_v.displacements)[r]
                         : ((const uint32_t *)
                                a_.dst.info_v.displacements)[r]) *
                    ds;
            }
            buf_ = (uint8_t *)a_.dst.info_v.buffer;
        } else {
            size_t ds    = ucc_dt_size(a_.dst.info.datatype);
            size_t block = a_.dst.info.count * ds / n_;
            for (uint32_t r = 0; r < n_; r++) {
                cnt_[r] = block;
                dsp_[r] = r * block;
            }
            buf_ = (uint8_t *)a_.dst.info.buffer;
        }
        if (!inplace) {
            ec_cpu::copy(buf_ + dsp_[me_], a_.src.info.buffer, cnt_[me_]);
        }
        round_ = 0;
        phase_ = 0;
        status = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    ucc_status_t progress_()
    {
        const uint32_t right = (me_ + 1) % n_;
        const uint32_t left  = (me_ + n_ - 1) % n_;
        while (round_ < (int)n_ - 1) {
            if (phase_ == 0) {
                uint32_t sb = (me_ + n_ - round_) % n_;     /* send block */
                uint32_t rb = (me_ + n_ - round_ - 1) % n_; /* recv block */
                if (cnt_[sb]) {
                    send_to(right, (uint32_t)round_, buf_ + dsp_[sb],
                            cnt_[sb]);
                }
                if (cnt_[rb]) {
                    recv_from(left, (uint32_t)round_, buf_ + dsp_[rb],
                              cnt_[rb]);
                }
                phase_ = 1;
            }
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            phase_ = 0;
            round_++;
        }
        return UCC_OK;
    }

    std::vector<size_t> cnt_, dsp_;
    uint8_t            *buf_ = nullptr;
};

/* ---- linear batched allgather(v) (reference tl/ucp allgather linear
 * / allgather_batched_num_posts role, re-derived): every rank sends
 * its own block DIRECTLY to all n-1 peers and receives each peer's
 * block into its final position — one hop per block (the ring
 * forwards each block n-1 times through intermediate ranks), at the
 * cost of n-1 simultaneous connections. AG_LINEAR_NUM_POSTS throttles
 * how many peers are in flight (0 = all at once). */
class TcpAllgatherLinearTask final : public TcpTask {
  public:
    using TcpTask::TcpTask;

    ucc_status_t post() override
    {
        begin();
        const bool is_v    = a_.coll_type == UCC_COLL_TYPE_ALLGATHERV;
        const bool inplace = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        cnt_.resize(n_);
        dsp_.resize(n_);
        if (is_v) {
            size_t ds = ucc_dt_size(a_.dst.info_v.datatype);
            for (uint32_t r = 0; r < n_; r++) {
                cnt_[r] = ((a_.flags & UCC_COLL_ARGS_FLAG_COUNT_64BIT)
                               ? ((const uint64_t *)
                                      a_.dst.info_v.counts)[r]
                               : ((const uint32_t *)
                                      a_.dst.info_v.counts)[r]) *
                          ds;
                dsp_[r] =
                    ((a_.flags & UCC_COLL_ARGS_FLAG_DISPLACEMENTS_64BIT)
                         ? ((const uint64_t *)
                                a_.dst.info_v.displacements)[r]
                         : ((const uint32_t *)
                                a_.dst.info_v.displacements)[r]) *
                    ds;
            }
            buf_ = (uint8_t *)a_.dst.info_v.buffer;
        } else {
            size_t ds    = ucc_dt_size(a_.dst.info.datatype);
            size_t block = a_.dst.info.count * ds / n_;
            for (uint32_t r = 0; r < n_; r++) {
                cnt_[r] = block;
                dsp_[r] = r * block;
            }
            buf_ = (uint8_t *)a_.dst.info.buffer;
        }
        if (!inplace) {
            ec_cpu::copy(buf_ + dsp_[me_], a_.src.info.buffer,
                         cnt_[me_]);
        }
        round_ = 1;
        phase_ = 0;
        status = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    ucc_status_t progress_()
    {
        const uint32_t np = (uint32_t)Config::instance().get_int(
            "TL_TCP", "AG_LINEAR_NUM_POSTS", 0);
        const uint32_t w = np == 0 ? n_ : (np < 1 ? 1 : np);
        while (round_ < (int)n_) {
            if (phase_ == 0) {
                uint32_t batch = 0;
                while (round_ + (int)batch < (int)n_ && batch < w) {
                    int      rd   = round_ + (int)batch;
                    uint32_t to   = (me_ + rd) % n_;
                    uint32_t from = (me_ + n_ - rd) % n_;
                    if (cnt_[me_]) {
                        send_to(to, (uint32_t)rd, buf_ + dsp_[me_],
                                cnt_[me_]);
                    }
                    if (cnt_[from]) {
                        recv_from(from, (uint32_t)rd,
                                  buf_ + dsp_[from], cnt_[from]);
                    }
                    batch++;
                }
                batch_ = (int)batch;
                phase_ = 1;
            }
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            phase_ = 0;
            round_ += batch_;
        }
        return UCC_OK;
    }

    std::vector<size_t> cnt_, dsp_;
    uint8_t            *buf_ = nullptr;
    int                 batch_ = 0;
};

/* ---- alltoall(v): pairwise exchange */
/* ---- alltoallv hybrid (reference tl/ucp alltoallv_hybrid.c role,
 * re-derived): pairs >= thr move pairwise-direct; pairs < thr travel
 * as [src,dest,len,bytes] envelopes through a Bruck digit exchange —
 * at round b every rank sends ONE aggregated message (its pending
 * envelopes whose remaining cyclic distance has bit b set) to
 * me+2^b, collapsing up to n-1 tiny messages per rank into
 * ceil(log2 n) aggregated ones. The dominant regime is skewed MoE
 * dispatch where most pairs are small and a few are huge. */
class TcpAlltoallvHybridTask final : public TcpTask {
  public:
    TcpAlltoallvHybridTask(Context *ctx, TcpTlTeam *tt,
                           const ucc_coll_args_t &args, size_t thr)
        : TcpTask(ctx, tt, args),
          thr_(thr > UINT32_MAX ? (size_t)UINT32_MAX : thr)
    {
    }

    ucc_status_t post() override
    {
        begin();
        scnt_.resize(n_);
        sdsp_.resize(n_);
        rcnt_.resize(n_);
        rdsp_.resize(n_);
        size_t ss = ucc_dt_size(a_.src.info_v.datatype);
        size_t ds = ucc_dt_size(a_.dst.info_v.datatype);
        for (uint32_t r = 0; r < n_; r++) {
            auto cat = [&](const void *c, uint32_t i) {
                return (a_.flags & UCC_COLL_ARGS_FLAG_COUNT_64BIT)
                           ? (size_t)((const uint64_t *)c)[i]
                           : (size_t)((const uint32_t *)c)[i];
            };
            auto dat = [&](const void *d, uint32_t i) {
                return (a_.flags &
                        UCC_COLL_ARGS_FLAG_DISPLACEMENTS_64BIT)
                           ? (size_t)((const uint64_t *)d)[i]
                           : (size_t)((const uint32_t *)d)[i];
            };
            scnt_[r] = cat(a_.src.info_v.counts, r) * ss;
            sdsp_[r] = dat(a_.src.info_v.displacements, r) * ss;
            rcnt_[r] = cat(a_.dst.info_v.counts, r) * ds;
            rdsp_[r] = dat(a_.dst.info_v.displacements, r) * ds;
        }
        sbuf_ = (const uint8_t *)a_.src.info_v.buffer;
        dbuf_ = (uint8_t *)a_.dst.info_v.buffer;
        ec_cpu::copy(dbuf_ + rdsp_[me_], sbuf_ + sdsp_[me_],
                     std::min(scnt_[me_], rcnt_[me_]));
        /* large pairs: direct, one message per pair (step tag 1) */
        for (uint32_t r = 1; r < n_; r++) {
            uint32_t to   = (me_ + r) % n_;
            uint32_t from = (me_ + n_ - r) % n_;
            if (scnt_[to] >= thr_ && scnt_[to]) {
                send_to(to, 1, sbuf_ + sdsp_[to], scnt_[to]);
            }
            if (rcnt_[from] >= thr_ && rcnt_[from]) {
                recv_from(from, 1, dbuf_ + rdsp_[from], rcnt_[from]);
            }
        }
        /* small pairs become envelopes */
        pend_.clear();
        for (uint32_t r = 1; r < n_; r++) {
            uint32_t d = (me_ + r) % n_;
            if (scnt_[d] < thr_ && scnt_[d]) {
                Env e;
                e.src  = me_;
                e.dest = d;
                e.data.assign(sbuf_ + sdsp_[d],
                              sbuf_ + sdsp_[d] + scnt_[d]);
                pend_.push_back(std::move(e));
            }
        }
        nrounds_ = 0;
        while ((1u << nrounds_) < n_) {
            nrounds_++;
        }
        sendbufs_.assign(nrounds_, {});
        round_ = 0;
        phase_ = 0;
        status = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    struct Env {
        uint32_t             src, dest;
        std::vector<uint8_t> data;
    };

    void deliver(const Env &e)
    {
        size_t len = std::min((size_t)e.data.size(), rcnt_[e.src]);
        memcpy(dbuf_ + rdsp_[e.src], e.data.data(), len);
    }

    ucc_status_t progress_()
    {
        while (round_ < (int)nrounds_) {
            const uint32_t p    = 1u << round_;
            const uint32_t to   = (me_ + p) % n_;
            const uint32_t from = (me_ + n_ - p) % n_;
            if (phase_ == 0) {
                /* serialize pending envelopes with bit p set in their
                 * remaining distance; keep the rest */
                auto &sb = sendbufs_[round_];
                sb.assign(4, 0); /* u32 count header */
                uint32_t            cnt = 0;
                std::vector<Env>    keep;
                for (auto &e : pend_) {
                    uint32_t dist = (e.dest + n_ - me_) % n_;
                    if (dist & p) {
                        uint32_t hdr[3] = {e.src, e.dest,
                                           (uint32_t)e.data.size()};
                        sb.insert(sb.end(), (uint8_t *)hdr,
                                  (uint8_t *)hdr + sizeof(hdr));
                        sb.insert(sb.end(), e.data.begin(),
                                  e.data.end());
                        cnt++;
                    } else {
                        keep.push_back(std::move(e));
                    }
                }
                pend_ = std::move(keep);
                memcpy(sb.data(), &cnt, 4);
                rsend_ = send_to(to, 0x40 + (uint32_t)round_, sb.data(),
                                 sb.size());
                phase_ = 1;
            }
            /* wait my send + steal the aggregated unexpected message */
            tt_->progress();
            if (!rsend_->done) {
                return UCC_INPROGRESS;
            }
            std::vector<uint8_t> rb;
            if (!tt_->conns_[from].take_unexp(
                    tt_->mktag(seq_, 0x40 + (uint32_t)round_), &rb)) {
                return UCC_INPROGRESS;
            }
            uint32_t cnt = 0;
            if (rb.size() >= 4) {
                memcpy(&cnt, rb.data(), 4);
            }
            size_t off = 4;
            for (uint32_t i = 0; i < cnt; i++) {
                if (off + 12 > rb.size()) {
                    return UCC_ERR_NO_MESSAGE;
                }
                uint32_t hdr[3];
                memcpy(hdr, rb.data() + off, 12);
                off += 12;
                if (off + hdr[2] > rb.size() || hdr[0] >= n_ ||
                    hdr[1] >= n_) {
                    return UCC_ERR_NO_MESSAGE; /* malformed envelope */
                }
                Env e;
                e.src  = hdr[0];
                e.dest = hdr[1];
                e.data.assign(rb.data() + off, rb.data() + off + hdr[2]);
                off += hdr[2];
                if (e.dest == me_) {
                    deliver(e);
                } else {
                    pend_.push_back(std::move(e));
                }
            }
            round_++;
            phase_ = 0;
        }
        if (!ops_done()) { /* direct large-pair messages */
            return UCC_INPROGRESS;
        }
        clear_ops();
        return UCC_OK;
    }

    size_t                            thr_;
    std::vector<size_t>               scnt_, sdsp_, rcnt_, rdsp_;
    const uint8_t                    *sbuf_ = nullptr;
    uint8_t                          *dbuf_ = nullptr;
    std::vector<Env>                  pend_;
    std::vector<std::vector<uint8_t>> sendbufs_;
    SendOp                           *rsend_   = nullptr;
    uint32_t                          nrounds_ = 0;
};

class TcpAlltoallTask final : public TcpTask {
  public:
    using TcpTask::TcpTask;

    ucc_status_t post() override
    {
        begin();
        const bool is_v = a_.coll_type == UCC_COLL_TYPE_ALLTOALLV;
        scnt_.resize(n_);
        sdsp_.resize(n_);
        rcnt_.resize(n_);
        rdsp_.resize(n_);
        if (is_v) {
            size_t ss = ucc_dt_size(a_.src.info_v.datatype);
            size_t ds = ucc_dt_size(a_.dst.info_v.datatype);
            for (uint32_t r = 0; r < n_; r++) {
                auto cat = [&](const void *c, uint32_t i) {
                    return (a_.flags & UCC_COLL_ARGS_FLAG_COUNT_64BIT)
                               ? (size_t)((const uint64_t *)c)[i]
                               : (size_t)((const uint32_t *)c)[i];
                };
                auto dat = [&](const void *d, uint32_t i) {
                    return (a_.flags &
                            UCC_COLL_ARGS_FLAG_DISPLACEMENTS_64BIT)
                               ? (size_t)((const uint64_t *)d)[i]
                               : (size_t)((const uint32_t *)d)[i];
                };
                scnt_[r] = cat(a_.src.info_v.counts, r) * ss;
                sdsp_[r] = dat(a_.src.info_v.displacements, r) * ss;
                rcnt_[r] = cat(a_.dst.info_v.counts, r) * ds;
                rdsp_[r] = dat(a_.dst.info_v.displacements, r) * ds;
            }
            sbuf_ = (const uint8_t *)a_.src.info_v.buffer;
            dbuf_ = (uint8_t *)a_.dst.info_v.buffer;
        } else {
            size_t ds    = ucc_dt_size(a_.dst.info.datatype);
            size_t block = a_.dst.info.count * ds / n_;
            for (uint32_t r = 0; r < n_; r++) {
                scnt_[r] = rcnt_[r] = block;
                sdsp_[r] = rdsp_[r] = r * block;
            }
            sbuf_ = (const uint8_t *)a_.src.info.buffer;
            dbuf_ = (uint8_t *)a_.dst.info.buffer;
        }
        ec_cpu::copy(dbuf_ + rdsp_[me_], sbuf_ + sdsp_[me_], scnt_[me_]);
        round_ = 1;
        phase_ = 0;
        status = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    ucc_status_t progress_()
    {
        /* num_posts batching (reference tl/ucp alltoall pairwise
         * num_posts role): post up to W peer pairs' sends+recvs before
         * waiting — data moves straight between user buffers (no
         * staging), so overlapping rounds costs no memory and hides
         * per-peer latency. 0 = post all n-1 rounds at once. */
        const uint32_t np = (uint32_t)Config::instance().get_int(
            "TL_TCP", "A2A_NUM_POSTS", 0);
        const uint32_t w = np == 0 ? n_ : (np < 1 ? 1 : np);
        while (round_ < (int)n_) {
            if (phase_ == 0) {
                uint32_t batch = 0;
                while (round_ + (int)batch < (int)n_ && batch < w) {
                    int      rd   = round_ + (int)batch;
                    uint32_t to   = (me_ + rd) % n_;
                    uint32_t from = (me_ + n_ - rd) % n_;
                    if (scnt_[to]) {
                        send_to(to, (uint32_t)rd, sbuf_ + sdsp_[to],
                                scnt_[to]);
                    }
                    if (rcnt_[from]) {
                        recv_from(from, (uint32_t)rd,
                                  dbuf_ + rdsp_[from], rcnt_[from]);
                    }
                    batch++;
                }
                batch_ = (int)batch;
                phase_ = 1;
            }
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            phase_ = 0;
            round_ += batch_;
        }
        return UCC_OK;
    }

    std::vector<size_t> scnt_, sdsp_, rcnt_, rdsp_;
    const uint8_t      *sbuf_ = nullptr;
    uint8_t            *dbuf_ = nullptr;
    int                 batch_ = 0;
};

/* ---- Bruck alltoall for small messages: ceil(log2 n) rounds of
 * aggregated exchanges instead of n-1 pairwise sends, so the per-round
 * socket latency is paid log(n) times. Reference parity: the Bruck
 * pattern (coll_patterns/bruck_alltoall.h + tl_ucp alltoall bruck),
 * re-derived from the standard formulation:
 *   1. tmp[j]   = src[(me + j) mod n]
 *   2. step s=1,2,4..: send blocks {j: j&s} to (me+s) mod n, receive
 *      the same positions from (me-s) mod n
 *   3. dst[(me - j) mod n] = tmp[j]
 * Fixed block size only (non-v); registered below TCP_BRUCK_MAX. */
class TcpAlltoallBruckTask final : public TcpTask {
  public:
    using TcpTask::TcpTask;

    ucc_status_t post() override
    {
        begin();
        size_t ds = ucc_dt_size(a_.dst.info.datatype);
        blk_      = a_.dst.info.count * ds / n_;
        sbuf_     = (const uint8_t *)a_.src.info.buffer;
        dbuf_     = (uint8_t *)a_.dst.info.buffer;
        tmp_.resize((size_t)n_ * blk_);
        xs_.resize(((size_t)n_ / 2 + 1) * blk_);
        xr_.resize(((size_t)n_ / 2 + 1) * blk_);
        for (uint32_t j = 0; j < n_; j++) {
            memcpy(tmp_.data() + (size_t)j * blk_,
                   sbuf_ + (size_t)((me_ + j) % n_) * blk_, blk_);
        }
        step_  = 1;
        phase_ = 0;
        status = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    ucc_status_t progress_()
    {
        while (step_ < n_) {
            if (phase_ == 0) { /* pack + exchange */
                nblk_ = 0;
                for (uint32_t j = 0; j < n_; j++) {
                    if (j & step_) {
                        memcpy(xs_.data() + (size_t)nblk_ * blk_,
                               tmp_.data() + (size_t)j * blk_, blk_);
                        nblk_++;
                    }
                }
                uint32_t to   = (me_ + step_) % n_;
                uint32_t from = (me_ + n_ - step_) % n_;
                if (nblk_ && blk_) {
                    send_to(to, 0x4000u + step_, xs_.data(),
                            (size_t)nblk_ * blk_);
                    recv_from(from, 0x4000u + step_, xr_.data(),
                              (size_t)nblk_ * blk_);
                }
                phase_ = 1;
            }
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            uint32_t i = 0;
            for (uint32_t j = 0; j < n_; j++) {
                if (j & step_) {
                    memcpy(tmp_.data() + (size_t)j * blk_,
                           xr_.data() + (size_t)i * blk_, blk_);
                    i++;
                }
            }
            phase_ = 0;
            step_ <<= 1;
        }
        for (uint32_t j = 0; j < n_; j++) {
            memcpy(dbuf_ + (size_t)((me_ + n_ - j) % n_) * blk_,
                   tmp_.data() + (size_t)j * blk_, blk_);
        }
        return UCC_OK;
    }

    size_t               blk_ = 0;
    uint32_t             step_ = 1, nblk_ = 0;
    const uint8_t       *sbuf_ = nullptr;
    uint8_t             *dbuf_ = nullptr;
    std::vector<uint8_t> tmp_, xs_, xr_;
};

/* ---- reduce: linear recv+reduce at root (small n); gather/scatter(v):
 * linear to/from root; reduce_scatter(v): reduce@0 + scatterv */
/* ---- ring reduce-scatter (reference tl/ucp reduce_scatter ring /
 * recursive-halving role, re-derived as the bandwidth-optimal ring):
 * n-1 rounds over a work copy; at round r, send block (me-1-r) to the
 * right, receive block (me-2-r) from the left and reduce — the fully
 * reduced block landing at rank me after the last round is block me,
 * which goes to dst. (n-1)/n * S bytes moved per rank. */
class TcpReduceScatterRingTask final : public TcpTask {
  public:
    using TcpTask::TcpTask;

    ucc_status_t post() override
    {
        begin();
        const bool inplace = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        const bool is_v = a_.coll_type == UCC_COLL_TYPE_REDUCE_SCATTERV;
        dt_   = is_v ? a_.dst.info_v.datatype : a_.dst.info.datatype;
        op_   = a_.op == UCC_OP_AVG ? UCC_OP_SUM : a_.op;
        dtsz_ = ucc_dt_size(dt_);
        cnt_.resize(n_);
        dsp_.resize(n_);
        uint64_t maxb = 0;
        if (is_v) {
            /* v-variant (reference reduce_scatterv ring role): block r
             * is counts[r] elements, laid out contiguously in counts
             * order in the source vector */
            size_t off = 0;
            for (uint32_t r = 0; r < n_; r++) {
                cnt_[r] = ((a_.flags & UCC_COLL_ARGS_FLAG_COUNT_64BIT)
                               ? ((const uint64_t *)
                                      a_.dst.info_v.counts)[r]
                               : ((const uint32_t *)
                                      a_.dst.info_v.counts)[r]) *
                          dtsz_;
                dsp_[r] = off;
                off += cnt_[r];
                maxb = cnt_[r] > maxb ? cnt_[r] : maxb;
            }
            total_ = off;
        } else {
            uint64_t per =
                (inplace ? a_.dst.info.count / n_ : a_.dst.info.count) *
                dtsz_;
            total_ = per * n_;
            if (per == 0 ||
                (inplace && per * n_ != a_.dst.info.count * dtsz_)) {
                return UCC_ERR_NOT_SUPPORTED; /* ragged: linear */
            }
            for (uint32_t r = 0; r < n_; r++) {
                cnt_[r] = per;
                dsp_[r] = (size_t)r * per;
            }
            maxb = per;
        }
        if (total_ == 0) {
            return UCC_OK; /* zero-length vector: nothing to move */
        }
        work_.resize(total_);
        const void *src;
        if (is_v) {
            src = inplace ? a_.dst.info_v.buffer : a_.src.info.buffer;
        } else {
            src = inplace ? a_.dst.info.buffer : a_.src.info.buffer;
        }
        memcpy(work_.data(), src, total_);
        tmp_.resize(maxb);
        /* bidirectional mode (reference tl_ucp reduce_scatter ring
         * bidirectional knob): each block splits in two halves that
         * travel opposite ring directions, using both duplex
         * directions of every link each round */
        bidir_ = Config::instance().get_bool("TL_TCP", "RS_RING_BIDIR",
                                             false) &&
                 n_ > 2;
        if (bidir_) {
            tmpb_.resize(maxb);
        }
        round_ = 0;
        phase_ = 0;
        status = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    uint8_t *blk(uint32_t b) { return work_.data() + dsp_[b]; }
    size_t   ha(uint32_t b) const /* fwd half, element-aligned */
    {
        return cnt_[b] / 2 / dtsz_ * dtsz_;
    }

    ucc_status_t progress_()
    {
        const uint32_t right = (me_ + 1) % n_;
        const uint32_t left  = (me_ + n_ - 1) % n_;
        while (round_ < (int)n_ - 1) {
            uint32_t sf = (me_ + 2 * n_ - 1 - (uint32_t)round_) % n_;
            uint32_t rf = (me_ + 2 * n_ - 2 - (uint32_t)round_) % n_;
            uint32_t sb = (me_ + 1 + (uint32_t)round_) % n_;
            uint32_t rb = (me_ + 2 + (uint32_t)round_) % n_;
            if (phase_ == 0) {
                if (!bidir_) {
                    send_to(right, (uint32_t)round_, blk(sf), cnt_[sf]);
                    recv_from(left, (uint32_t)round_, tmp_.data(),
                              cnt_[rf]);
                } else {
                    /* forward ring carries each block's first half */
                    send_to(right, (uint32_t)round_, blk(sf), ha(sf));
                    recv_from(left, (uint32_t)round_, tmp_.data(),
                              ha(rf));
                    /* backward ring carries the second half */
                    send_to(left, 0x100u + (uint32_t)round_,
                            blk(sb) + ha(sb), cnt_[sb] - ha(sb));
                    recv_from(right, 0x100u + (uint32_t)round_,
                              tmpb_.data(), cnt_[rb] - ha(rb));
                }
                phase_ = 1;
            }
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            if (!bidir_) {
                if (cnt_[rf]) {
                    const void *srcs[2] = {blk(rf), tmp_.data()};
                    ec_cpu::reduce(blk(rf), srcs, 2, cnt_[rf] / dtsz_,
                                   dt_, op_);
                }
            } else {
                if (ha(rf)) {
                    const void *srcs[2] = {blk(rf), tmp_.data()};
                    ec_cpu::reduce(blk(rf), srcs, 2, ha(rf) / dtsz_,
                                   dt_, op_);
                }
                if (cnt_[rb] - ha(rb)) {
                    const void *srcs[2] = {blk(rb) + ha(rb),
                                           tmpb_.data()};
                    ec_cpu::reduce(blk(rb) + ha(rb), srcs, 2,
                                   (cnt_[rb] - ha(rb)) / dtsz_, dt_,
                                   op_);
                }
            }
            phase_ = 0;
            round_++;
        }
        const bool is_v = a_.coll_type == UCC_COLL_TYPE_REDUCE_SCATTERV;
        uint8_t *dst = (uint8_t *)(is_v ? a_.dst.info_v.buffer
                                        : a_.dst.info.buffer);
        if (a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE) {
            dst += dsp_[me_];
        }
        memcpy(dst, blk(me_), cnt_[me_]);
        if (a_.op == UCC_OP_AVG && cnt_[me_]) {
            const void *srcs[1] = {dst};
            ec_cpu::reduce(dst, srcs, 1, cnt_[me_] / dtsz_, dt_,
                           UCC_OP_SUM, 1.0 / (double)n_);
        }
        return UCC_OK;
    }

    ucc_datatype_t     dt_ = UCC_DT_FLOAT32;
    ucc_reduction_op_t op_ = UCC_OP_SUM;
    size_t   dtsz_ = 4;
    uint64_t total_ = 0;
    bool     bidir_ = false;
    std::vector<size_t>  cnt_, dsp_;
    std::vector<uint8_t> work_, tmp_, tmpb_;
};

/* ---- recursive-halving reduce_scatter (reference tl/ucp
 * reduce_scatter knomial / Rabenseifner role, re-derived): log2(m)
 * rounds; at each round the active group and its block range halve,
 * partners exchange the half each keeps and reduce — (m-1)/m * S
 * bytes moved in log rounds, vs the ring's n-1 rounds. Non-power-of-2
 * n folds the first n-m odd ranks into even proxies up front
 * (m = largest power of two <= n); because m may not divide the n
 * output blocks, a final redistribution sends each block from the
 * active that finished owning it to the rank the API assigns it to
 * (a no-op when n is a power of two). */
class TcpReduceScatterHalvingTask final : public TcpTask {
  public:
    using TcpTask::TcpTask;

    ucc_status_t post() override
    {
        begin();
        const bool inplace = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        dt_   = a_.dst.info.datatype;
        op_   = a_.op == UCC_OP_AVG ? UCC_OP_SUM : a_.op;
        dtsz_ = ucc_dt_size(dt_);
        if (inplace) {
            total_ = a_.dst.info.count;
            per_   = total_ / n_;
        } else {
            per_   = a_.dst.info.count;
            total_ = per_ * n_;
        }
        if (per_ == 0 || per_ * n_ != total_ || n_ < 2) {
            return UCC_ERR_NOT_SUPPORTED; /* ragged: linear handles */
        }
        m_ = 1;
        while (m_ * 2 <= n_) {
            m_ *= 2;
        }
        nex_ = n_ - m_; /* extras: odd ranks < 2*nex_ */
        work_.resize((size_t)total_ * dtsz_);
        memcpy(work_.data(),
               inplace ? a_.dst.info.buffer : a_.src.info.buffer,
               (size_t)total_ * dtsz_);
        tmp_.resize((size_t)total_ * dtsz_);
        extra_  = me_ < 2 * nex_ && (me_ % 2) == 1;
        vr_     = extra_ ? UINT32_MAX
                         : (me_ < 2 * nex_ ? me_ / 2 : me_ - nex_);
        lo_     = 0;
        hi_     = n_;
        vlo_    = 0;
        vn_     = m_;
        round_  = 0;
        phase_  = 0;
        status  = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    uint8_t *blk(uint64_t b) { return work_.data() + b * per_ * dtsz_; }
    uint32_t real_of(uint32_t v) const
    {
        return v < nex_ ? 2 * v : v + nex_;
    }
    /* the active virtual rank that ends the halving owning block b */
    uint32_t owner_of(uint64_t b) const
    {
        uint64_t lo = 0, hi = n_;
        uint32_t vlo = 0, vn = m_;
        while (vn > 1) {
            uint64_t mid = lo + (hi - lo) / 2;
            if (b < mid) {
                hi = mid;
            } else {
                lo  = mid;
                vlo += vn / 2;
            }
            vn /= 2;
        }
        return vlo;
    }
    void scale_avg(uint8_t *p)
    {
        if (a_.op == UCC_OP_AVG) {
            const void *s[1] = {p};
            ec_cpu::reduce(p, s, 1, per_, dt_, UCC_OP_SUM,
                           1.0 / (double)n_);
        }
    }

    ucc_status_t progress_()
    {
        /* phase A: fold extras into their even proxies */
        if (round_ == 0 && phase_ == 0) {
            if (extra_) {
                send_to(me_ - 1, 1000, work_.data(),
                        (size_t)total_ * dtsz_);
            } else if (vr_ < nex_) {
                recv_from(me_ + 1, 1000, tmp_.data(),
                          (size_t)total_ * dtsz_);
            }
            phase_ = 1;
        }
        if (round_ == 0 && phase_ == 1) {
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            if (!extra_ && vr_ < nex_) {
                const void *srcs[2] = {work_.data(), tmp_.data()};
                ec_cpu::reduce(work_.data(), srcs, 2, total_, dt_, op_);
            }
            round_ = 1;
            phase_ = 0;
        }
        /* phase B: recursive halving among the m_ actives */
        if (!extra_) {
            while (vn_ > 1) {
                uint64_t mid    = lo_ + (hi_ - lo_) / 2;
                uint32_t half_v = vn_ / 2;
                bool     low    = vr_ < vlo_ + half_v;
                uint32_t peer   = real_of(low ? vr_ + half_v
                                              : vr_ - half_v);
                uint64_t kb = low ? lo_ : mid;   /* keep   */
                uint64_t ke = low ? mid : hi_;
                uint64_t sb = low ? mid : lo_;   /* send   */
                uint64_t se = low ? hi_ : mid;
                if (phase_ == 0) {
                    send_to(peer, (uint32_t)round_ * 4, blk(sb),
                            (se - sb) * per_ * dtsz_);
                    recv_from(peer, (uint32_t)round_ * 4, tmp_.data(),
                              (ke - kb) * per_ * dtsz_);
                    phase_ = 1;
                }
                if (!ops_done()) {
                    return UCC_INPROGRESS;
                }
                clear_ops();
                const void *srcs[2] = {blk(kb), tmp_.data()};
                ec_cpu::reduce(blk(kb), srcs, 2,
                               (ke - kb) * per_, dt_, op_);
                lo_ = kb;
                hi_ = ke;
                if (!low) {
                    vlo_ += half_v;
                }
                vn_ = half_v;
                round_++;
                phase_ = 0;
            }
        }
        /* phase C: redistribute finished blocks to their API owners */
        if (phase_ == 0) {
            uint8_t *dst = (uint8_t *)a_.dst.info.buffer;
            if (a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE) {
                dst += (size_t)me_ * per_ * dtsz_;
            }
            if (!extra_) {
                for (uint64_t b = lo_; b < hi_; b++) {
                    if ((uint32_t)b == me_) {
                        memcpy(dst, blk(b), per_ * dtsz_);
                        scale_avg(dst);
                    } else {
                        send_to((uint32_t)b, 2000, blk(b),
                                per_ * dtsz_);
                    }
                }
            }
            uint32_t own = real_of(owner_of(me_));
            if (own != me_) {
                recv_from(own, 2000, dst, per_ * dtsz_);
                need_scale_ = true;
            }
            phase_ = 2;
        }
        if (!ops_done()) {
            return UCC_INPROGRESS;
        }
        clear_ops();
        if (need_scale_) {
            uint8_t *dst = (uint8_t *)a_.dst.info.buffer;
            if (a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE) {
                dst += (size_t)me_ * per_ * dtsz_;
            }
            scale_avg(dst);
        }
        return UCC_OK;
    }

    ucc_datatype_t     dt_ = UCC_DT_FLOAT32;
    ucc_reduction_op_t op_ = UCC_OP_SUM;
    size_t   dtsz_ = 4;
    uint64_t total_ = 0, per_ = 0, lo_ = 0, hi_ = 0;
    uint32_t m_ = 1, nex_ = 0, vr_ = 0, vlo_ = 0, vn_ = 1;
    bool     extra_ = false, need_scale_ = false;
    std::vector<uint8_t> work_, tmp_;
};

/* ---- k-nomial gather/scatter (reference tl/ucp gather/scatter
 * knomial role, re-derived): blocks travel in VIRTUAL-rank order
 * (vr = (rank - root) mod n) so every subtree owns a contiguous range
 * [vr, vr + q) and each tree edge moves ONE contiguous message.
 * Scatter: the root lays the team blocks out in virtual order in a
 * scratch buffer, then each parent forwards each child its subtree
 * range; ranks keep their own first block. Gather is the exact
 * reverse (children aggregate their subtree ranges up; the root
 * unpacks virtual order back to team order). log_k(n) latency vs the
 * linear task's root fan of n-1 messages. */
class TcpGatherScatterKnTask final : public TcpTask {
  public:
    TcpGatherScatterKnTask(Context *ctx, TcpTlTeam *tt,
                           const ucc_coll_args_t &args, uint32_t radix)
        : TcpTask(ctx, tt, args), k_(radix < 2 ? 2 : radix)
    {
    }

    ucc_status_t post() override
    {
        begin();
        gather_ = a_.coll_type == UCC_COLL_TYPE_GATHER;
        root_   = (uint32_t)a_.root;
        vr_     = (me_ + n_ - root_) % n_;
        const bool inplace = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        if (gather_) {
            dtsz_ = ucc_dt_size(a_.src.info.datatype);
            /* root convention: count is the gathered TOTAL (dst side);
             * leaves pass their single block count */
            blk_ = me_ == root_ ? a_.dst.info.count * dtsz_ / n_
                                : a_.src.info.count * dtsz_;
        } else {
            dtsz_ = ucc_dt_size(me_ == root_ ? a_.src.info.datatype
                                             : a_.dst.info.datatype);
            blk_  = me_ == root_
                        ? a_.src.info.count * dtsz_ / n_
                        : a_.dst.info.count * dtsz_;
        }
        if (blk_ == 0) {
            return UCC_ERR_NOT_SUPPORTED;
        }
        /* subtree size: q = my fan-out level (lowest nonzero base-k
         * digit of vr; n for the root) */
        sub_ = 1;
        if (vr_ == 0) {
            while (sub_ < n_) {
                sub_ *= k_;
            }
        } else {
            uint64_t q = 1;
            while ((vr_ / q) % k_ == 0) {
                q *= k_;
            }
            sub_ = q;
        }
        nsub_ = std::min<uint64_t>(sub_, n_ - vr_);
        work_.resize(nsub_ * blk_);
        if (gather_) {
            /* my own block at virtual offset 0 */
            const void *mysrc =
                (me_ == root_ && inplace)
                    ? (const uint8_t *)a_.dst.info.buffer + me_ * blk_
                    : a_.src.info.buffer;
            memcpy(work_.data(), mysrc, blk_);
            /* children push their subtree ranges up */
            for (uint64_t q = sub_ / k_; q >= 1; q /= k_) {
                for (uint32_t i = 1; i < k_; i++) {
                    uint64_t c = vr_ + (uint64_t)i * q;
                    if (c < n_) {
                        uint64_t cn = std::min<uint64_t>(q, n_ - c);
                        recv_from(to_team((uint32_t)c), 0,
                                  work_.data() + (c - vr_) * blk_,
                                  cn * blk_);
                    }
                }
                if (q == 1) {
                    break;
                }
            }
            phase_ = 0;
        } else { /* scatter */
            if (vr_ == 0) {
                /* root: virtual-order layout */
                const uint8_t *src =
                    (const uint8_t *)(inplace ? a_.dst.info.buffer
                                              : a_.src.info.buffer);
                for (uint64_t v = 0; v < n_; v++) {
                    memcpy(work_.data() + v * blk_,
                           src + ((v + root_) % n_) * blk_, blk_);
                }
                send_children();
                phase_ = 1;
            } else {
                uint32_t digit  = 0;
                uint64_t q      = 1;
                while ((vr_ / q) % k_ == 0) {
                    q *= k_;
                }
                digit = (uint32_t)((vr_ / q) % k_);
                recv_from(to_team(vr_ - digit * (uint32_t)q), 0,
                          work_.data(), nsub_ * blk_);
                phase_ = 0;
            }
        }
        status = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    uint32_t to_team(uint32_t v) const { return (v + root_) % n_; }

    void send_children()
    {
        for (uint64_t q = sub_ / k_; q >= 1; q /= k_) {
            for (uint32_t i = 1; i < k_; i++) {
                uint64_t c = vr_ + (uint64_t)i * q;
                if (c < n_) {
                    uint64_t cn = std::min<uint64_t>(q, n_ - c);
                    send_to(to_team((uint32_t)c), 0,
                            work_.data() + (c - vr_) * blk_,
                            cn * blk_);
                }
            }
            if (q == 1) {
                break;
            }
        }
    }

    ucc_status_t progress_()
    {
        if (gather_) {
            if (phase_ == 0) { /* all children arrived? */
                if (!ops_done()) {
                    return UCC_INPROGRESS;
                }
                clear_ops();
                if (vr_ != 0) { /* push my aggregated range up */
                    uint64_t q = sub_;
                    uint32_t digit =
                        (uint32_t)((vr_ / q) % k_);
                    send_to(to_team(vr_ - digit * (uint32_t)q), 0,
                            work_.data(), nsub_ * blk_);
                }
                phase_ = 1;
            }
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            if (vr_ == 0) { /* unpack virtual -> team order */
                uint8_t *dst = (uint8_t *)a_.dst.info.buffer;
                for (uint64_t v = 0; v < n_; v++) {
                    memcpy(dst + ((v + root_) % n_) * blk_,
                           work_.data() + v * blk_, blk_);
                }
            }
            return UCC_OK;
        }
        /* scatter */
        if (phase_ == 0) { /* wait my range, then forward children */
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            send_children();
            phase_ = 1;
        }
        if (!ops_done()) {
            return UCC_INPROGRESS;
        }
        clear_ops();
        if (me_ == root_) {
            /* root's own block: already in place when IN_PLACE; the
             * root's dst may legitimately be null (send-only root) */
            if (!(a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE) &&
                a_.dst.info.buffer) {
                memcpy(a_.dst.info.buffer,
                       (const uint8_t *)a_.src.info.buffer +
                           me_ * blk_,
                       blk_);
            }
        } else {
            memcpy(a_.dst.info.buffer, work_.data(), blk_);
        }
        return UCC_OK;
    }

    uint32_t k_ = 2, root_ = 0, vr_ = 0;
    bool     gather_ = false;
    uint64_t sub_ = 1, nsub_ = 1;
    size_t   dtsz_ = 4, blk_ = 0;
    std::vector<uint8_t> work_;
};

class TcpRootedTask final : public TcpTask {
  public:
    using TcpTask::TcpTask;

    ucc_status_t post() override
    {
        begin();
        ct_ = a_.coll_type;
        const bool inplace = a_.flags & UCC_COLL_ARGS_FLAG_IN_PLACE;
        root_              = (uint32_t)a_.root;
        switch (ct_) {
        case UCC_COLL_TYPE_REDUCE: {
            dt_    = a_.src.info.datatype;
            op_    = a_.op;
            dtsz_  = ucc_dt_size(dt_);
            count_ = a_.src.info.count;
            bytes_ = count_ * dtsz_;
            if (me_ == root_) {
                dst_ = (uint8_t *)a_.dst.info.buffer;
                if (!inplace) {
                    ec_cpu::copy(dst_, a_.src.info.buffer, bytes_);
                }
                tmp_.resize(bytes_ * (n_ - 1));
                for (uint32_t r = 0; r < n_; r++) {
                    if (r != me_) {
                        uint32_t slot = r < me_ ? r : r - 1;
                        recv_from(r, 0, tmp_.data() + slot * bytes_,
                                  bytes_);
                    }
                }
            } else {
                send_to(root_, 0, a_.src.info.buffer, bytes_);
            }
            break;
        }
        case UCC_COLL_TYPE_REDUCE_SCATTER:
        case UCC_COLL_TYPE_REDUCE_SCATTERV: {
            /* reduce at 0 into tmp, then scatter slices */
            const bool is_v = ct_ == UCC_COLL_TYPE_REDUCE_SCATTERV;
            dt_  = is_v ? a_.dst.info_v.datatype : a_.dst.info.datatype;
            op_  = a_.op;
            dtsz_ = ucc_dt_size(dt_);
            cnt_.resize(n_);
            dsp_.resize(n_);
            if (is_v) {
                size_t off = 0;
                for (uint32_t r = 0; r < n_; r++) {
                    cnt_[r] = ((a_.flags & UCC_COLL_ARGS_FLAG_COUNT_64BIT)
                                   ? ((const uint64_t *)
                                          a_.dst.info_v.counts)[r]
                                   : ((const uint32_t *)
                                          a_.dst.info_v.counts)[r]) *
                              dtsz_;
                    dsp_[r] = off;
                    off += cnt_[r];
                }
                bytes_ = off;
                dst_   = inplace ? (uint8_t *)a_.dst.info_v.buffer + dsp_[me_]
                                 : (uint8_t *)a_.dst.info_v.buffer;
                sptr_  = inplace ? (const uint8_t *)a_.dst.info_v.buffer
                                 : (const uint8_t *)a_.src.info.buffer;
            } else {
                size_t out_b;
                if (inplace) {
                    bytes_ = a_.dst.info.count * dtsz_;
                    out_b  = bytes_ / n_;
                    sptr_  = (const uint8_t *)a_.dst.info.buffer;
                    dst_   = (uint8_t *)a_.dst.info.buffer + me_ * out_b;
                } else {
                    out_b  = a_.dst.info.count * dtsz_;
                    bytes_ = out_b * n_;
                    sptr_  = (const uint8_t *)a_.src.info.buffer;
                    dst_   = (uint8_t *)a_.dst.info.buffer;
                }
                for (uint32_t r = 0; r < n_; r++) {
                    cnt_[r] = out_b;
                    dsp_[r] = r * out_b;
                }
            }
            count_ = bytes_ / dtsz_;
            if (me_ == 0) {
                tmp_.resize(bytes_ * n_); /* [acc][peers...] */
                ec_cpu::copy(tmp_.data(), sptr_, bytes_);
                for (uint32_t r = 1; r < n_; r++) {
                    recv_from(r, 0, tmp_.data() + r * bytes_, bytes_);
                }
            } else {
                send_to(0, 0, sptr_, bytes_);
            }
            break;
        }
        case UCC_COLL_TYPE_GATHER:
        case UCC_COLL_TYPE_GATHERV: {
            const bool is_v = ct_ == UCC_COLL_TYPE_GATHERV;
            if (me_ == root_) {
                dt_ = is_v ? a_.dst.info_v.datatype : a_.dst.info.datatype;
                dtsz_ = ucc_dt_size(dt_);
                dst_  = is_v ? (uint8_t *)a_.dst.info_v.buffer
                             : (uint8_t *)a_.dst.info.buffer;
                cnt_.resize(n_);
                dsp_.resize(n_);
                if (is_v) {
                    for (uint32_t r = 0; r < n_; r++) {
                        cnt_[r] =
                            ((a_.flags & UCC_COLL_ARGS_FLAG_COUNT_64BIT)
                                 ? ((const uint64_t *)
                                        a_.dst.info_v.counts)[r]
                                 : ((const uint32_t *)
                                        a_.dst.info_v.counts)[r]) *
                            dtsz_;
                        dsp_[r] =
                            ((a_.flags &
                              UCC_COLL_ARGS_FLAG_DISPLACEMENTS_64BIT)
                                 ? ((const uint64_t *)
                                        a_.dst.info_v.displacements)[r]
                                 : ((const uint32_t *)
                                        a_.dst.info_v.displacements)[r]) *
                            dtsz_;
                    }
                } else {
                    size_t block = a_.dst.info.count * dtsz_ / n_;
                    for (uint32_t r = 0; r < n_; r++) {
                        cnt_[r] = block;
                        dsp_[r] = r * block;
                    }
                }
                if (!inplace) {
                    ec_cpu::copy(dst_ + dsp_[me_], a_.src.info.buffer,
                                 cnt_[me_]);
                }
                for (uint32_t r = 0; r < n_; r++) {
                    if (r != me_) {
                        recv_from(r, 0, dst_ + dsp_[r], cnt_[r]);
                    }
                }
            } else {
                size_t b = a_.src.info.count *
                           ucc_dt_size(a_.src.info.datatype);
                send_to(root_, 0, a_.src.info.buffer, b);
            }
            break;
        }
        case UCC_COLL_TYPE_SCATTER:
        case UCC_COLL_TYPE_SCATTERV: {
            const bool is_v = ct_ == UCC_COLL_TYPE_SCATTERV;
            if (me_ == root_) {
                dt_ = is_v ? a_.src.info_v.datatype : a_.src.info.datatype;
                dtsz_ = ucc_dt_size(dt_);
                auto *sb = is_v ? (const uint8_t *)a_.src.info_v.buffer
                                : (const uint8_t *)a_.src.info.buffer;
                cnt_.resize(n_);
                dsp_.resize(n_);
                if (is_v) {
                    for (uint32_t r = 0; r < n_; r++) {
                        cnt_[r] =
                            ((a_.flags & UCC_COLL_ARGS_FLAG_COUNT_64BIT)
                                 ? ((const uint64_t *)
                                        a_.src.info_v.counts)[r]
                                 : ((const uint32_t *)
                                        a_.src.info_v.counts)[r]) *
                            dtsz_;
                        dsp_[r] =
                            ((a_.flags &
                              UCC_COLL_ARGS_FLAG_DISPLACEMENTS_64BIT)
                                 ? ((const uint64_t *)
                                        a_.src.info_v.displacements)[r]
                                 : ((const uint32_t *)
                                        a_.src.info_v.displacements)[r]) *
                            dtsz_;
                    }
                } else {
                    size_t block = a_.src.info.count * dtsz_ / n_;
                    for (uint32_t r = 0; r < n_; r++) {
                        cnt_[r] = block;
                        dsp_[r] = r * block;
                    }
                }
                for (uint32_t r = 0; r < n_; r++) {
                    if (r != me_) {
                        send_to(r, 0, sb + dsp_[r], cnt_[r]);
                    }
                }
                if (!inplace && a_.dst.info.buffer) {
                    ec_cpu::copy(a_.dst.info.buffer, sb + dsp_[me_],
                                 cnt_[me_]);
                }
            } else {
                size_t b = a_.dst.info.count *
                           ucc_dt_size(a_.dst.info.datatype);
                recv_from(root_, 0, a_.dst.info.buffer, b);
            }
            break;
        }
        default:
            return UCC_ERR_NOT_SUPPORTED;
        }
        phase_ = 0;
        status = UCC_INPROGRESS;
        return progress_();
    }
    ucc_status_t progress() override { return progress_(); }

  private:
    ucc_status_t progress_()
    {
        if (phase_ == 0) {
            if (!ops_done()) {
                return UCC_INPROGRESS;
            }
            clear_ops();
            if (ct_ == UCC_COLL_TYPE_REDUCE && me_ == root_ && n_ > 1) {
                std::vector<const void *> srcs;
                srcs.push_back(dst_);
                for (uint32_t r = 0; r + 1 < n_; r++) {
                    srcs.push_back(tmp_.data() + r * bytes_);
                }
                ec_cpu::reduce(dst_, srcs.data(), (int)srcs.size(), count_,
                               dt_, op_,
                               op_ == UCC_OP_AVG ? 1.0 / n_ : 1.0);
            }
            if ((ct_ == UCC_COLL_TYPE_REDUCE_SCATTER ||
                 ct_ == UCC_COLL_TYPE_REDUCE_SCATTERV)) {
                if (me_ == 0) {
                    std::vector<const void *> srcs;
                    for (uint32_t r = 0; r < n_; r++) {
                        srcs.push_back(tmp_.data() + r * bytes_);
                    }
                    ec_cpu::reduce(tmp_.data(), srcs.data(),
                                   (int)srcs.size(), count_, dt_, op_,
                                   op_ == UCC_OP_AVG ? 1.0 / n_ : 1.0);
                    /* scatter slices */
                    for (uint32_t r = 1; r < n_; r++) {
                        if (cnt_[r]) {
                            send_to(r, 1, tmp_.data() + dsp_[r], cnt_[r]);
                        }
                    }
                    ec_cpu::copy(dst_, tmp_.data() + dsp_[0], cnt_[0]);
                } else {
                    if (cnt_[me_]) {
                        recv_from(0, 1, dst_, cnt_[me_]);
                    }
                }
                phase_ = 1;
                if (!ops_done()) {
                    return UCC_INPROGRESS;
                }
                clear_ops();
                return UCC_OK;
            }
            return UCC_OK;
        }
        if (!ops_done()) {
            return UCC_INPROGRESS;
        }
        clear_ops();
        return UCC_OK;
    }

    ucc_coll_type_t    ct_ = UCC_COLL_TYPE_REDUCE;
    ucc_datatype_t     dt_ = UCC_DT_FLOAT32;
    ucc_reduction_op_t op_ = UCC_OP_SUM;
    size_t             dtsz_ = 4, count_ = 0, bytes_ = 0;
    uint32_t           root_ = 0;
    uint8_t           *dst_  = nullptr;
    const uint8_t     *sptr_ = nullptr;
    std::vector<uint8_t> tmp_;
    std::vector<size_t>  cnt_, dsp_;
};

/* --------------------------------------------------------------- iface  */
class TcpTl final : public Tl {
  public:
    const char *name() const override { return "tcp"; }
    int         default_score() const override { return 5; }

    TlContext *context_create(Context *ctx) override
    {
        auto &cfg = Config::instance();
        cfg.declare("TL_TCP", "ENABLE", "1",
                    "enable the TCP host transport (inter-node fallback)");
        cfg.declare("TL_TCP", "KN_RADIX", "4",
                    "k-nomial bcast tree radix");
        cfg.declare("TL_TCP", "DBT_MIN", "8192",
                    "double-binary-tree bcast/reduce lower bound bytes");
        cfg.declare("TL_TCP", "DBT_MAX", "4m",
                    "double-binary-tree bcast/reduce upper bound bytes");
        cfg.declare("TL_TCP", "SLIDING_MIN", "64m",
                    "sliding-window allreduce threshold bytes");
        cfg.declare("TL_TCP", "SLIDING_WINDOW", "8m",
                    "sliding-window allreduce window bytes");
        cfg.declare("TL_TCP", "SLIDING_DEPTH", "2",
                    "sliding-window allreduce windows in flight");
        cfg.declare("TL_TCP", "AG_BRUCK_MAX", "64k",
                    "Bruck allgather upper bound bytes");
        cfg.declare("TL_TCP", "AG_SPARBIT_MAX", "256k",
                    "sparbit (data-ordered log-round) allgather upper "
                    "bound bytes");
        cfg.declare("TL_TCP", "AG_NEIGHBOR_MIN", "256k",
                    "neighbor-exchange allgather lower bound bytes "
                    "(even team sizes only)");
        cfg.declare("TL_TCP", "RS_RING_BIDIR", "0",
                    "reduce_scatter ring: split halves over both ring "
                    "directions (full-duplex links)");
        cfg.declare("TL_TCP", "SOCKBUF", "4m",
                    "SO_SNDBUF/SO_RCVBUF bytes (0 = kernel default)");
        cfg.declare("TL_TCP", "AG_LINEAR_MIN", "256k",
                    "linear direct allgather lower bound bytes");
        cfg.declare("TL_TCP", "AG_LINEAR_MAX", "0",
                    "linear direct allgather upper bound bytes "
                    "(0 = unbounded)");
        cfg.declare("TL_TCP", "AG_LINEAR_NUM_POSTS", "0",
                    "linear allgather: peers posted before waiting "
                    "(0 = all at once)");
        cfg.declare("TL_TCP", "A2A_NUM_POSTS", "0",
                    "pairwise alltoall(v): peer pairs posted before "
                    "waiting (0 = all at once)");
        cfg.declare("TL_TCP", "A2AV_HYBRID_THRESH", "4096",
                    "alltoallv hybrid: pairs below this ride the Bruck "
                    "digit exchange (0 disables the hybrid alg)");
        if (!cfg.get_bool("TL_TCP", "ENABLE", true)) {
            return nullptr;
        }
        auto *c = new TcpTlContext(ctx, this);
        if (c->lfd_ < 0) {
            delete c;
            return nullptr;
        }
        return c;
    }

    TlTeam *team_create(TlContext *tlc, Team *team) override
    {
        if (team->size < 2 || team->size > 64) {
            return nullptr;
        }
        /* in-process multi-rank teams: every rank has its own context and
         * listen socket, so the mesh works; allow. */
        return new TcpTlTeam(tlc, team);
    }
};

static TcpTl g_tcp_tl;

Tl *TcpTlContext::iface() { return &g_tcp_tl; }

void TcpTlTeam::get_scores(Team *team, ScoreMap &map)
{
    (void)team;
    TcpTlTeam *self = this;
    int sc = (int)Config::instance().get_int("TL_TCP", "SCORE",
                                             g_tcp_tl.default_score());
    auto add = [&](ucc_coll_type_t ct, auto maker) {
        ScoreRange r;
        r.start    = 0;
        r.end      = SIZE_MAX;
        r.score    = sc;
        r.tl_name  = "tcp";
        r.alg_name = "p2p";
        r.init     = [self, maker, ct](const ucc_coll_args_t &args,
                                   Team *t, Task **task) -> ucc_status_t {
            const ucc_generic_dt_ops_t *g =
                ucc_dt_generic_ops(args.src.info.datatype);
            if (g && !(g->flags & UCC_GENERIC_DT_OPS_FLAG_CONTIG) &&
                ct != UCC_COLL_TYPE_BCAST) {
                /* pack/unpack movement implemented for bcast */
                return UCC_ERR_NOT_SUPPORTED;
            }
            *task = maker(t->ctx, self, args);
            return UCC_OK;
        };
        map.add(ct, UCC_MEMORY_TYPE_HOST, r);
    };
    auto mk = [](auto *tag) {
        using T = std::remove_pointer_t<decltype(tag)>;
        return [](Context *c, TcpTlTeam *tt, const ucc_coll_args_t &a)
                   -> Task * { return new T(c, tt, a); };
    };
    {
        size_t sra_min = Config::instance().get_size("TL_TCP", "SRA_MIN",
                                                     64 * 1024);
        ScoreRange r;
        r.start    = sra_min;
        r.end      = SIZE_MAX;
        r.score    = sc + 1;
        r.tl_name  = "tcp";
        r.alg_name = "sra_ring";
        r.init     = [self](const ucc_coll_args_t &args, Team *t2,
                        Task **task) -> ucc_status_t {
            *task = new TcpAllreduceSraTask(t2->ctx, self, args);
            return UCC_OK;
        };
        map.add(UCC_COLL_TYPE_ALLREDUCE, UCC_MEMORY_TYPE_HOST, r);
    }
    {
        /* radix-k knomial allreduce for the small-message band */
        uint32_t radix = (uint32_t)Config::instance().get_int(
            "TL_TCP", "KN_RADIX", 4);
        size_t kn_max = Config::instance().get_size(
            "TL_TCP", "KN_AR_MAX", 64 * 1024);
        if (radix > 2 && kn_max > 0) {
            ScoreRange r;
            r.start    = 0;
            r.end      = kn_max;
            r.score    = sc + 1;
            r.tl_name  = "tcp";
            r.alg_name = "knomial";
            r.init     = [self, radix](const ucc_coll_args_t &args,
                                   Team *t2,
                                   Task **task) -> ucc_status_t {
                if (!ucc_dt_is_predefined(args.dst.info.datatype)) {
                    return UCC_ERR_NOT_SUPPORTED;
                }
                *task = new TcpAllreduceKnomialTask(t2->ctx, self,
                                                    args, radix);
                return UCC_OK;
            };
            map.add(UCC_COLL_TYPE_ALLREDUCE, UCC_MEMORY_TYPE_HOST, r);
        }
    }
    {
        /* sliding-window overlap for huge host messages (reference
         * allreduce_sliding_window.c role) */
        size_t sw_min = Config::instance().get_size(
            "TL_TCP", "SLIDING_MIN", 64 * 1024 * 1024);
        size_t sw_win = Config::instance().get_size(
            "TL_TCP", "SLIDING_WINDOW", 8 * 1024 * 1024);
        int sw_depth = (int)Config::instance().get_int(
            "TL_TCP", "SLIDING_DEPTH", 2);
        ScoreRange r;
        r.start    = sw_min;
        r.end      = SIZE_MAX;
        r.score    = sc + 2;
        r.tl_name  = "tcp";
        r.alg_name = "sliding_window";
        r.init     = [self, sw_win, sw_depth](
                     const ucc_coll_args_t &args, Team *t2,
                     Task **task) -> ucc_status_t {
            if (!ucc_dt_is_predefined(args.dst.info.datatype)) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            *task = new TcpAllreduceSlidingTask(t2->ctx, self, args,
                                                sw_win, sw_depth);
            return UCC_OK;
        };
        map.add(UCC_COLL_TYPE_ALLREDUCE, UCC_MEMORY_TYPE_HOST, r);
    }
    add(UCC_COLL_TYPE_ALLREDUCE, mk((TcpAllreduceTask *)nullptr));
    {
        size_t sag_min = Config::instance().get_size("TL_TCP", "SAG_MIN",
                                                     64 * 1024);
        ScoreRange r;
        r.start    = sag_min;
        r.end      = SIZE_MAX;
        r.score    = sc + 1;
        r.tl_name  = "tcp";
        r.alg_name = "sag_ring";
        r.init     = [self](const ucc_coll_args_t &args, Team *t2,
                        Task **task) -> ucc_status_t {
            const ucc_generic_dt_ops_t *g =
                ucc_dt_generic_ops(args.src.info.datatype);
            if ((args.mask & UCC_COLL_ARGS_FIELD_ACTIVE_SET) ||
                (g && !(g->flags & UCC_GENERIC_DT_OPS_FLAG_CONTIG))) {
                return UCC_ERR_NOT_SUPPORTED; /* binomial handles these */
            }
            *task = new TcpBcastSagTask(t2->ctx, self, args);
            return UCC_OK;
        };
        map.add(UCC_COLL_TYPE_BCAST, UCC_MEMORY_TYPE_HOST, r);
    }
    {
        /* k-nomial bcast with a radix knob (small-medium latency band)
         * + DBT bcast/reduce (medium bandwidth*latency band) —
         * reference recursive_knomial.h / double_binary_tree.h */
        uint32_t radix = (uint32_t)Config::instance().get_int(
            "TL_TCP", "KN_RADIX", 4);
        size_t dbt_min = Config::instance().get_size("TL_TCP",
                                                     "DBT_MIN", 8192);
        size_t dbt_max = Config::instance().get_size(
            "TL_TCP", "DBT_MAX", 4 * 1024 * 1024);
        ScoreRange kr;
        kr.start    = 0;
        kr.end      = dbt_min;
        kr.score    = sc + 1;
        kr.tl_name  = "tcp";
        kr.alg_name = "knomial";
        kr.init     = [self, radix](const ucc_coll_args_t &args, Team *t2,
                                Task **task) -> ucc_status_t {
            const ucc_generic_dt_ops_t *g =
                ucc_dt_generic_ops(args.src.info.datatype);
            if ((args.mask & UCC_COLL_ARGS_FIELD_ACTIVE_SET) ||
                (g && !(g->flags & UCC_GENERIC_DT_OPS_FLAG_CONTIG))) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            *task = new TcpBcastKnomialTask(t2->ctx, self, args, radix);
            return UCC_OK;
        };
        map.add(UCC_COLL_TYPE_BCAST, UCC_MEMORY_TYPE_HOST, kr);

        ScoreRange db;
        db.start    = dbt_min;
        db.end      = dbt_max;
        db.score    = sc + 2;
        db.tl_name  = "tcp";
        db.alg_name = "dbt";
        db.init     = [self](const ucc_coll_args_t &args, Team *t2,
                         Task **task) -> ucc_status_t {
            const ucc_generic_dt_ops_t *g =
                ucc_dt_generic_ops(args.src.info.datatype);
            if ((args.mask & UCC_COLL_ARGS_FIELD_ACTIVE_SET) ||
                (g && !(g->flags & UCC_GENERIC_DT_OPS_FLAG_CONTIG))) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            *task = new TcpBcastDbtTask(t2->ctx, self, args);
            return UCC_OK;
        };
        map.add(UCC_COLL_TYPE_BCAST, UCC_MEMORY_TYPE_HOST, db);

        ScoreRange dr;
        dr.start    = dbt_min;
        dr.end      = dbt_max;
        dr.score    = sc + 2;
        dr.tl_name  = "tcp";
        dr.alg_name = "dbt";
        dr.init     = [self](const ucc_coll_args_t &args, Team *t2,
                         Task **task) -> ucc_status_t {
            if ((args.mask & UCC_COLL_ARGS_FIELD_ACTIVE_SET) ||
                !ucc_dt_is_predefined(args.src.info.datatype)) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            *task = new TcpReduceDbtTask(t2->ctx, self, args);
            return UCC_OK;
        };
        map.add(UCC_COLL_TYPE_REDUCE, UCC_MEMORY_TYPE_HOST, dr);

        /* DBT allreduce: reduce + bcast through the two shifted BSTs
         * (selectable; the recursive-doubling/SRA defaults measured
         * better on loopback — profiles/tcp_host_crossovers_r02.md) */
        ScoreRange ar;
        ar.start    = dbt_min;
        ar.end      = dbt_max;
        ar.score    = sc; /* tie with the default: tune to enable */
        ar.tl_name  = "tcp";
        ar.alg_name = "dbt";
        ar.init     = [self](const ucc_coll_args_t &args, Team *t2,
                         Task **task) -> ucc_status_t {
            if ((args.mask & UCC_COLL_ARGS_FIELD_ACTIVE_SET) ||
                !ucc_dt_is_predefined(args.dst.info.datatype) ||
                args.dst.info.count < 2) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            *task = new TcpAllreduceDbtTask(t2->ctx, self, args);
            return UCC_OK;
        };
        map.add(UCC_COLL_TYPE_ALLREDUCE, UCC_MEMORY_TYPE_HOST, ar);

        /* k-nomial reduce below the DBT band: log_k(n) hops instead
         * of the linear task's n-1 root fan */
        ScoreRange rk;
        rk.start    = 0;
        rk.end      = dbt_min;
        rk.score    = sc + 1;
        rk.tl_name  = "tcp";
        rk.alg_name = "knomial";
        rk.init     = [self, radix](const ucc_coll_args_t &args,
                                Team *t2, Task **task) -> ucc_status_t {
            if ((args.mask & UCC_COLL_ARGS_FIELD_ACTIVE_SET) ||
                !ucc_dt_is_predefined(args.src.info.datatype)) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            *task = new TcpReduceKnomialTask(t2->ctx, self, args, radix);
            return UCC_OK;
        };
        map.add(UCC_COLL_TYPE_REDUCE, UCC_MEMORY_TYPE_HOST, rk);
    }
    add(UCC_COLL_TYPE_BCAST, mk((TcpBcastTask *)nullptr));
    add(UCC_COLL_TYPE_BARRIER, mk((TcpBarrierTask *)nullptr));
    add(UCC_COLL_TYPE_FANIN, mk((TcpBarrierTask *)nullptr));
    add(UCC_COLL_TYPE_FANOUT, mk((TcpBarrierTask *)nullptr));
    {
        /* Bruck allgather: log-latency for small blocks */
        size_t bmax = Config::instance().get_size(
            "TL_TCP", "AG_BRUCK_MAX", 64 * 1024);
        ScoreRange r;
        r.start    = 0;
        r.end      = bmax;
        r.score    = sc + 3; /* measured: beats sparbit below 64k
                                (profiles/tcp_host_crossovers_r02.md) */
        r.tl_name  = "tcp";
        r.alg_name = "bruck";
        r.init     = [self](const ucc_coll_args_t &args, Team *t2,
                        Task **task) -> ucc_status_t {
            if (!ucc_dt_is_predefined(args.dst.info.datatype)) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            /* equal blocks required: ragged totals fall back at
             * INIT time (post-time rejection has no fallback chain) */
            if ((args.dst.info.count *
                 ucc_dt_size(args.dst.info.datatype)) %
                    t2->size !=
                0) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            *task = new TcpAllgatherBruckTask(t2->ctx, self, args);
            return UCC_OK;
        };
        map.add(UCC_COLL_TYPE_ALLGATHER, UCC_MEMORY_TYPE_HOST, r);
    }
    {
        /* radix-k knomial allgather: ceil(log_k n) rounds with k-1
         * exchanges each (selectable; sparbit is the k=2 default) */
        uint32_t agk = (uint32_t)Config::instance().get_int(
            "TL_TCP", "KN_RADIX", 4);
        ScoreRange r;
        r.start    = 0;
        r.end      = Config::instance().get_size("TL_TCP",
                                                 "AG_BRUCK_MAX",
                                                 64 * 1024);
        r.score    = sc; /* tie with default: tune to enable */
        r.tl_name  = "tcp";
        r.alg_name = "knomial";
        r.init     = [self, agk](const ucc_coll_args_t &args, Team *t2,
                        Task **task) -> ucc_status_t {
            if (!ucc_dt_is_predefined(args.dst.info.datatype) ||
                (args.dst.info.count *
                 ucc_dt_size(args.dst.info.datatype)) %
                        t2->size !=
                    0) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            *task = new TcpAllgatherKnomialTask(t2->ctx, self, args,
                                                agk);
            return UCC_OK;
        };
        map.add(UCC_COLL_TYPE_ALLGATHER, UCC_MEMORY_TYPE_HOST, r);
    }
    {
        /* sparbit role: log-round AND data-ordered (no work buffer, no
         * rotation) — preferred over bruck in the small-block band */
        size_t smax = Config::instance().get_size(
            "TL_TCP", "AG_SPARBIT_MAX", 256 * 1024);
        ScoreRange r;
        r.start    = 0;
        r.end      = smax;
        r.score    = sc + 2; /* wins 64k-256k in the measured sweep
                                (crossover re-checked after SOCKBUF) */
        r.tl_name  = "tcp";
        r.alg_name = "sparbit";
        r.init     = [self](const ucc_coll_args_t &args, Team *t2,
                        Task **task) -> ucc_status_t {
            if (!ucc_dt_is_predefined(args.dst.info.datatype)) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            /* equal blocks required: ragged totals fall back at
             * INIT time (post-time rejection has no fallback chain) */
            if ((args.dst.info.count *
                 ucc_dt_size(args.dst.info.datatype)) %
                    t2->size !=
                0) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            *task = new TcpAllgatherSparbitTask(t2->ctx, self, args);
            return UCC_OK;
        };
        map.add(UCC_COLL_TYPE_ALLGATHER, UCC_MEMORY_TYPE_HOST, r);
    }
    {
        /* linear direct: one hop per block, all peers in flight —
         * wins the band between the log-round and ring regimes */
        size_t lmin = Config::instance().get_size(
            "TL_TCP", "AG_LINEAR_MIN", 256 * 1024);
        size_t lmax = Config::instance().get_size(
            "TL_TCP", "AG_LINEAR_MAX", 0); /* 0 = unbounded */
        if (lmax == 0) {
            lmax = SIZE_MAX;
        }
        ScoreRange r;
        r.start    = lmin;
        r.end      = lmax;
        r.score    = sc + 2; /* one hop per block: measured 2-2.7x over
                                the ring forwarding path >= 1 MiB */
        r.tl_name  = "tcp";
        r.alg_name = "linear";
        r.init     = [self](const ucc_coll_args_t &args, Team *t2,
                        Task **task) -> ucc_status_t {
            ucc_datatype_t dt =
                args.coll_type == UCC_COLL_TYPE_ALLGATHERV
                    ? args.dst.info_v.datatype
                    : args.dst.info.datatype;
            if (!ucc_dt_is_predefined(dt)) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            if (args.coll_type == UCC_COLL_TYPE_ALLGATHER &&
                (args.dst.info.count * ucc_dt_size(dt)) % t2->size !=
                    0) {
                return UCC_ERR_NOT_SUPPORTED; /* ragged: default task */
            }
            *task = new TcpAllgatherLinearTask(t2->ctx, self, args);
            return UCC_OK;
        };
        map.add(UCC_COLL_TYPE_ALLGATHER, UCC_MEMORY_TYPE_HOST, r);
        map.add(UCC_COLL_TYPE_ALLGATHERV, UCC_MEMORY_TYPE_HOST, r);
    }
    {
        /* neighbor exchange: n/2 rounds of 2-block swaps (even n) —
         * halves the per-round latency count vs the ring at the same
         * wire volume; wins for mid/large blocks on even teams */
        size_t nmin = Config::instance().get_size(
            "TL_TCP", "AG_NEIGHBOR_MIN", 256 * 1024);
        ScoreRange r;
        r.start    = nmin;
        r.end      = SIZE_MAX;
        r.score    = sc + 1;
        r.tl_name  = "tcp";
        r.alg_name = "neighbor";
        r.init     = [self](const ucc_coll_args_t &args, Team *t2,
                        Task **task) -> ucc_status_t {
            if (t2->size % 2 != 0 || t2->size < 4 ||
                !ucc_dt_is_predefined(args.dst.info.datatype)) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            /* equal blocks required: ragged totals fall back at
             * INIT time (post-time rejection has no fallback chain) */
            if ((args.dst.info.count *
                 ucc_dt_size(args.dst.info.datatype)) %
                    t2->size !=
                0) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            *task = new TcpAllgatherNeighborTask(t2->ctx, self, args);
            return UCC_OK;
        };
        map.add(UCC_COLL_TYPE_ALLGATHER, UCC_MEMORY_TYPE_HOST, r);
    }
    add(UCC_COLL_TYPE_ALLGATHER, mk((TcpAllgatherTask *)nullptr));
    add(UCC_COLL_TYPE_ALLGATHERV, mk((TcpAllgatherTask *)nullptr));
    add(UCC_COLL_TYPE_ALLTOALL, mk((TcpAlltoallTask *)nullptr));
    {
        /* Bruck wins while per-round latency dominates: log2(n) rounds
         * vs n-1, at the cost of 2x data volume (each block moves
         * ~log(n)/2 times) */
        size_t bruck_max = Config::instance().get_size(
            "TL_TCP", "BRUCK_MAX", 64 * 1024);
        if (bruck_max > 0) {
            ScoreRange r;
            r.start    = 0;
            r.end      = bruck_max;
            r.score    = sc + 1;
            r.tl_name  = "tcp";
            r.alg_name = "bruck";
            r.init     = [self](const ucc_coll_args_t &args, Team *t2,
                            Task **task) -> ucc_status_t {
                const ucc_generic_dt_ops_t *g =
                    ucc_dt_generic_ops(args.dst.info.datatype);
                if ((args.flags & UCC_COLL_ARGS_FLAG_IN_PLACE) ||
                    (g && !(g->flags & UCC_GENERIC_DT_OPS_FLAG_CONTIG))) {
                    return UCC_ERR_NOT_SUPPORTED;
                }
                *task = new TcpAlltoallBruckTask(t2->ctx, self, args);
                return UCC_OK;
            };
            map.add(UCC_COLL_TYPE_ALLTOALL, UCC_MEMORY_TYPE_HOST, r);
        }
    }
    {
        /* hybrid a2av: Bruck digit-exchange for small pairs + direct
         * pairwise for large (skewed MoE dispatch regime) */
        size_t thr = Config::instance().get_size(
            "TL_TCP", "A2AV_HYBRID_THRESH", 4096);
        if (thr > 0) {
            ScoreRange r;
            r.start    = 0;
            r.end      = SIZE_MAX;
            r.score    = sc + 1;
            r.tl_name  = "tcp";
            r.alg_name = "hybrid";
            r.init     = [self, thr](const ucc_coll_args_t &args, Team *t2,
                                 Task **task) -> ucc_status_t {
                if ((args.mask & UCC_COLL_ARGS_FIELD_ACTIVE_SET) ||
                    !ucc_dt_is_predefined(args.src.info_v.datatype) ||
                    (args.flags & UCC_COLL_ARGS_FLAG_IN_PLACE)) {
                    return UCC_ERR_NOT_SUPPORTED;
                }
                *task = new TcpAlltoallvHybridTask(t2->ctx, self, args,
                                                   thr);
                return UCC_OK;
            };
            map.add(UCC_COLL_TYPE_ALLTOALLV, UCC_MEMORY_TYPE_HOST, r);
        }
    }
    add(UCC_COLL_TYPE_ALLTOALLV, mk((TcpAlltoallTask *)nullptr));
    {
        size_t srg_min = Config::instance().get_size("TL_TCP", "SRG_MIN",
                                                     64 * 1024);
        ScoreRange r;
        r.start    = srg_min;
        r.end      = SIZE_MAX;
        r.score    = sc + 1;
        r.tl_name  = "tcp";
        r.alg_name = "srg_ring";
        r.init     = [self](const ucc_coll_args_t &args, Team *t2,
                        Task **task) -> ucc_status_t {
            const ucc_generic_dt_ops_t *g =
                ucc_dt_generic_ops(args.src.info.datatype);
            if ((args.mask & UCC_COLL_ARGS_FIELD_ACTIVE_SET) || g) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            *task = new TcpReduceSrgTask(t2->ctx, self, args);
            return UCC_OK;
        };
        map.add(UCC_COLL_TYPE_REDUCE, UCC_MEMORY_TYPE_HOST, r);
    }
    add(UCC_COLL_TYPE_REDUCE, mk((TcpRootedTask *)nullptr));
    {
        size_t rs_min = Config::instance().get_size(
            "TL_TCP", "RS_RING_MIN", 64 * 1024);
        ScoreRange r;
        r.start    = rs_min;
        r.end      = SIZE_MAX;
        r.score    = sc + 1;
        r.tl_name  = "tcp";
        r.alg_name = "ring";
        r.init     = [self](const ucc_coll_args_t &args, Team *t2,
                        Task **task) -> ucc_status_t {
            if (!ucc_dt_is_predefined(args.dst.info.datatype) ||
                ((args.flags & UCC_COLL_ARGS_FLAG_IN_PLACE) &&
                 args.dst.info.count % t2->size != 0)) {
                return UCC_ERR_NOT_SUPPORTED; /* ragged: linear */
            }
            *task = new TcpReduceScatterRingTask(t2->ctx, self, args);
            return UCC_OK;
        };
        map.add(UCC_COLL_TYPE_REDUCE_SCATTER, UCC_MEMORY_TYPE_HOST, r);
        /* v-variant through the same ring engine (per-block v sizes,
         * reference reduce_scatterv ring role) */
        ScoreRange rv = r;
        rv.init       = [self](const ucc_coll_args_t &args, Team *t2,
                         Task **task) -> ucc_status_t {
            if (!ucc_dt_is_predefined(args.dst.info_v.datatype)) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            *task = new TcpReduceScatterRingTask(t2->ctx, self, args);
            return UCC_OK;
        };
        map.add(UCC_COLL_TYPE_REDUCE_SCATTERV, UCC_MEMORY_TYPE_HOST,
                rv);
        /* recursive halving below the ring band: log2 rounds beat the
         * ring's n-1 while the message sizes still fit latency-bound
         * traffic (reference reduce_scatter knomial role) */
        ScoreRange h;
        h.start    = 0;
        h.end      = rs_min;
        h.score    = sc + 1;
        h.tl_name  = "tcp";
        h.alg_name = "knomial";
        h.init     = [self](const ucc_coll_args_t &args, Team *t2,
                        Task **task) -> ucc_status_t {
            if (!ucc_dt_is_predefined(args.dst.info.datatype)) {
                return UCC_ERR_NOT_SUPPORTED;
            }
            *task = new TcpReduceScatterHalvingTask(t2->ctx, self, args);
            return UCC_OK;
        };
        map.add(UCC_COLL_TYPE_REDUCE_SCATTER, UCC_MEMORY_TYPE_HOST, h);
    }
    add(UCC_COLL_TYPE_REDUCE_SCATTER, mk((TcpRootedTask *)nullptr));
    add(UCC_COLL_TYPE_REDUCE_SCATTERV, mk((TcpRootedTask *)nullptr));
    {
        uint32_t radix = (uint32_t)Config::instance().get_int(
            "TL_TCP", "KN_RADIX", 4);
        for (auto ct : {UCC_COLL_TYPE_GATHER, UCC_COLL_TYPE_SCATTER}) {
            ScoreRange r;
            r.start    = 0;
            r.end      = SIZE_MAX;
            r.score    = sc + 1;
            r.tl_name  = "tcp";
            r.alg_name = "knomial";
            r.init     = [self, radix, ct](const ucc_coll_args_t &args,
                                       Team *t2,
                                       Task **task) -> ucc_status_t {
                const bool   root = t2->rank == (uint32_t)args.root;
                ucc_datatype_t dt =
                    ct == UCC_COLL_TYPE_GATHER
                        ? args.src.info.datatype
                        : (root ? args.src.info.datatype
                                : args.dst.info.datatype);
                if ((args.mask & UCC_COLL_ARGS_FIELD_ACTIVE_SET) ||
                    !ucc_dt_is_predefined(dt)) {
                    return UCC_ERR_NOT_SUPPORTED;
                }
                *task = new TcpGatherScatterKnTask(t2->ctx, self, args,
                                                   radix);
                return UCC_OK;
            };
            map.add(ct, UCC_MEMORY_TYPE_HOST, r);
        }
    }
    add(UCC_COLL_TYPE_GATHER, mk((TcpRootedTask *)nullptr));
    add(UCC_COLL_TYPE_GATHERV, mk((TcpRootedTask *)nullptr));
    add(UCC_COLL_TYPE_SCATTER, mk((TcpRootedTask *)nullptr));
    add(UCC_COLL_TYPE_SCATTERV, mk((TcpRootedTask *)nullptr));
}

} // namespace tcp

Tl *tl_tcp_iface() { return &tcp::g_tcp_tl; }

} // namespace ucc
