/* Tagged nonblocking point-to-point engine over TCP sockets: the
 * transport under tl/tcp's collective algorithms.
 *
 * Reference parity: the role of UCX tag-matching under tl/ucp
 * (tl_ucp_sendrecv.h tag layout, posted/completed counter completion,
 * tl_ucp_coll.h:124-199) — re-derived on plain sockets: full-mesh
 * nonblocking connections, 16-byte wire header {tag, len}, expected-recv
 * matching with an unexpected-message queue, progress() drives partial
 * reads/writes. Single-node testing uses 127.0.0.1; the same code spans
 * nodes (the only inter-node path MI355X boxes need besides RCCL/IB). */
#ifndef UCC_AMD_TL_TCP_P2P_H_
#define UCC_AMD_TL_TCP_P2P_H_

#include <cstdint>
#include <cstring>
#include <deque>
#include <list>
#include <map>
#include <vector>

namespace ucc {
namespace tcp {

struct WireHdr {
    uint64_t tag;
    uint64_t len;
};

struct SendOp {
    uint64_t             tag;
    const uint8_t       *buf;
    size_t               len;
    size_t               off = 0; /* bytes of hdr+payload written */
    bool                 done = false;
};

struct RecvOp {
    uint64_t  tag;
    uint8_t  *buf;
    size_t    len;
    bool      done = false;
};

struct UnexpMsg {
    uint64_t             tag;
    std::vector<uint8_t> data;
    bool                 done = false; /* payload fully received */
};

/* One peer connection. */
class Conn {
  public:
    int  fd = -1;
    /* ---- send side */
    std::deque<SendOp *> sendq;
    /* ---- recv side */
    std::deque<RecvOp *>  recvq;     /* posted expectations (FIFO/tag)  */
    /* std::list (not deque): take_unexp() erases from the middle while
     * cur_un may point into another element's payload vector — list
     * erase leaves every other element (and cur_un) stable. */
    std::list<UnexpMsg>   unexp;
    /* in-flight incoming message state */
    WireHdr               hdr{};
    size_t                hdr_got  = 0;
    std::vector<uint8_t> *cur_un   = nullptr; /* unexpected payload dst */
    RecvOp               *cur_recv = nullptr; /* matched expected recv  */
    size_t                pay_got  = 0;

    void progress();

    /* steal a COMPLETED unexpected message by tag (dynamic-size recv:
     * the sender determines the length — used by the pack/unpack
     * generic-datatype path). */
    bool take_unexp(uint64_t tag, std::vector<uint8_t> *out)
    {
        for (auto it = unexp.begin(); it != unexp.end(); ++it) {
            if (it->tag == tag && it->done) {
                *out = std::move(it->data);
                unexp.erase(it);
                return true;
            }
        }
        return false;
    }

  private:
    void match_header();
};

} // namespace tcp
} // namespace ucc

#endif
