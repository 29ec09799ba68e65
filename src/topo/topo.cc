/* See topo.h. */
#include "topo.h"

#include <algorithm>
#include <mutex>

#ifdef UCC_AMD_HAS_HIP
#include <hip/hip_runtime.h>
#endif

namespace ucc {
namespace topo {

Sbgp build_sbgp(const Team *team, SbgpType type)
{
    Sbgp s;
    s.type = type;
    const auto &procs = team->procs;
    const uint32_t n  = team->size;
    switch (type) {
    case SbgpType::FULL:
        for (uint32_t r = 0; r < n; r++) {
            s.ranks.push_back(r);
        }
        s.my_idx = (int)team->rank;
        break;
    case SbgpType::NODE: {
        uint64_t my_host = procs[team->rank].host_hash;
        for (uint32_t r = 0; r < n; r++) {
            if (procs[r].host_hash == my_host) {
                if (r == team->rank) {
                    s.my_idx = (int)s.ranks.size();
                }
                s.ranks.push_back(r);
            }
        }
        break;
    }
    case SbgpType::SOCKET:
    case SbgpType::NUMA: {
        /* intra-node grouping by socket/NUMA id; unknown ids (-1)
         * degrade to one NODE-wide group (correct, untuned) */
        uint64_t my_host = procs[team->rank].host_hash;
        auto     key     = [&](uint32_t r) {
            return type == SbgpType::SOCKET ? procs[r].socket_id
                                                : procs[r].numa_id;
        };
        int my_key = key(team->rank);
        for (uint32_t r = 0; r < n; r++) {
            if (procs[r].host_hash == my_host && key(r) == my_key) {
                if (r == team->rank) {
                    s.my_idx = (int)s.ranks.size();
                }
                s.ranks.push_back(r);
            }
        }
        break;
    }
    case SbgpType::SOCKET_LEADERS:
    case SbgpType::NUMA_LEADERS: {
        /* lowest rank per distinct (socket|numa) id ON MY NODE */
        uint64_t my_host = procs[team->rank].host_hash;
        auto     key     = [&](uint32_t r) {
            return type == SbgpType::SOCKET_LEADERS
                           ? procs[r].socket_id
                           : procs[r].numa_id;
        };
        std::vector<int> seen;
        for (uint32_t r = 0; r < n; r++) {
            if (procs[r].host_hash != my_host) {
                continue;
            }
            int k = key(r);
            if (std::find(seen.begin(), seen.end(), k) == seen.end()) {
                seen.push_back(k);
                if (r == team->rank) {
                    s.my_idx = (int)s.ranks.size();
                }
                s.ranks.push_back(r);
            }
        }
        break;
    }
    case SbgpType::NODE_LEADERS: {
        /* lowest team rank per distinct host hash, in rank order */
        std::vector<uint64_t> seen;
        for (uint32_t r = 0; r < n; r++) {
            uint64_t h = procs[r].host_hash;
            if (std::find(seen.begin(), seen.end(), h) == seen.end()) {
                seen.push_back(h);
                if (r == team->rank) {
                    s.my_idx = (int)s.ranks.size();
                }
                s.ranks.push_back(r);
            }
        }
        break;
    }
    }
    return s;
}

bool team_same_cpu(const Team *team)
{
    for (const auto &p : team->procs) {
        if (p.cpu_hash != team->procs[0].cpu_hash) {
            return false;
        }
    }
    return true;
}

const GpuLinks &gpu_links()
{
    static GpuLinks    g;
    static std::once_flag once;
    std::call_once(once, [] {
#ifdef UCC_AMD_HAS_HIP
        int ndev = 0;
        if (hipGetDeviceCount(&ndev) != hipSuccess) {
            return;
        }
        g.ndev = ndev;
        g.peer.assign(ndev, std::vector<int>(ndev, 0));
        g.hops.assign(ndev, std::vector<int>(ndev, -1));
        for (int i = 0; i < ndev; i++) {
            g.peer[i][i] = 1;
            g.hops[i][i] = 0;
            for (int j = 0; j < ndev; j++) {
                if (i == j) {
                    continue;
                }
                int can = 0;
                if (hipDeviceCanAccessPeer(&can, i, j) == hipSuccess) {
                    g.peer[i][j] = can;
                }
                uint32_t hops = 0, type = 0;
                if (hipExtGetLinkTypeAndHopCount(i, j, &type, &hops) ==
                    hipSuccess) {
                    g.hops[i][j] = (int)hops;
                }
            }
        }
#endif
    });
    return g;
}

} // namespace topo
} // namespace ucc
