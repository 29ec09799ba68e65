/* Topology component: process placement subgroups + the xGMI link matrix
 * of the local node's GPUs.
 *
 * Reference parity: src/components/topo/ (ucc_topo.h team topo + sbgps,
 * ucc_sbgp.h subgroup kinds, topo/cuda NVML link discovery) re-derived
 * for one MI355X node: processes are grouped by host hash (NODE /
 * NODE_LEADERS / FULL sbgps — the ones single-node + future multi-node
 * composition needs), and GPU connectivity comes from
 * hipDeviceCanAccessPeer + hipExtGetLinkTypeAndHopCount instead of NVML
 * (8 fully-connected GPUs, 7 xGMI links each). */
#ifndef UCC_AMD_TOPO_H_
#define UCC_AMD_TOPO_H_

#include <cstdint>
#include <vector>

#include "../core/core.h"

namespace ucc {
namespace topo {

enum class SbgpType {
    NODE,           /* my node's ranks                       */
    NODE_LEADERS,   /* lowest rank of each node              */
    SOCKET,         /* my node's ranks on my CPU socket      */
    SOCKET_LEADERS, /* lowest rank of each socket on my node */
    NUMA,           /* my node's ranks on my NUMA domain     */
    NUMA_LEADERS,   /* lowest rank of each NUMA on my node   */
    FULL,
};

/* CPU-model consensus across the team (reference ucc_topo.h:88-95):
 * symmetric tuning defaults are only safe when every rank runs on the
 * same CPU model. */
bool team_same_cpu(const Team *team);

struct Sbgp {
    SbgpType              type;
    std::vector<uint32_t> ranks;   /* team ranks, sorted          */
    int                   my_idx = -1; /* -1 if not a member      */
};

/* Build a subgroup for `team` from its proc infos. */
Sbgp build_sbgp(const Team *team, SbgpType type);

/* Per-device link info of the local node (HIP runtime query). */
struct GpuLinks {
    int ndev = 0;
    /* [i][j]: 1 if peer access possible (xGMI reachable), 0 otherwise */
    std::vector<std::vector<int>> peer;
    /* [i][j]: link hop count (1 = direct xGMI) or -1 */
    std::vector<std::vector<int>> hops;
};
const GpuLinks &gpu_links();

} // namespace topo
} // namespace ucc

#endif
