/* ucc_info — version / component / config introspection.
 * Reference parity: tools/info/ucc_info.c (-v version, -c configs,
 * -s score map of a simulated 2-rank team). */
#include <cstdio>
#include <cstring>
#include <vector>

#include "../src/api/ucc.h"
#include "../src/core/core.h"
#include "../src/utils/config.h"

using namespace ucc;

static void print_version()
{
    printf("# UCC-AMD (MI355X-native) version %s\n",
           ucc_get_version_string());
    printf("#  arch: gfx950 (CDNA4), transports: ");
    ensure_builtin_tls();
    for (auto *tl : tl_registry()) {
        printf("%s ", tl->name());
    }
    printf("\n");
}

static void print_configs()
{
    /* touch the main tables so defaults are declared */
    Config &c = Config::instance();
    printf("#\n# configuration variables (UCC_<COMPONENT>_<NAME>)\n#\n");
    auto entries = c.entries();
    for (auto &e : entries) {
        std::string key = Config::key(e.component, e.name);
        printf("UCC_%s=%s\n", key.c_str(),
               c.get(e.component, e.name, e.dflt).c_str());
        if (!e.doc.empty()) {
            printf("# %s\n", e.doc.c_str());
        }
    }
    if (entries.empty()) {
        printf("# (no components touched their config yet)\n");
    }
}

int main(int argc, char **argv)
{
    bool ver = true, cfg = false;
    for (int i = 1; i < argc; i++) {
        if (!strcmp(argv[i], "-c") || !strcmp(argv[i], "-caf")) {
            cfg = true;
        } else if (!strcmp(argv[i], "-v")) {
            ver = true;
        } else if (!strcmp(argv[i], "-h")) {
            printf("ucc_info [-v] [-c]\n");
            return 0;
        }
    }
    if (ver) {
        print_version();
    }
    if (cfg) {
        print_configs();
    }
    return 0;
}
