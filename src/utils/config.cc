#include "config.h"
#include "log.h"

#include <algorithm>
#include <cctype>
#include <cstdlib>
#include <cstring>
#include <fstream>

namespace ucc {

std::string Config::key(const std::string &component, const std::string &name)
{
    std::string k = "UCC_";
    if (!component.empty()) {
        k += component + "_";
    }
    k += name;
    std::transform(k.begin(), k.end(), k.begin(), ::toupper);
    return k;
}

Config::Config()
{
    /* core variables always present in `ucc_info -c` dumps */
    declare("", "LOG_LEVEL", "warn",
            "fatal|error|warn|info|debug|trace|trace_poll");
    declare("", "TUNE", "",
            "score-map override string: coll:msgrange:mem:@alg:score,...");
    declare("", "CONFIG_FILE", "ucc.conf", "ini config path");
    declare("", "LOCK_FREE_PROGRESS_Q", "1",
            "lock-free MPMC progress queue in THREAD_MULTIPLE");
    declare("", "COLL_TRACE", "0",
            "print per-collective algorithm selection at INFO");
    declare("", "FAKE_NODE_SPLIT", "0",
            "testing: spread contexts over k pseudo-nodes");
    /* ini file: UCC_CONFIG_FILE > ./ucc.conf ; lines "KEY = value",
     * '#'/';' comments, section headers ignored (keys are globally unique
     * via the UCC_ prefix convention). */
    const char *path = getenv("UCC_CONFIG_FILE");
    std::string fname = path ? path : "ucc.conf";
    std::ifstream f(fname);
    if (!f.good()) {
        return;
    }
    std::string line;
    while (std::getline(f, line)) {
        size_t h = line.find_first_of("#;");
        if (h != std::string::npos) {
            line = line.substr(0, h);
        }
        size_t eq = line.find('=');
        if (eq == std::string::npos) {
            continue;
        }
        auto trim = [](std::string s) {
            size_t b = s.find_first_not_of(" \t\r\n");
            size_t e = s.find_last_not_of(" \t\r\n");
            return b == std::string::npos ? std::string()
                                          : s.substr(b, e - b + 1);
        };
        std::string k = trim(line.substr(0, eq));
        std::string v = trim(line.substr(eq + 1));
        if (!k.empty()) {
            std::transform(k.begin(), k.end(), k.begin(), ::toupper);
            file_vals_[k] = v;
        }
    }
}

Config &Config::instance()
{
    static Config cfg;
    return cfg;
}

std::string Config::get(const std::string &component, const std::string &name,
                        const std::string &dflt)
{
    std::string k = key(component, name);
    auto        it = overrides_.find(k);
    if (it != overrides_.end()) {
        return it->second;
    }
    const char *e = getenv(k.c_str());
    if (e) {
        return e;
    }
    it = file_vals_.find(k);
    if (it != file_vals_.end()) {
        return it->second;
    }
    return dflt;
}

void Config::set(const std::string &component, const std::string &name,
                 const std::string &value)
{
    overrides_[key(component, name)] = value;
}

int64_t Config::get_int(const std::string &component, const std::string &name,
                        int64_t dflt)
{
    std::string v = get(component, name, "");
    if (v.empty()) {
        return dflt;
    }
    return strtoll(v.c_str(), nullptr, 0);
}

size_t parse_size(const std::string &s, size_t dflt)
{
    if (s.empty()) {
        return dflt;
    }
    char  *end = nullptr;
    double v   = strtod(s.c_str(), &end);
    if (end == s.c_str()) {
        return dflt;
    }
    switch (tolower(*end)) {
    case 'k': v *= 1024.0; break;
    case 'm': v *= 1024.0 * 1024.0; break;
    case 'g': v *= 1024.0 * 1024.0 * 1024.0; break;
    default: break;
    }
    return (size_t)v;
}

size_t Config::get_size(const std::string &component, const std::string &name,
                        size_t dflt)
{
    return parse_size(get(component, name, ""), dflt);
}

bool Config::get_bool(const std::string &component, const std::string &name,
                      bool dflt)
{
    std::string v = get(component, name, "");
    if (v.empty()) {
        return dflt;
    }
    return v == "1" || !strcasecmp(v.c_str(), "y") ||
           !strcasecmp(v.c_str(), "yes") || !strcasecmp(v.c_str(), "true") ||
           !strcasecmp(v.c_str(), "on");
}

double Config::get_double(const std::string &component,
                          const std::string &name, double dflt)
{
    std::string v = get(component, name, "");
    if (v.empty()) {
        return dflt;
    }
    return strtod(v.c_str(), nullptr);
}

void Config::declare(const std::string &component, const std::string &name,
                     const std::string &dflt, const std::string &doc)
{
    entries_.push_back({component, name, dflt, doc});
}

std::vector<Config::Entry> Config::entries() const { return entries_; }

} // namespace ucc
