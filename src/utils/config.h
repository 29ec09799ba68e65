/* ucc_amd — table-driven env + file configuration.
 * Reference parity: UCS-wrapped parser (utils/ucc_parser.{h,c}, ini.c,
 * per-component tables, ucc.conf search). Native re-implementation: a global
 * key/value store seeded from an optional ini file (UCC_CONFIG_FILE or
 * ./ucc.conf) and overridden by UCC_* environment variables; components
 * declare typed entries with defaults + docstrings so `ucc_info -caf`-style
 * dumps work. */
#ifndef UCC_AMD_CONFIG_H_
#define UCC_AMD_CONFIG_H_

#include <cstdint>
#include <map>
#include <string>
#include <vector>

namespace ucc {

class Config {
  public:
    /* Singleton, seeded on first use. */
    static Config &instance();

    /* "component" is e.g. "" (core), "TL_CDNA4", "EC_HIP". Lookup order:
     * env UCC_<COMPONENT>_<NAME> (or UCC_<NAME> for core) -> config file
     * entry of the same key -> registered default. */
    std::string get(const std::string &component, const std::string &name,
                    const std::string &dflt);
    int64_t     get_int(const std::string &component, const std::string &name,
                        int64_t dflt);
    size_t      get_size(const std::string &component, const std::string &name,
                         size_t dflt); /* supports k/m/g suffixes */
    bool        get_bool(const std::string &component, const std::string &name,
                         bool dflt);
    double      get_double(const std::string &component,
                           const std::string &name, double dflt);

    /* Runtime override (ucc_context_config_modify). */
    void set(const std::string &component, const std::string &name,
             const std::string &value);

    /* Declare an entry for documentation dumps. */
    void declare(const std::string &component, const std::string &name,
                 const std::string &dflt, const std::string &doc);
    struct Entry {
        std::string component, name, dflt, doc;
    };
    std::vector<Entry> entries() const;

    static std::string key(const std::string &component,
                           const std::string &name);

  private:
    Config();
    std::map<std::string, std::string> file_vals_;
    std::map<std::string, std::string> overrides_;
    std::vector<Entry>                 entries_;
};

size_t parse_size(const std::string &s, size_t dflt);

} // namespace ucc

#endif
