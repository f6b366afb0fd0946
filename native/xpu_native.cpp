// kata_xpu_device_plugin_amd._native — C++ fast paths.
//
// The reference's runtime is native (Go); this extension keeps our hot
// paths native too (SURVEY.md §2.2 — runtime native where the reference's
// is):
//   * scan_pci(): the sysfs PCI walk behind discovery AND the per-call
//     Allocate revalidation (reference hot path does 2 sysfs reads per
//     device per Allocate — generic_device_plugin.go:329-338; one C++
//     openat/readlinkat pass is ~10× cheaper than the Python equivalent),
//   * revalidate_group(): single-group check used by Allocate,
//   * select_preferred(): exact (hive,numa)-bucket composition search for
//     GetPreferredAllocation (mirrors topology/hive.py::preferred_sets).
//
// Build: g++ via setup.py (no HIP here — this runs on any node, GPU-less
// control planes included).

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <algorithm>
#include <cstdint>
#include <cstring>
#include <dirent.h>
#include <functional>
#include <fcntl.h>
#include <map>
#include <optional>
#include <string>
#include <unistd.h>
#include <vector>

namespace py = pybind11;

namespace {

// Read a small sysfs attribute relative to dirfd; returns empty on error.
static std::string read_attr(int dirfd, const char* name) {
    int fd = ::openat(dirfd, name, O_RDONLY | O_CLOEXEC);
    if (fd < 0) return {};
    char buf[256];
    ssize_t n = ::read(fd, buf, sizeof(buf) - 1);
    ::close(fd);
    if (n <= 0) return {};
    buf[n] = 0;
    // strip trailing whitespace/newline
    while (n > 0 && (buf[n - 1] == '\n' || buf[n - 1] == ' ')) buf[--n] = 0;
    return std::string(buf, (size_t)n);
}

static std::optional<long> parse_hex(const std::string& s) {
    if (s.empty()) return std::nullopt;
    errno = 0;
    char* end = nullptr;
    long v = std::strtol(s.c_str(), &end, 16);
    if (errno || end == s.c_str()) return std::nullopt;
    return v;
}

static std::optional<long> parse_int(const std::string& s) {
    if (s.empty()) return std::nullopt;
    errno = 0;
    char* end = nullptr;
    long v = std::strtol(s.c_str(), &end, 10);
    if (errno || end == s.c_str()) return std::nullopt;
    return v;
}

// Basename of a symlink target relative to dirfd ("" on error).
static std::string link_base(int dirfd, const char* name) {
    char buf[512];
    ssize_t n = ::readlinkat(dirfd, name, buf, sizeof(buf) - 1);
    if (n <= 0) return {};
    buf[n] = 0;
    const char* slash = std::strrchr(buf, '/');
    return std::string(slash ? slash + 1 : buf);
}

struct Fn {
    std::string bdf;
    long vendor = 0, device = 0, class_code = 0;
    std::string driver, iommu_group, physfn;
    long numa = -1, totalvfs = 0, numvfs = 0;
};

static std::optional<Fn> read_function(int base_fd, const std::string& bdf,
                                       const std::vector<long>& vendors) {
    int dfd = ::openat(base_fd, bdf.c_str(), O_RDONLY | O_DIRECTORY | O_CLOEXEC);
    if (dfd < 0) return std::nullopt;
    Fn fn;
    fn.bdf = bdf;
    auto vendor = parse_hex(read_attr(dfd, "vendor"));
    if (!vendor ||
        std::find(vendors.begin(), vendors.end(), *vendor) == vendors.end()) {
        ::close(dfd);
        return std::nullopt;
    }
    fn.vendor = *vendor;
    auto device = parse_hex(read_attr(dfd, "device"));
    if (!device) { ::close(dfd); return std::nullopt; }
    fn.device = *device;
    fn.class_code = parse_hex(read_attr(dfd, "class")).value_or(0);
    fn.driver = link_base(dfd, "driver");
    fn.iommu_group = link_base(dfd, "iommu_group");
    fn.physfn = link_base(dfd, "physfn");
    fn.numa = parse_int(read_attr(dfd, "numa_node")).value_or(-1);
    fn.totalvfs = parse_int(read_attr(dfd, "sriov_totalvfs")).value_or(0);
    fn.numvfs = parse_int(read_attr(dfd, "sriov_numvfs")).value_or(0);
    ::close(dfd);
    return fn;
}

// scan_pci(devices_dir, vendor_allowlist) → list of dicts matching
// discovery.sysfs.PCIFunction's constructor kwargs.
py::list scan_pci(const std::string& devices_dir, const std::vector<long>& vendors) {
    py::list out;
    int base_fd = ::open(devices_dir.c_str(), O_RDONLY | O_DIRECTORY | O_CLOEXEC);
    if (base_fd < 0) return out;
    std::vector<std::string> names;
    {
        DIR* d = ::fdopendir(::dup(base_fd));
        if (d) {
            while (dirent* ent = ::readdir(d)) {
                if (ent->d_name[0] == '.') continue;
                names.emplace_back(ent->d_name);
            }
            ::closedir(d);
        }
    }
    std::sort(names.begin(), names.end());
    for (const auto& bdf : names) {
        auto fn = read_function(base_fd, bdf, vendors);
        if (!fn) continue;
        py::dict d;
        d["bdf"] = fn->bdf;
        d["vendor"] = fn->vendor;
        d["device"] = fn->device;
        d["class_code"] = fn->class_code;
        d["driver"] = fn->driver.empty() ? py::object(py::none())
                                         : py::object(py::str(fn->driver));
        d["iommu_group"] = fn->iommu_group.empty()
                               ? py::object(py::none())
                               : py::object(py::str(fn->iommu_group));
        d["numa_node"] = fn->numa;
        d["sriov_totalvfs"] = fn->totalvfs;
        d["sriov_numvfs"] = fn->numvfs;
        d["physfn_bdf"] = fn->physfn.empty() ? py::object(py::none())
                                             : py::object(py::str(fn->physfn));
        out.append(std::move(d));
    }
    ::close(base_fd);
    return out;
}

// revalidate_group(devices_dir, group_id, bdfs, vendors, required_driver)
// → "" if OK else a reason string. The Allocate() hot path.
std::string revalidate_group(const std::string& devices_dir,
                             const std::string& group_id,
                             const std::vector<std::string>& bdfs,
                             const std::vector<long>& vendors,
                             const std::string& required_driver) {
    int base_fd = ::open(devices_dir.c_str(), O_RDONLY | O_DIRECTORY | O_CLOEXEC);
    if (base_fd < 0) return "cannot open " + devices_dir;
    std::string err;
    for (const auto& bdf : bdfs) {
        int dfd = ::openat(base_fd, bdf.c_str(), O_RDONLY | O_DIRECTORY | O_CLOEXEC);
        if (dfd < 0) { err = "device " + bdf + " vanished"; break; }
        if (link_base(dfd, "iommu_group") != group_id) {
            err = "device " + bdf + " no longer in IOMMU group " + group_id;
            ::close(dfd);
            break;
        }
        auto vendor = parse_hex(read_attr(dfd, "vendor"));
        if (!vendor ||
            std::find(vendors.begin(), vendors.end(), *vendor) == vendors.end()) {
            err = "device " + bdf + " vendor not allowed";
            ::close(dfd);
            break;
        }
        if (link_base(dfd, "driver") != required_driver) {
            err = "device " + bdf + " not bound to " + required_driver;
            ::close(dfd);
            break;
        }
        ::close(dfd);
    }
    ::close(base_fd);
    return err;
}

// Batched form: validate several groups with ONE directory fd — the
// Allocate() hot path calls this once per container request.
std::string revalidate_groups(
    const std::string& devices_dir,
    const std::vector<std::pair<std::string, std::vector<std::string>>>& groups,
    const std::vector<long>& vendors,
    const std::string& required_driver) {
    int base_fd = ::open(devices_dir.c_str(), O_RDONLY | O_DIRECTORY | O_CLOEXEC);
    if (base_fd < 0) return "cannot open " + devices_dir;
    std::string err;
    for (const auto& [gid, bdfs] : groups) {
        for (const auto& bdf : bdfs) {
            int dfd = ::openat(base_fd, bdf.c_str(),
                               O_RDONLY | O_DIRECTORY | O_CLOEXEC);
            if (dfd < 0) { err = "device " + bdf + " vanished"; goto done; }
            if (link_base(dfd, "iommu_group") != gid) {
                ::close(dfd);
                err = "device " + bdf + " no longer in IOMMU group " + gid;
                goto done;
            }
            auto vendor = parse_hex(read_attr(dfd, "vendor"));
            if (!vendor || std::find(vendors.begin(), vendors.end(), *vendor) ==
                               vendors.end()) {
                ::close(dfd);
                err = "device " + bdf + " vendor not allowed";
                goto done;
            }
            if (link_base(dfd, "driver") != required_driver) {
                ::close(dfd);
                err = "device " + bdf + " not bound to " + required_driver;
                goto done;
            }
            ::close(dfd);
        }
    }
done:
    ::close(base_fd);
    return err;
}

// ---------------------------------------------------------------------------
// Preferred-set selection: exact composition search over (hive, numa)
// buckets; must mirror topology/hive.py::preferred_sets (tested for parity).
// ---------------------------------------------------------------------------

// Tier-separated weights (mirror topology/hive.py): one xGMI pair beats
// every possible NUMA pair even on a 1024-device node, etc.
constexpr int64_t W_XGMI = 1000000000000LL;
constexpr int64_t W_NUMA = 1000000;
// Search-effort cap (mirror hive.py::MAX_SEARCH_NODES): past this the
// selector keeps the concentration-greedy result of the first descent.
constexpr long MAX_SEARCH_NODES = 50000;

struct Bucket {
    std::string hive;
    long numa;
    std::vector<std::string> ids;
};

std::vector<std::string> select_preferred(
    const std::map<std::string, std::pair<std::string, long>>& locality,
    // device id → (hive key, numa)
    const std::vector<std::string>& available,
    const std::vector<std::string>& must_include,
    int size) {
    // dedupe, preserve order
    std::vector<std::string> avail;
    for (const auto& d : available)
        if (std::find(avail.begin(), avail.end(), d) == avail.end())
            avail.push_back(d);
    std::vector<std::string> must;
    for (const auto& d : must_include)
        if (std::find(must.begin(), must.end(), d) == must.end())
            must.push_back(d);
    if (size <= 0 || (size_t)size > avail.size()) return {};
    for (const auto& m : must)
        if (std::find(avail.begin(), avail.end(), m) == avail.end()) return {};
    if ((int)must.size() >= size)
        return std::vector<std::string>(must.begin(), must.begin() + size);

    const int need = size - (int)must.size();
    std::map<std::pair<std::string, long>, Bucket> bmap;
    for (const auto& d : avail) {
        if (std::find(must.begin(), must.end(), d) != must.end()) continue;
        auto it = locality.find(d);
        std::pair<std::string, long> key =
            it != locality.end() ? it->second : std::make_pair(std::string(), -1L);
        auto& b = bmap[key];
        b.hive = key.first;
        b.numa = key.second;
        b.ids.push_back(d);
    }
    std::vector<Bucket> buckets;
    for (auto& kv : bmap) {
        std::sort(kv.second.ids.begin(), kv.second.ids.end(),
                  [](const std::string& a, const std::string& b) {
                      return a.size() != b.size() ? a.size() < b.size() : a < b;
                  });
        buckets.push_back(std::move(kv.second));
    }
    // Cap-DESCENDING: the first DFS path (max take from the biggest
    // buckets) is the concentration optimum, so the branch-and-bound
    // prunes nearly everything else; best-fit preference comes from the
    // packing term, not search order (mirror of hive.py).
    // secondary sort by NUMA then hive so indistinguishable buckets sit
    // adjacent (symmetry reduction below; mirror of hive.py)
    std::sort(buckets.begin(), buckets.end(), [](const Bucket& a, const Bucket& b) {
        if (a.ids.size() != b.ids.size()) return a.ids.size() > b.ids.size();
        if (a.numa != b.numa) return a.numa < b.numa;
        return a.hive < b.hive;
    });

    const int nb = (int)buckets.size();
    // affinity of each bucket to the must-set
    std::vector<int64_t> affinity(nb, 0);
    for (int i = 0; i < nb; i++) {
        for (const auto& m : must) {
            auto it = locality.find(m);
            std::string mh = it != locality.end() ? it->second.first : "";
            long mn = it != locality.end() ? it->second.second : -1;
            if (!buckets[i].hive.empty() && mh == buckets[i].hive)
                affinity[i] += W_XGMI;
            else if (mn == buckets[i].numa && buckets[i].numa != -1)
                affinity[i] += W_NUMA;
        }
    }
    std::vector<int> caps(nb), suffix(nb + 1, 0);
    for (int i = 0; i < nb; i++) caps[i] = (int)buckets[i].ids.size();
    for (int i = nb - 1; i >= 0; i--) suffix[i] = suffix[i + 1] + caps[i];

    // ---- symmetry reduction (mirror of hive.py): whole-hive buckets
    // with equal capacity, same NUMA, same must-affinity, whose hives
    // live entirely in their one bucket, are interchangeable — search
    // only canonical non-increasing takes across equivalent runs.
    std::map<std::string, int> hive_span;
    for (int i = 0; i < nb; i++)
        if (!buckets[i].hive.empty()) hive_span[buckets[i].hive] += caps[i];
    std::vector<char> equiv(nb, 0);
    for (int i = 1; i < nb; i++) {
        const Bucket& a = buckets[i - 1];
        const Bucket& b = buckets[i];
        equiv[i] = !a.hive.empty() && !b.hive.empty() &&
                   caps[i] == caps[i - 1] && a.numa == b.numa &&
                   affinity[i] == affinity[i - 1] &&
                   hive_span[a.hive] == caps[i - 1] &&
                   hive_span[b.hive] == caps[i];
    }

    // packing-hive key per bucket: hives pack together; hiveless pack by numa
    auto pkey = [&](int i) -> std::pair<std::string, long> {
        if (!buckets[i].hive.empty()) return {buckets[i].hive, -1};
        return {std::string("numa"), buckets[i].numa};
    };
    std::map<std::pair<std::string, long>, int> hive_free;
    for (int i = 0; i < nb; i++) hive_free[pkey(i)] += caps[i];

    // suffix max affinity and per-suffix remaining hive capacities, for
    // the admissible upper bound (mirror of hive.py's branch-and-bound).
    std::vector<int64_t> max_aff_suffix(nb + 1, 0);
    for (int i = nb - 1; i >= 0; i--)
        max_aff_suffix[i] = std::max(max_aff_suffix[i + 1], affinity[i]);
    std::vector<std::map<std::string, int>> hive_cap_suffix(nb + 1);
    std::vector<std::map<long, int>> numa_cap_suffix(nb + 1);
    for (int i = nb - 1; i >= 0; i--) {
        hive_cap_suffix[i] = hive_cap_suffix[i + 1];
        numa_cap_suffix[i] = numa_cap_suffix[i + 1];
        if (!buckets[i].hive.empty()) hive_cap_suffix[i][buckets[i].hive] += caps[i];
        if (buckets[i].numa != -1) numa_cap_suffix[i][buckets[i].numa] += caps[i];
    }

    std::vector<int> take(nb, 0), best_take;
    int64_t best_score = -1;
    int64_t best_pack = 0;
    std::map<std::string, int> taken_hive;
    std::map<long, int> taken_numa;

    auto bucket_pair = [&](int i, int c) -> int64_t {
        int64_t w = !buckets[i].hive.empty() ? W_XGMI
                    : buckets[i].numa != -1 ? W_NUMA : 0;
        return w * c * (c - 1) / 2;
    };
    auto packing = [&]() -> int64_t {
        std::map<std::pair<std::string, long>, int> taken;
        for (int i = 0; i < nb; i++)
            if (take[i]) taken[pkey(i)] += take[i];
        int64_t p = 0;
        for (auto& kv : taken) p -= hive_free[kv.first] - kv.second;
        return p;
    };
    // Admissible bound on future pairs: marginal gain of the j-th future
    // device in group g (holding t) is t+j; the sum of the `left` largest
    // marginals over-estimates every feasible placement. Each group's
    // marginals are the integer range [t, t+cap), so the top-k sum is a
    // THRESHOLD binary search over ranges — O(H log V) per node instead
    // of materializing every marginal (mirror of hive.py::_topk_relaxed).
    auto topk_ranges = [](const std::vector<std::pair<int, int>>& ranges,
                          int left) -> int64_t {
        if (ranges.empty() || left <= 0) return 0;
        int lo = 0, hi = 0;
        for (auto& r : ranges) hi = std::max(hi, r.first + r.second);
        while (lo < hi) {
            int mid = (lo + hi + 1) / 2;
            long cnt = 0;
            for (auto& r : ranges) {
                int top = r.first + r.second;
                if (top > mid) cnt += top - std::max(mid, r.first);
            }
            if (cnt >= left) lo = mid; else hi = mid - 1;
        }
        const int T = lo;
        int64_t total = 0;
        long cnt_above = 0;
        for (auto& r : ranges) {
            int top = r.first + r.second;
            int start = std::max(T + 1, r.first);
            if (top > start) {
                int64_t k = top - start;
                total += k * (start + top - 1) / 2;
                cnt_above += k;
            }
        }
        total += (int64_t)(left - cnt_above) * T;
        return total;
    };
    // EXACT tier max when every remaining group starts empty: fill
    // largest-capacity groups fully (exchange argument) — tight, so the
    // first greedy descent prunes symmetric requests at the root.
    auto concentration_exact = [](const std::vector<int>& caps_desc,
                                  int left) -> int64_t {
        int64_t total = 0;
        for (int cap : caps_desc) {
            if (left <= 0) break;
            int64_t c = cap < left ? cap : left;
            total += c * (c - 1) / 2;
            left -= (int)c;
        }
        return total;
    };
    // per-suffix cap-descending group capacity lists
    std::vector<std::vector<int>> hive_caps_desc(nb + 1), numa_caps_desc(nb + 1);
    for (int i = 0; i <= nb; i++) {
        for (auto& kv : hive_cap_suffix[i]) hive_caps_desc[i].push_back(kv.second);
        for (auto& kv : numa_cap_suffix[i]) numa_caps_desc[i].push_back(kv.second);
        std::sort(hive_caps_desc[i].rbegin(), hive_caps_desc[i].rend());
        std::sort(numa_caps_desc[i].rbegin(), numa_caps_desc[i].rend());
    }
    auto xgmi_upper = [&](int i, int left) -> int64_t {
        bool untouched = true;
        for (auto& kv : hive_cap_suffix[i]) {
            auto it = taken_hive.find(kv.first);
            if (it != taken_hive.end() && it->second > 0) { untouched = false; break; }
        }
        if (untouched) return concentration_exact(hive_caps_desc[i], left);
        std::vector<std::pair<int, int>> ranges;
        for (auto& kv : hive_cap_suffix[i]) {
            auto it = taken_hive.find(kv.first);
            int t = it != taken_hive.end() ? it->second : 0;
            ranges.emplace_back(t, kv.second);
        }
        return topk_ranges(ranges, left);
    };
    // same marginal-gain bound for the NUMA tier
    auto numa_upper = [&](int i, int left) -> int64_t {
        bool untouched = true;
        for (auto& kv : numa_cap_suffix[i]) {
            auto it = taken_numa.find(kv.first);
            if (it != taken_numa.end() && it->second > 0) { untouched = false; break; }
        }
        if (untouched) return concentration_exact(numa_caps_desc[i], left);
        std::vector<std::pair<int, int>> ranges;
        for (auto& kv : numa_cap_suffix[i]) {
            auto it = taken_numa.find(kv.first);
            int t = it != taken_numa.end() ? it->second : 0;
            ranges.emplace_back(t, kv.second);
        }
        return topk_ranges(ranges, left);
    };

    long nodes_visited = 0;
    std::function<void(int, int, int64_t)> dfs = [&](int i, int left, int64_t acc) {
        // Budget only bites once a complete solution exists: the first
        // greedy descent (<= nb nodes) always finishes, so a feasible
        // request never degrades to {} (mirrors hive.py).
        if (++nodes_visited > MAX_SEARCH_NODES && !best_take.empty()) return;
        if (left == 0) {
            int64_t pack = packing();
            if (best_take.empty() || acc > best_score ||
                (acc == best_score && pack > best_pack)) {
                best_score = acc;
                best_pack = pack;
                best_take = take;
            }
            return;
        }
        if (i >= nb || suffix[i] < left) return;
        if (!best_take.empty()) {
            int64_t ub = acc + W_XGMI * xgmi_upper(i, left) +
                         W_NUMA * numa_upper(i, left) +
                         max_aff_suffix[i] * left;
            if (ub < best_score || (ub == best_score && best_pack == 0)) return;
        }
        const std::string& hive = buckets[i].hive;
        const long numa = buckets[i].numa;
        int cmax = std::min(caps[i], left);
        if (equiv[i]) cmax = std::min(cmax, take[i - 1]);  // canonical order
        for (int c = cmax; c >= 0; c--) {
            int64_t inc = bucket_pair(i, c) + affinity[i] * c;
            if (c) {
                if (!hive.empty()) inc += W_XGMI * c * taken_hive[hive];
                if (numa != -1) inc += W_NUMA * c * taken_numa[numa];
                take[i] = c;
                if (!hive.empty()) taken_hive[hive] += c;
                if (numa != -1) taken_numa[numa] += c;
            }
            dfs(i + 1, left - c, acc + inc);
            if (c) {
                take[i] = 0;
                if (!hive.empty()) taken_hive[hive] -= c;
                if (numa != -1) taken_numa[numa] -= c;
            }
        }
    };
    dfs(0, need, 0);
    if (best_take.empty() && need > 0) return {};

    std::vector<std::string> chosen = must;
    for (int i = 0; i < nb; i++)
        for (int c = 0; c < best_take[i]; c++) chosen.push_back(buckets[i].ids[c]);
    return chosen;
}

}  // namespace

PYBIND11_MODULE(_native, m) {
    m.doc() = "Native fast paths for kata-xpu-device-plugin-amd";
    m.def("scan_pci", &scan_pci, py::arg("devices_dir"), py::arg("vendors"),
          "Walk a PCI devices dir, return vendor-matched functions");
    m.def("revalidate_group", &revalidate_group, py::arg("devices_dir"),
          py::arg("group_id"), py::arg("bdfs"), py::arg("vendors"),
          py::arg("required_driver"),
          "Allocate-path group revalidation; empty string = OK");
    m.def("revalidate_groups", &revalidate_groups, py::arg("devices_dir"),
          py::arg("groups"), py::arg("vendors"), py::arg("required_driver"),
          "Batched group revalidation (one dirfd); empty string = OK");
    m.def("select_preferred", &select_preferred, py::arg("locality"),
          py::arg("available"), py::arg("must_include"), py::arg("size"),
          "Exact xGMI/NUMA bucket-composition preferred-set selection");
}
