// sched-pipeline: optimal-throughput pipeline partitioner (MI355X rebuild).
//
// CLI- and YAML-compatible re-implementation of the reference's native
// scheduler (reference src-native/sched-pipeline.cpp:132-172 CLI flags;
// schedule.cpp:92-267 algorithm): reads models.yml + device_types.yml +
// devices.yml, minimizes the maximum per-stage cost (compute or comm time)
// over all contiguous-layer partitions and device assignments via dynamic
// programming over (last layer, multiset of used device types, current
// type), then assigns concrete hosts round-robin per type and prints the
// schedule as YAML: "- host: [layer_l, layer_r]" (1-indexed inclusive).
//
// MI355X deltas vs the reference: the dtype table supports bf16/fp16/fp64
// (the reference rejects anything but torch.float32,
// sched-pipeline.cpp:227-232), and no yaml-cpp dependency (minyaml.h).
#include <getopt.h>

#include <algorithm>
#include <cmath>
#include <cstdio>
#include <iostream>
#include <limits>
#include <map>
#include <optional>
#include <string>
#include <vector>

#include "minyaml.h"

namespace {

constexpr double INF = std::numeric_limits<double>::infinity();

struct DeviceType {
    std::string name;
    double mem_mb = 0;
    double bw_mbps = 0;
    std::vector<double> time_s;  // per-layer profiled compute time
    std::vector<std::string> hosts;
};

struct ModelInfo {
    long layers = 0;
    long parameters_in = 0;
    std::vector<long> parameters_out;
    std::vector<double> mem_mb;
};

struct Options {
    std::string dtype = "torch.float32";
    long batch_size = 8;
    long buffers_in = 2;
    long buffers_out = 2;
    std::string model_name = "google/vit-base-patch16-224";
    std::string models_file = "models.yml";
    std::string dev_types_file = "device_types.yml";
    std::string dev_file = "devices.yml";
};

long dtype_bytes(const std::string& dtype) {
    if (dtype == "torch.float32") return 4;
    if (dtype == "torch.float16" || dtype == "torch.bfloat16") return 2;
    if (dtype == "torch.float64") return 8;
    throw std::runtime_error("unsupported dtype: " + dtype);
}

ModelInfo load_model(const std::string& path, const std::string& name) {
    auto root = minyaml::parse_file(path);
    const auto& m = root->at(name);
    ModelInfo info;
    info.layers = m->at("layers")->as_long();
    info.parameters_in = m->at("parameters_in")->as_long();
    info.parameters_out = m->at("parameters_out")->as_long_list();
    info.mem_mb = m->at("mem_MB")->as_double_list();
    if ((long)info.parameters_out.size() != info.layers ||
        (long)info.mem_mb.size() != info.layers)
        throw std::runtime_error("models.yml: list lengths != layers");
    return info;
}

std::vector<DeviceType> load_device_types(const Options& opt) {
    auto root = minyaml::parse_file(opt.dev_types_file);
    auto hosts_root = minyaml::parse_file(opt.dev_file);
    std::vector<DeviceType> out;
    for (const auto& [name, props] : root->map) {
        DeviceType dt;
        dt.name = name;
        dt.mem_mb = props->at("mem_MB")->as_double();
        dt.bw_mbps = props->at("bw_Mbps")->as_double();
        if (props->has("model_profiles")) {
            const auto& profs = props->at("model_profiles");
            if (profs->has(opt.model_name)) {
                for (const auto& prof : profs->at(opt.model_name)->seq) {
                    if (prof->at("dtype")->scalar == opt.dtype &&
                        prof->at("batch_size")->as_long() == opt.batch_size) {
                        dt.time_s = prof->at("time_s")->as_double_list();
                        break;
                    }
                }
            }
        }
        if (hosts_root->has(name)) {
            for (const auto& h : hosts_root->at(name)->seq)
                dt.hosts.push_back(h->scalar);
        }
        out.push_back(std::move(dt));
    }
    return out;
}

struct Stage {
    size_t type_idx;
    long layer_l, layer_r;  // 1-indexed inclusive
};

// memory feasibility: weights + in/out transfer buffers + processing copies
// (model mirrored from pipeedge_amd/sched/__init__.py mem_bytes)
bool layers_fit(const ModelInfo& m, const DeviceType& dt, long l, long r,
                long dtype_sz, long ubatch, long buf_in, long buf_out) {
    double mem_model = 0;
    for (long i = l - 1; i < r; ++i) mem_model += m.mem_mb[i] * 1024 * 1024;
    const long params_in =
        l == 1 ? m.parameters_in : m.parameters_out[l - 2];
    const double dat_in = (double)params_in * ubatch * dtype_sz;
    const double dat_out = (double)m.parameters_out[r - 1] * ubatch * dtype_sz;
    double buffers = dat_in + dat_out;
    if (l > 1) buffers += dat_in * buf_in;
    buffers += dat_out * buf_out;
    return dt.mem_mb * 1024 * 1024 > mem_model + buffers;
}

std::vector<Stage> schedule(const ModelInfo& model,
                            const std::vector<DeviceType>& types,
                            const Options& opt) {
    const long L = model.layers;
    const size_t T = types.size();
    const long dsz = dtype_bytes(opt.dtype);

    std::vector<size_t> counts(T);
    for (size_t t = 0; t < T; ++t) counts[t] = types[t].hosts.size();

    // mixed-radix encoding of the used-device-type multiset
    std::vector<size_t> prefix(T + 1, 1);
    for (size_t t = 0; t < T; ++t) prefix[t + 1] = prefix[t] * (counts[t] + 1);
    const size_t mask = prefix[T];

    // precompute per-type cumulative compute times and feasibility
    std::vector<std::vector<double>> cumt(T);
    for (size_t t = 0; t < T; ++t) {
        cumt[t].assign(L + 1, 0.0);
        if ((long)types[t].time_s.size() == L)
            for (long i = 0; i < L; ++i)
                cumt[t][i + 1] = cumt[t][i] + types[t].time_s[i];
    }
    auto comp_time = [&](size_t t, long l, long r) {
        return cumt[t][r] - cumt[t][l - 1];
    };
    auto comm_time = [&](long r, size_t u, size_t v) {
        const double bytes =
            (double)model.parameters_out[r - 1] * opt.batch_size * dsz;
        const double bw = std::min(types[u].bw_mbps, types[v].bw_mbps);
        return bytes / (bw * 1024 * 1024 / 8);
    };

    // h[i][S][u]: best max-stage-cost covering layers [1, i]
    auto idx = [&](long i, size_t S, size_t u) {
        return ((size_t)i * mask + S) * T + u;
    };
    std::vector<double> h((L + 1) * mask * T, INF);
    std::vector<std::pair<int32_t, int32_t>> parent((L + 1) * mask * T,
                                                    {-1, -1});
    for (size_t u = 0; u < T; ++u) h[idx(0, 0, u)] = 0;

    double best = INF;
    std::optional<std::tuple<long, size_t, size_t>> best_idx;

    for (long i = 0; i < L; ++i) {
        for (size_t S = 0; S < mask; ++S) {
            for (size_t u = 0; u < T; ++u) {
                const double cur = h[idx(i, S, u)];
                if (cur == INF) continue;
                if (S / prefix[u] % (counts[u] + 1) == counts[u])
                    continue;  // type u exhausted in S
                if (counts[u] == 0 || (long)types[u].time_s.size() != L)
                    continue;
                for (long j = i + 1; j <= L; ++j) {
                    if (!layers_fit(model, types[u], i + 1, j, dsz,
                                    opt.batch_size, opt.buffers_in,
                                    opt.buffers_out))
                        continue;
                    const double ct = comp_time(u, i + 1, j);
                    if (j == L) {
                        const double cost = std::max(cur, ct);
                        if (cost < best) {
                            best = cost;
                            best_idx = {i, S, u};
                        }
                    }
                    for (size_t v = 0; v < T; ++v) {
                        if (S / prefix[v] % (counts[v] + 1) == counts[v])
                            continue;
                        const double cost =
                            std::max({cur, ct, comm_time(j, u, v)});
                        const size_t S2 = S + prefix[u];
                        if (cost < h[idx(j, S2, v)]) {
                            h[idx(j, S2, v)] = cost;
                            parent[idx(j, S2, v)] = {(int32_t)i, (int32_t)u};
                        }
                    }
                }
            }
        }
    }

    std::vector<Stage> stages;
    if (!best_idx) return stages;
    auto [layer, S, u] = *best_idx;
    stages.push_back({u, layer + 1, L});
    while (layer > 0) {
        auto [pl, pu] = parent[idx(layer, S, u)];
        stages.push_back({(size_t)pu, pl + 1, layer});
        S -= prefix[pu];
        layer = pl;
        u = (size_t)pu;
    }
    std::sort(stages.begin(), stages.end(),
              [](const Stage& a, const Stage& b) { return a.layer_l < b.layer_l; });
    return stages;
}

void print_schedule(const std::vector<Stage>& stages,
                    const std::vector<DeviceType>& types) {
    // concrete hosts round-robin per type, YAML list of single-entry maps
    std::vector<size_t> next(types.size(), 0);
    for (const auto& st : stages) {
        const auto& hosts = types[st.type_idx].hosts;
        const std::string& host = hosts[next[st.type_idx]++];
        std::printf("- %s: [%ld, %ld]\n", host.c_str(), st.layer_l,
                    st.layer_r);
    }
}

[[noreturn]] void usage(int code) {
    auto& os = code ? std::cerr : std::cout;
    os << "Usage: sched-pipeline [OPTION]...\n\n"
       << "Optimal-throughput pipeline partition scheduling.\n\n"
       << "  -h, --help                 Print this message and exit\n"
       << "  -d, --dtype=NAME           Data type (default torch.float32)\n"
       << "  -b, --batch-size=N         Microbatch size (default 8)\n"
       << "  -i, --buffers-in=N         Inbound data buffers (default 2)\n"
       << "  -o, --buffers-out=N        Outbound data buffers (default 2)\n"
       << "  -m, --model-name=NAME      Model name\n"
       << "  -M, --models-file=PATH     Models YAML (default models.yml)\n"
       << "  -T, --dev-types-file=PATH  Device types YAML\n"
       << "  -D, --dev-file=PATH        Devices YAML (default devices.yml)\n";
    std::exit(code);
}

}  // namespace

int main(int argc, char** argv) {
    Options opt;
    static const char short_opts[] = "hd:b:i:o:m:M:T:D:";
    static const struct option long_opts[] = {
        {"help", no_argument, nullptr, 'h'},
        {"dtype", required_argument, nullptr, 'd'},
        {"batch-size", required_argument, nullptr, 'b'},
        {"buffers-in", required_argument, nullptr, 'i'},
        {"buffers-out", required_argument, nullptr, 'o'},
        {"model-name", required_argument, nullptr, 'm'},
        {"models-file", required_argument, nullptr, 'M'},
        {"dev-types-file", required_argument, nullptr, 'T'},
        {"dev-file", required_argument, nullptr, 'D'},
        {nullptr, 0, nullptr, 0}};
    int c;
    while ((c = getopt_long(argc, argv, short_opts, long_opts, nullptr)) !=
           -1) {
        switch (c) {
            case 'h': usage(0);
            case 'd': opt.dtype = optarg; break;
            case 'b': opt.batch_size = std::atol(optarg); break;
            case 'i': opt.buffers_in = std::atol(optarg); break;
            case 'o': opt.buffers_out = std::atol(optarg); break;
            case 'm': opt.model_name = optarg; break;
            case 'M': opt.models_file = optarg; break;
            case 'T': opt.dev_types_file = optarg; break;
            case 'D': opt.dev_file = optarg; break;
            default: usage(1);
        }
    }
    try {
        const auto model = load_model(opt.models_file, opt.model_name);
        const auto types = load_device_types(opt);
        const auto stages = schedule(model, types, opt);
        if (stages.empty()) {
            std::cerr << "No viable schedule found" << std::endl;
            return 2;
        }
        print_schedule(stages, types);
    } catch (const std::exception& exc) {
        std::cerr << "sched-pipeline: " << exc.what() << std::endl;
        return 1;
    }
    return 0;
}
