// Torch glue for the CDNA4 fused multi-tensor kernels (fused_ops.hip).
//
// Responsibilities:
//   * validate tensor lists (fp32, contiguous, same device)
//   * build + cache the int64 metadata buffer (pointers, numels,
//     chunk->tensor map) on device; cache key = the pointer/numel multiset,
//     so steady-state training steps perform NO host->device traffic and
//     are hipGraph-capture safe
//   * launch through the extern "C" launchers compiled by hipcc

#include <torch/extension.h>

#include <ATen/hip/HIPContext.h>

#include <map>
#include <vector>

extern "C" {
void swq_launch_fused_sgd(const long long*, int, int, float, float, float,
                          float, int, int, void*);
void swq_launch_fused_adam(const long long*, int, int, float, float, float,
                           float, float, int, int, void*);
void swq_launch_multi_tensor_accum(const long long*, int, int, float, void*);
void swq_launch_multi_tensor_l2norm_sq(const long long*, int, int, float*,
                                       void*);
void swq_launch_gns_window_stats(const long long*, int, long long, float*,
                                 void*);
}

namespace {

constexpr long long kChunkElems = 32768;  // keep in sync with fused_ops.hip

struct MetaEntry {
    torch::Tensor device_buf;
    int num_chunks;
};

using Key = std::vector<long long>;
std::map<Key, MetaEntry> g_meta_cache;

void check_lists(const std::vector<std::vector<torch::Tensor>>& lists) {
    TORCH_CHECK(!lists.empty() && !lists[0].empty(), "empty tensor lists");
    const size_t T = lists[0].size();
    for (const auto& list : lists) {
        TORCH_CHECK(list.size() == T, "tensor list length mismatch");
        for (size_t t = 0; t < T; ++t) {
            const auto& ten = list[t];
            TORCH_CHECK(ten.is_cuda(), "swq ops require device tensors");
            TORCH_CHECK(ten.scalar_type() == torch::kFloat32,
                        "swq ops are fp32-only (got ", ten.scalar_type(), ")");
            // flat elementwise kernels: any dense non-overlapping layout is
            // fine (NCHW-contiguous or channels_last) as long as every list
            // shares the layout per tensor index
            TORCH_CHECK(ten.is_non_overlapping_and_dense(),
                        "tensor ", t, " not dense");
            TORCH_CHECK(ten.numel() == lists[0][t].numel(),
                        "numel mismatch across lists at tensor ", t);
            // layouts must agree so the flat kernels pair elements
            // correctly; strides of size-1 dims are don't-care (a 1x1
            // conv's NCHW-contiguous grad is layout-equal to its
            // channels_last param even though the stride tuples differ)
            const auto& ref = lists[0][t];
            for (int64_t d = 0; d < ten.dim(); ++d) {
                if (ten.size(d) <= 1) continue;
                TORCH_CHECK(ten.stride(d) == ref.stride(d),
                            "stride/layout mismatch across lists at tensor ",
                            t, " dim ", d);
            }
        }
    }
}

const MetaEntry& get_meta(const std::vector<std::vector<torch::Tensor>>& lists) {
    const int NL = (int)lists.size();
    const int T = (int)lists[0].size();
    Key key;
    key.reserve(NL * T + T);
    for (const auto& list : lists)
        for (const auto& ten : list)
            key.push_back((long long)ten.data_ptr());
    for (const auto& ten : lists[0]) key.push_back(ten.numel());

    auto it = g_meta_cache.find(key);
    if (it != g_meta_cache.end()) return it->second;

    // build chunk map
    std::vector<long long> host;
    host.reserve(NL * T + T);
    for (const auto& list : lists)
        for (const auto& ten : list)
            host.push_back((long long)ten.data_ptr());
    int num_chunks = 0;
    std::vector<long long> pairs;
    for (int t = 0; t < T; ++t) {
        const long long n = lists[0][t].numel();
        host.push_back(n);
        const int chunks = (int)((n + kChunkElems - 1) / kChunkElems);
        for (int c = 0; c < chunks; ++c) {
            pairs.push_back(t);
            pairs.push_back(c);
            ++num_chunks;
        }
    }
    host.insert(host.end(), pairs.begin(), pairs.end());

    auto cpu = torch::from_blob(host.data(), {(long long)host.size()},
                                torch::kInt64)
                   .clone();
    auto dev = cpu.to(lists[0][0].device(), /*non_blocking=*/false);
    auto res = g_meta_cache.emplace(std::move(key),
                                    MetaEntry{dev, num_chunks});
    return res.first->second;
}

void* current_stream(const torch::Tensor& ref) {
    return (void*)at::hip::getCurrentHIPStream(ref.device().index()).stream();
}

}  // namespace

void fused_sgd(std::vector<torch::Tensor> params,
               std::vector<torch::Tensor> grads,
               std::vector<torch::Tensor> momentum_bufs, double lr,
               double momentum, double dampening, double weight_decay,
               bool nesterov, bool buf_initialized) {
    std::vector<std::vector<torch::Tensor>> lists{params, grads, momentum_bufs};
    check_lists(lists);
    const auto& meta = get_meta(lists);
    swq_launch_fused_sgd((const long long*)meta.device_buf.data_ptr<int64_t>(),
                         (int)params.size(), meta.num_chunks, (float)lr,
                         (float)momentum, (float)dampening,
                         (float)weight_decay, nesterov ? 1 : 0,
                         buf_initialized ? 1 : 0, current_stream(params[0]));
}

void fused_adam(std::vector<torch::Tensor> params,
                std::vector<torch::Tensor> grads,
                std::vector<torch::Tensor> exp_avgs,
                std::vector<torch::Tensor> exp_avg_sqs, double lr,
                double beta1, double beta2, double eps, double weight_decay,
                int64_t step, bool adamw) {
    std::vector<std::vector<torch::Tensor>> lists{params, grads, exp_avgs,
                                                  exp_avg_sqs};
    check_lists(lists);
    const auto& meta = get_meta(lists);
    swq_launch_fused_adam((const long long*)meta.device_buf.data_ptr<int64_t>(),
                          (int)params.size(), meta.num_chunks, (float)lr,
                          (float)beta1, (float)beta2, (float)eps,
                          (float)weight_decay, (int)step, adamw ? 1 : 0,
                          current_stream(params[0]));
}

void multi_tensor_accum(std::vector<torch::Tensor> dsts,
                        std::vector<torch::Tensor> srcs, double alpha) {
    std::vector<std::vector<torch::Tensor>> lists{dsts, srcs};
    check_lists(lists);
    const auto& meta = get_meta(lists);
    swq_launch_multi_tensor_accum((const long long*)meta.device_buf.data_ptr<int64_t>(),
                                  (int)dsts.size(), meta.num_chunks,
                                  (float)alpha, current_stream(dsts[0]));
}

torch::Tensor multi_tensor_l2norm_sq(std::vector<torch::Tensor> tensors,
                                     torch::Tensor out) {
    std::vector<std::vector<torch::Tensor>> lists{tensors};
    check_lists(lists);
    TORCH_CHECK(out.numel() == (long long)tensors.size(),
                "out must have one element per tensor");
    TORCH_CHECK(out.scalar_type() == torch::kFloat32 && out.is_cuda());
    out.zero_();
    const auto& meta = get_meta(lists);
    swq_launch_multi_tensor_l2norm_sq(
        (const long long*)meta.device_buf.data_ptr<int64_t>(), (int)tensors.size(),
        meta.num_chunks, out.data_ptr<float>(), current_stream(tensors[0]));
    return out;
}

torch::Tensor gns_window_stats(std::vector<torch::Tensor> grads,
                               torch::Tensor out) {
    std::vector<std::vector<torch::Tensor>> lists{grads};
    check_lists(lists);
    const long long n = grads[0].numel();
    for (const auto& g : grads) TORCH_CHECK(g.numel() == n);
    TORCH_CHECK(out.numel() >= 2 && out.is_cuda() &&
                out.scalar_type() == torch::kFloat32);
    out.zero_();

    Key key;
    for (const auto& g : grads) key.push_back((long long)g.data_ptr());
    key.push_back(n);
    auto it = g_meta_cache.find(key);
    if (it == g_meta_cache.end()) {
        std::vector<long long> host;
        for (const auto& g : grads) host.push_back((long long)g.data_ptr());
        auto cpu = torch::from_blob(host.data(), {(long long)host.size()},
                                    torch::kInt64)
                       .clone();
        auto dev = cpu.to(grads[0].device(), false);
        it = g_meta_cache.emplace(std::move(key), MetaEntry{dev, 0}).first;
    }
    swq_launch_gns_window_stats((const long long*)it->second.device_buf.data_ptr<int64_t>(),
                                (int)grads.size(), n, out.data_ptr<float>(),
                                current_stream(grads[0]));
    return out;
}

void clear_meta_cache() { g_meta_cache.clear(); }

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.doc() = "shockwave_amd CDNA4 fused multi-tensor kernels";
    m.def("fused_sgd", &fused_sgd, "fused multi-tensor SGD-momentum step");
    m.def("fused_adam", &fused_adam, "fused multi-tensor Adam/AdamW step");
    m.def("multi_tensor_accum", &multi_tensor_accum, "dst += alpha*src");
    m.def("multi_tensor_l2norm_sq", &multi_tensor_l2norm_sq,
          "per-tensor squared L2 norms");
    m.def("gns_window_stats", &gns_window_stats,
          "window-average + current grad squared norms");
    m.def("clear_meta_cache", &clear_meta_cache);
}
