// Python bindings for the gcbf_amd HIP/CDNA4 kernels (gfx950).
//
// Thin torch glue: tensor checks + allocation here, all device code in the
// .hip translation units (compiled with --offload-arch=gfx950).
#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

extern "C" void launch_seg_attn_fwd(const float* msg, const float* gate,
                                    const int* ptr, float* att, float* out,
                                    int N, int D, hipStream_t stream);
extern "C" void launch_seg_attn_bwd(const float* grad_out, const float* msg,
                                    const float* att, const int* ptr,
                                    float* dmsg, float* dgate, int N, int D,
                                    hipStream_t stream);
extern "C" void launch_seg_max_fwd(const float* values, const int* ptr,
                                   float* out, int* argmax, int N, int D,
                                   hipStream_t stream);
extern "C" void launch_seg_max_bwd(const float* grad_out, const int* argmax,
                                   float* dval, int N, int D,
                                   hipStream_t stream);
extern "C" void launch_radius_count(const float* pos, int* counts, int B,
                                    int N, int n_rec, int P, float r,
                                    int topk, hipStream_t stream);
extern "C" void launch_radius_fill(const float* pos, const float* states,
                                   const int* offsets, long* edge_index,
                                   float* edge_attr, long E_total, int B,
                                   int N, int n_rec, int P, int S, int A,
                                   float r, int topk, int attr_kind,
                                   hipStream_t stream);
extern "C" void launch_fused_linear_bf16(const void* A, const void* W,
                                         const float* bias, void* Cb,
                                         float* Cf, int M, int N, int K,
                                         int act, int out_f32,
                                         hipStream_t stream);
extern "C" void launch_pad_edges(const int* offsets, const int* counts,
                                 long* edge_index, long* seg,
                                 float* edge_attr, int* e_count, int rows,
                                 long E_max, long N_total, int A,
                                 hipStream_t stream);
extern "C" void launch_fused_masks(const float* states, bool* safe,
                                   bool* unsafe, bool* collision, int B,
                                   int N, int n_rec, int S, float r, int kind,
                                   hipStream_t stream);
extern "C" void launch_dubins_step(const float* states, const float* goal,
                                   const float* action, float* new_states,
                                   float* u_ref_next, float* reward,
                                   bool* reach, bool* collision, int N, int n,
                                   float dt, float r, float sl, float d2g,
                                   float act_lim, hipStream_t stream);
extern "C" void launch_car_step(const float* states, const float* goal,
                                const float* action, const float* K,
                                float* new_states, float* u_ref_next,
                                float* reward, bool* reach, bool* collision,
                                int N, float dt, float r, float sl, float d2g,
                                float act_lim, hipStream_t stream);
extern "C" void launch_drone_step(const float* states, const float* goal,
                                  const float* action, const float* K,
                                  float* new_states, float* u_ref_next,
                                  float* reward, bool* reach,
                                  bool* collision, int N, int n, float dt,
                                  float r, float sl, float d2g,
                                  float act_lim, hipStream_t stream);

namespace {

#define CHECK_IN(x)                                                    \
    TORCH_CHECK(x.is_cuda(), #x " must be on GPU");                    \
    TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")

hipStream_t current_stream() {
    return c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

std::vector<torch::Tensor> segment_attn_fwd(torch::Tensor msg,
                                            torch::Tensor gate,
                                            torch::Tensor ptr) {
    CHECK_IN(msg);
    CHECK_IN(gate);
    CHECK_IN(ptr);
    TORCH_CHECK(msg.scalar_type() == torch::kFloat32, "msg must be fp32");
    TORCH_CHECK(ptr.scalar_type() == torch::kInt32, "ptr must be int32");
    const int64_t E = msg.size(0), D = msg.size(1);
    const int64_t N = ptr.size(0) - 1;
    // att is zero-initialized for the same reason as the backward outputs:
    // edges past the last CSR segment (sentinel pads) are never written by
    // the kernel and must not surface uninitialized memory
    auto att = torch::zeros({E}, msg.options());
    auto out = torch::empty({N, D}, msg.options());
    if (N > 0)
        launch_seg_attn_fwd(msg.data_ptr<float>(), gate.data_ptr<float>(),
                            ptr.data_ptr<int>(), att.data_ptr<float>(),
                            out.data_ptr<float>(), (int)N, (int)D,
                            current_stream());
    return {att, out};
}

std::vector<torch::Tensor> segment_attn_bwd(torch::Tensor grad_out,
                                            torch::Tensor msg,
                                            torch::Tensor att,
                                            torch::Tensor ptr) {
    CHECK_IN(grad_out);
    CHECK_IN(msg);
    CHECK_IN(att);
    CHECK_IN(ptr);
    const int64_t E = msg.size(0), D = msg.size(1);
    const int64_t N = ptr.size(0) - 1;
    // zeros, not empty: the kernel only writes edges covered by a CSR
    // segment, and padded edge buffers (sentinel segment id past the last
    // node) legitimately contain uncovered edges — leaving those rows
    // uninitialized poisoned every gradient downstream of the aggregation
    // in the captured update engine
    auto dmsg = torch::zeros_like(msg);
    auto dgate = torch::zeros({E}, msg.options());
    if (N > 0)
        launch_seg_attn_bwd(grad_out.data_ptr<float>(), msg.data_ptr<float>(),
                            att.data_ptr<float>(), ptr.data_ptr<int>(),
                            dmsg.data_ptr<float>(), dgate.data_ptr<float>(),
                            (int)N, (int)D, current_stream());
    return {dmsg, dgate};
}

std::vector<torch::Tensor> segment_max_fwd(torch::Tensor values,
                                           torch::Tensor ptr) {
    CHECK_IN(values);
    CHECK_IN(ptr);
    const int64_t D = values.size(1);
    const int64_t N = ptr.size(0) - 1;
    auto out = torch::empty({N, D}, values.options());
    auto argmax = torch::empty({N, D},
                               values.options().dtype(torch::kInt32));
    if (N > 0)
        launch_seg_max_fwd(values.data_ptr<float>(), ptr.data_ptr<int>(),
                           out.data_ptr<float>(), argmax.data_ptr<int>(),
                           (int)N, (int)D, current_stream());
    return {out, argmax};
}

torch::Tensor segment_max_bwd(torch::Tensor grad_out, torch::Tensor argmax,
                              int64_t E) {
    CHECK_IN(grad_out);
    CHECK_IN(argmax);
    const int64_t N = grad_out.size(0), D = grad_out.size(1);
    auto dval = torch::zeros({E, D}, grad_out.options());
    if (N > 0)
        launch_seg_max_bwd(grad_out.data_ptr<float>(),
                           argmax.data_ptr<int>(), dval.data_ptr<float>(),
                           (int)N, (int)D, current_stream());
    return dval;
}

std::vector<torch::Tensor> build_graph(torch::Tensor pos,
                                       torch::Tensor states, int64_t B,
                                       int64_t n_rec, double r, int64_t topk,
                                       int64_t attr_kind, int64_t attr_dim) {
    CHECK_IN(pos);
    CHECK_IN(states);
    const int64_t N = pos.size(0) / B;
    const int64_t P = pos.size(1), S = states.size(1);
    auto stream = current_stream();
    auto counts = torch::empty({B * n_rec},
                               pos.options().dtype(torch::kInt32));
    launch_radius_count(pos.data_ptr<float>(), counts.data_ptr<int>(),
                        (int)B, (int)N, (int)n_rec, (int)P, (float)r,
                        (int)topk, stream);
    auto incl = counts.cumsum(0, torch::kInt32);
    auto offsets = (incl - counts).contiguous();
    const int64_t E = incl.numel() ? incl[-1].item<int64_t>() : 0;
    auto edge_index = torch::empty({2, E},
                                   pos.options().dtype(torch::kInt64));
    auto edge_attr = torch::empty({E, attr_dim}, pos.options());
    if (E > 0)
        launch_radius_fill(pos.data_ptr<float>(), states.data_ptr<float>(),
                           offsets.data_ptr<int>(),
                           edge_index.data_ptr<int64_t>(),
                           edge_attr.data_ptr<float>(), E, (int)B, (int)N,
                           (int)n_rec, (int)P, (int)S, (int)attr_dim,
                           (float)r, (int)topk, (int)attr_kind, stream);
    return {edge_index, edge_attr};
}

std::vector<torch::Tensor> fused_masks(torch::Tensor states, int64_t B,
                                       int64_t n_rec, double r, int64_t kind,
                                       bool want_safe, bool want_unsafe,
                                       bool want_coll) {
    CHECK_IN(states);
    const int64_t N = states.size(0) / B;
    const int64_t S = states.size(1);
    auto opts = states.options().dtype(torch::kBool);
    auto none = torch::Tensor();
    auto t_safe = want_safe ? torch::empty({B * n_rec}, opts) : none;
    auto t_uns = want_unsafe ? torch::empty({B * n_rec}, opts) : none;
    auto t_coll = want_coll ? torch::empty({B * n_rec}, opts) : none;
    launch_fused_masks(
        states.data_ptr<float>(),
        want_safe ? t_safe.data_ptr<bool>() : nullptr,
        want_unsafe ? t_uns.data_ptr<bool>() : nullptr,
        want_coll ? t_coll.data_ptr<bool>() : nullptr,
        (int)B, (int)N, (int)n_rec, (int)S, (float)r, (int)kind,
        current_stream());
    return {t_safe, t_uns, t_coll};
}

std::vector<torch::Tensor> build_graph_padded(
        torch::Tensor pos, torch::Tensor states, int64_t B, int64_t n_rec,
        double r, int64_t topk, int64_t attr_kind, int64_t attr_dim,
        int64_t E_max,
        c10::optional<std::vector<torch::Tensor>> out_opt = c10::nullopt) {
    // capture-safe variant: fixed E_max edge buffers, no host sync.
    CHECK_IN(pos);
    CHECK_IN(states);
    const int64_t N = pos.size(0) / B;
    const int64_t P = pos.size(1), S = states.size(1);
    auto stream = current_stream();
    auto counts = torch::empty({B * n_rec},
                               pos.options().dtype(torch::kInt32));
    launch_radius_count(pos.data_ptr<float>(), counts.data_ptr<int>(),
                        (int)B, (int)N, (int)n_rec, (int)P, (float)r,
                        (int)topk, stream);
    auto incl = counts.cumsum(0, torch::kInt32);
    auto offsets = (incl - counts).contiguous();
    torch::Tensor edge_index, seg, edge_attr, e_count;
    if (out_opt.has_value()) {
        TORCH_CHECK(out_opt->size() == 4,
                    "out must be {edge_index, seg, edge_attr, e_count}");
        for (const auto& t : *out_opt) { CHECK_IN(t); }
        edge_index = (*out_opt)[0];
        seg = (*out_opt)[1];
        edge_attr = (*out_opt)[2];
        e_count = (*out_opt)[3];
        TORCH_CHECK(edge_index.size(1) == E_max && seg.size(0) == E_max
                    && edge_attr.size(0) == E_max, "E_max mismatch");
    } else {
        edge_index = torch::empty({2, E_max},
                                  pos.options().dtype(torch::kInt64));
        seg = torch::empty({E_max}, pos.options().dtype(torch::kInt64));
        edge_attr = torch::empty({E_max, attr_dim}, pos.options());
        e_count = torch::empty({1}, pos.options().dtype(torch::kInt32));
    }
    launch_radius_fill(pos.data_ptr<float>(), states.data_ptr<float>(),
                       offsets.data_ptr<int>(),
                       edge_index.data_ptr<int64_t>(),
                       edge_attr.data_ptr<float>(), E_max, (int)B, (int)N,
                       (int)n_rec, (int)P, (int)S, (int)attr_dim, (float)r,
                       (int)topk, (int)attr_kind, stream);
    launch_pad_edges(offsets.data_ptr<int>(), counts.data_ptr<int>(),
                     edge_index.data_ptr<int64_t>(),
                     seg.data_ptr<int64_t>(), edge_attr.data_ptr<float>(),
                     e_count.data_ptr<int>(), (int)(B * n_rec), E_max,
                     B * N, (int)attr_dim, stream);
    return {edge_index, seg, edge_attr, e_count};
}

torch::Tensor fused_linear(torch::Tensor x, torch::Tensor w,
                           c10::optional<torch::Tensor> bias, int64_t act,
                           bool out_f32) {
    CHECK_IN(x);
    CHECK_IN(w);
    TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "x must be bf16");
    TORCH_CHECK(w.scalar_type() == torch::kBFloat16, "w must be bf16");
    const int64_t M = x.size(0), K = x.size(1), N = w.size(0);
    TORCH_CHECK(w.size(1) == K, "K mismatch");
    TORCH_CHECK(M % 128 == 0 && N % 128 == 0 && K % 64 == 0,
                "fused_linear needs M%128==0, N%128==0, K%64==0 (got ", M,
                "x", K, " -> ", N, ")");
    const float* bptr = nullptr;
    if (bias.has_value()) {
        CHECK_IN(bias.value());
        TORCH_CHECK(bias->scalar_type() == torch::kFloat32,
                    "bias must be fp32");
        bptr = bias->data_ptr<float>();
    }
    auto out = torch::empty({M, N}, x.options().dtype(
        out_f32 ? torch::kFloat32 : torch::kBFloat16));
    launch_fused_linear_bf16(
        x.data_ptr(), w.data_ptr(), bptr,
        out_f32 ? nullptr : out.data_ptr(),
        out_f32 ? out.data_ptr<float>() : nullptr,
        (int)M, (int)N, (int)K, (int)act, out_f32 ? 1 : 0,
        current_stream());
    return out;
}

struct StepOut {
    torch::Tensor new_states, u_ref_next, reward, reach, collision;
};

static std::vector<torch::Tensor> alloc_step_out(
        torch::Tensor states, int64_t n, int64_t A,
        const c10::optional<std::vector<torch::Tensor>>& out) {
    if (out.has_value()) {
        // caller-provided output buffers (ping-pong rollout capture: the
        // kernels write the next phase's inputs directly, no copy-back)
        TORCH_CHECK(out->size() == 5, "out must be "
                    "{new_states, u_ref_next, reward, reach, collision}");
        for (const auto& t : *out) { CHECK_IN(t); }
        return *out;
    }
    auto f = states.options();
    auto b = states.options().dtype(torch::kBool);
    return {torch::empty_like(states), torch::empty({n, A}, f),
            torch::empty({n}, f), torch::empty({n}, b),
            torch::empty({n}, b)};
}

std::vector<torch::Tensor> dubins_step(
        torch::Tensor states, torch::Tensor goal, torch::Tensor action,
        double dt, double r, double sl, double d2g, double act_lim,
        c10::optional<std::vector<torch::Tensor>> out_opt = c10::nullopt) {
    CHECK_IN(states); CHECK_IN(goal); CHECK_IN(action);
    const int64_t N = states.size(0), n = action.size(0);
    auto out = alloc_step_out(states, n, 2, out_opt);
    launch_dubins_step(states.data_ptr<float>(), goal.data_ptr<float>(),
                       action.data_ptr<float>(), out[0].data_ptr<float>(),
                       out[1].data_ptr<float>(), out[2].data_ptr<float>(),
                       out[3].data_ptr<bool>(), out[4].data_ptr<bool>(),
                       (int)N, (int)n, (float)dt, (float)r, (float)sl,
                       (float)d2g, (float)act_lim, current_stream());
    return out;
}

std::vector<torch::Tensor> car_step(
        torch::Tensor states, torch::Tensor goal, torch::Tensor action,
        torch::Tensor K, double dt, double r, double sl, double d2g,
        double act_lim,
        c10::optional<std::vector<torch::Tensor>> out_opt = c10::nullopt) {
    CHECK_IN(states); CHECK_IN(goal); CHECK_IN(action); CHECK_IN(K);
    const int64_t N = states.size(0);
    auto out = alloc_step_out(states, N, 2, out_opt);
    launch_car_step(states.data_ptr<float>(), goal.data_ptr<float>(),
                    action.data_ptr<float>(), K.data_ptr<float>(),
                    out[0].data_ptr<float>(), out[1].data_ptr<float>(),
                    out[2].data_ptr<float>(), out[3].data_ptr<bool>(),
                    out[4].data_ptr<bool>(), (int)N, (float)dt, (float)r,
                    (float)sl, (float)d2g, (float)act_lim, current_stream());
    return out;
}

std::vector<torch::Tensor> drone_step(
        torch::Tensor states, torch::Tensor goal, torch::Tensor action,
        torch::Tensor K, double dt, double r, double sl, double d2g,
        double act_lim,
        c10::optional<std::vector<torch::Tensor>> out_opt = c10::nullopt) {
    CHECK_IN(states); CHECK_IN(goal); CHECK_IN(action); CHECK_IN(K);
    const int64_t N = states.size(0), n = action.size(0);
    auto out = alloc_step_out(states, n, 3, out_opt);
    launch_drone_step(states.data_ptr<float>(), goal.data_ptr<float>(),
                      action.data_ptr<float>(), K.data_ptr<float>(),
                      out[0].data_ptr<float>(), out[1].data_ptr<float>(),
                      out[2].data_ptr<float>(), out[3].data_ptr<bool>(),
                      out[4].data_ptr<bool>(), (int)N, (int)n, (float)dt,
                      (float)r, (float)sl, (float)d2g, (float)act_lim,
                      current_stream());
    return out;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("fused_masks", &fused_masks,
          "batched safe/unsafe/collision agent masks in one pass");
    m.def("dubins_step", &dubins_step, "fused DubinsCar rollout step",
          py::arg("states"), py::arg("goal"), py::arg("action"),
          py::arg("dt"), py::arg("r"), py::arg("sl"), py::arg("d2g"),
          py::arg("act_lim"), py::arg("out") = py::none());
    m.def("car_step", &car_step, "fused SimpleCar rollout step",
          py::arg("states"), py::arg("goal"), py::arg("action"),
          py::arg("K"), py::arg("dt"), py::arg("r"), py::arg("sl"),
          py::arg("d2g"), py::arg("act_lim"), py::arg("out") = py::none());
    m.def("drone_step", &drone_step, "fused SimpleDrone rollout step",
          py::arg("states"), py::arg("goal"), py::arg("action"),
          py::arg("K"), py::arg("dt"), py::arg("r"), py::arg("sl"),
          py::arg("d2g"), py::arg("act_lim"), py::arg("out") = py::none());
    m.def("segment_attn_fwd", &segment_attn_fwd,
          "fused scatter-softmax + weighted scatter-sum (forward)");
    m.def("segment_attn_bwd", &segment_attn_bwd,
          "fused scatter-softmax + weighted scatter-sum (backward)");
    m.def("segment_max_fwd", &segment_max_fwd,
          "CSR segment max with argmax (forward)");
    m.def("segment_max_bwd", &segment_max_bwd,
          "CSR segment max scatter-back (backward)");
    m.def("build_graph", &build_graph,
          "batched dense radius graph + edge_attr (count/scan/fill)");
    m.def("build_graph_padded", &build_graph_padded,
          "capture-safe radius graph into fixed E_max buffers",
          py::arg("pos"), py::arg("states"), py::arg("B"), py::arg("n_rec"),
          py::arg("r"), py::arg("topk"), py::arg("attr_kind"),
          py::arg("attr_dim"), py::arg("E_max"),
          py::arg("out") = py::none());
    m.def("fused_linear", &fused_linear,
          "MFMA bf16 GEMM with fused bias + activation epilogue");
}
