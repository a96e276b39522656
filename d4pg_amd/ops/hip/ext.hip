// Python bindings for the fused D4PG engine (torch extension).
// Single translation unit: includes engine.hip (kernels + host engine).

#include "engine.hip"

#include <torch/extension.h>

#include <memory>
#include <unordered_map>

namespace d4pg {

static std::unordered_map<int64_t, std::unique_ptr<Engine>> g_engines;
static int64_t g_next = 1;

static Engine& get(int64_t h) {
    auto it = g_engines.find(h);
    if (it == g_engines.end())
        throw std::runtime_error("invalid engine handle");
    return *it->second;
}

static int64_t create(int64_t obs, int64_t act, int64_t hidden, int64_t atoms,
                      int64_t batch, int64_t capacity, double v_min,
                      double v_max, double gamma_n, double tau,
                      double lr_actor, double lr_critic, double per_alpha,
                      double per_beta0, int64_t per_beta_iters,
                      double per_eps, int64_t seed, bool is_weighting) {
    EngineCfg c{};
    c.obs = (int)obs; c.act = (int)act; c.hidden = (int)hidden;
    c.atoms = (int)atoms; c.batch = (int)batch;
    c.capacity = capacity;
    c.v_min = (float)v_min; c.v_max = (float)v_max;
    c.gamma_n = (float)gamma_n; c.tau = (float)tau;
    c.lr_actor = (float)lr_actor; c.lr_critic = (float)lr_critic;
    c.per_alpha = (float)per_alpha; c.per_beta0 = (float)per_beta0;
    c.per_beta_iters = per_beta_iters; c.per_eps = (float)per_eps;
    c.seed = (uint64_t)seed; c.is_weighting = is_weighting ? 1 : 0;
    int64_t h = g_next++;
    g_engines[h] = std::make_unique<Engine>(c);
    return h;
}

static void destroy(int64_t h) { g_engines.erase(h); }

static py::dict info(int64_t h) {
    Engine& e = get(h);
    py::dict d;
    d["n_params_actor"] = e.anet.n_params;
    d["n_params_critic"] = e.cnet.n_params;
    d["tree_cap"] = e.tree_cap;
    d["persistent"] = e.use_persistent();
    d["seed"] = (int64_t)e.cfg.seed;
    py::list al, cl;
    for (int i = 0; i < 4; ++i) {
        py::dict ld;
        ld["in1"] = e.anet.l[i].in1; ld["in2"] = e.anet.l[i].in2;
        ld["out"] = e.anet.l[i].out; ld["w_off"] = e.anet.l[i].w_off;
        ld["b_off"] = e.anet.l[i].b_off;
        al.append(ld);
        py::dict ld2;
        ld2["in1"] = e.cnet.l[i].in1; ld2["in2"] = e.cnet.l[i].in2;
        ld2["out"] = e.cnet.l[i].out; ld2["w_off"] = e.cnet.l[i].w_off;
        ld2["b_off"] = e.cnet.l[i].b_off;
        cl.append(ld2);
    }
    d["actor_layers"] = al;
    d["critic_layers"] = cl;
    return d;
}

static torch::Tensor f32_contig(torch::Tensor t) {
    return t.to(torch::kFloat32).contiguous().cpu();
}

// which: 0 actor, 1 actor_target, 2 critic, 3 critic_target
static float* slab_ptr(Engine& e, int64_t which, long& n) {
    switch (which) {
        case 0: n = e.anet.n_params; return e.p_actor;
        case 1: n = e.anet.n_params; return e.p_actor_t;
        case 2: n = e.cnet.n_params; return e.p_critic;
        case 3: n = e.cnet.n_params; return e.p_critic_t;
        case 4: n = e.anet.n_params; return e.g_actor;
        case 5: n = e.cnet.n_params; return e.g_critic;
        case 6: n = e.anet.n_params; return e.m_actor;
        case 7: n = e.anet.n_params; return e.v_actor;
        case 8: n = e.cnet.n_params; return e.m_critic;
        case 9: n = e.cnet.n_params; return e.v_critic;
    }
    throw std::runtime_error("bad slab id");
}

static void load_slab(int64_t h, int64_t which, torch::Tensor flat) {
    Engine& e = get(h);
    long n;
    float* dst = slab_ptr(e, which, n);
    auto t = f32_contig(flat);
    TORCH_CHECK(t.numel() == n, "slab size mismatch");
    e.load_slab(dst, t.data_ptr<float>(), n);
}

static torch::Tensor store_slab(int64_t h, int64_t which) {
    Engine& e = get(h);
    long n;
    float* src = slab_ptr(e, which, n);
    auto out = torch::empty({n}, torch::kFloat32);
    e.store_slab(out.data_ptr<float>(), src, n);
    return out;
}

static void synth_fill(int64_t h, int64_t n, int64_t seed) {
    get(h).synth_fill(n, (uint64_t)seed);
}

static void ingest(int64_t h, torch::Tensor s, torch::Tensor a,
                   torch::Tensor r, torch::Tensor s2, torch::Tensor d) {
    Engine& e = get(h);
    auto ts = f32_contig(s), ta = f32_contig(a), tr = f32_contig(r),
         ts2 = f32_contig(s2), td = f32_contig(d);
    int T = (int)tr.numel();
    TORCH_CHECK(ts.numel() == (long)T * e.cfg.obs, "bad states shape");
    TORCH_CHECK(ta.numel() == (long)T * e.cfg.act, "bad actions shape");
    e.ingest(ts.data_ptr<float>(), ta.data_ptr<float>(), tr.data_ptr<float>(),
             ts2.data_ptr<float>(), td.data_ptr<float>(), T);
}

// exact-resume support: restore the schedule counters to "steps_done
// completed" (adam/rng pre-advanced to steps_done+1, beta_t = steps_done —
// the same convention Engine::alloc establishes at step 0)
static void set_seed(int64_t h, int64_t seed) {
    Engine& e = get(h);
    e.cfg.seed = (uint64_t)seed;
    // the seed is baked into captured kernel ARGUMENTS (k_per_sample /
    // k_step_persistent take it by value), so a previously captured
    // hipGraph would silently keep sampling with the stale seed —
    // invalidate it; FusedEngine.set_seed resets _captured so the next
    // train_steps() recaptures with the new seed.
    e.invalidate_graph();
}

static void set_schedule(int64_t h, int64_t steps_done, double max_priority) {
    Engine& e = get(h);
    Counters c{};
    HIP_CHECK(hipMemcpy(&c, e.cnt, sizeof(c), hipMemcpyDeviceToHost));
    c.beta_t = steps_done;
    c.adam_t_actor = c.adam_t_critic = c.rng_epoch = steps_done + 1;
    c.max_priority = (float)max_priority;
    HIP_CHECK(hipMemcpy(e.cnt, &c, sizeof(c), hipMemcpyHostToDevice));
}

// on-HBM replay snapshot: SoA rows [0, size) + both segment trees
static py::dict replay_state(int64_t h) {
    Engine& e = get(h);
    Counters c{};
    HIP_CHECK(hipMemcpy(&c, e.cnt, sizeof(c), hipMemcpyDeviceToHost));
    long n = c.size;
    const int O = e.cfg.obs, A = e.cfg.act;
    py::dict d;
    auto grab = [&](const float* src, long rows, long w) {
        auto t = torch::empty({rows, w}, torch::kFloat32);
        if (rows)
            HIP_CHECK(hipMemcpy(t.data_ptr<float>(), src, rows * w * 4,
                                hipMemcpyDeviceToHost));
        return t;
    };
    d["s"] = grab(e.rs, n, O);
    d["a"] = grab(e.ra, n, A);
    d["r"] = grab(e.rr, n, 1);
    d["s2"] = grab(e.rs2, n, O);
    d["d"] = grab(e.rd, n, 1);
    auto st = torch::empty({2 * e.tree_cap}, torch::kFloat64);
    auto mt = torch::empty({2 * e.tree_cap}, torch::kFloat64);
    HIP_CHECK(hipMemcpy(st.data_ptr<double>(), e.sum_tree,
                        2 * e.tree_cap * 8, hipMemcpyDeviceToHost));
    HIP_CHECK(hipMemcpy(mt.data_ptr<double>(), e.min_tree,
                        2 * e.tree_cap * 8, hipMemcpyDeviceToHost));
    d["sum_tree"] = st;
    d["min_tree"] = mt;
    d["size"] = c.size;
    d["pos"] = c.pos;
    d["max_priority"] = c.max_priority;
    return d;
}

static void load_replay_state(int64_t h, torch::Tensor s, torch::Tensor a,
                              torch::Tensor r, torch::Tensor s2,
                              torch::Tensor dn, torch::Tensor sum_tree,
                              torch::Tensor min_tree, int64_t size,
                              int64_t pos, double max_priority) {
    Engine& e = get(h);
    const int O = e.cfg.obs, A = e.cfg.act;
    long n = size;
    TORCH_CHECK(sum_tree.numel() == 2 * e.tree_cap, "tree size mismatch");
    auto put = [&](float* dst, torch::Tensor t, long rows, long w) {
        auto tt = f32_contig(t);
        TORCH_CHECK(tt.numel() == rows * w, "replay slab size mismatch");
        if (rows)
            HIP_CHECK(hipMemcpy(dst, tt.data_ptr<float>(), rows * w * 4,
                                hipMemcpyHostToDevice));
    };
    put(e.rs, s, n, O);
    put(e.ra, a, n, A);
    put(e.rr, r, n, 1);
    put(e.rs2, s2, n, O);
    put(e.rd, dn, n, 1);
    auto st = sum_tree.to(torch::kFloat64).contiguous().cpu();
    auto mt = min_tree.to(torch::kFloat64).contiguous().cpu();
    HIP_CHECK(hipMemcpy(e.sum_tree, st.data_ptr<double>(),
                        2 * e.tree_cap * 8, hipMemcpyHostToDevice));
    HIP_CHECK(hipMemcpy(e.min_tree, mt.data_ptr<double>(),
                        2 * e.tree_cap * 8, hipMemcpyHostToDevice));
    Counters c{};
    HIP_CHECK(hipMemcpy(&c, e.cnt, sizeof(c), hipMemcpyDeviceToHost));
    c.size = size;
    c.pos = pos;
    c.max_priority = (float)max_priority;
    HIP_CHECK(hipMemcpy(e.cnt, &c, sizeof(c), hipMemcpyHostToDevice));
}

static void step(int64_t h, int64_t n) { get(h).step((int)n); }
static void step_part(int64_t h, int64_t mask) {
    get(h).step_part((int)mask);
}

// Zero-copy torch view of an engine slab (params/grads/moments) ON DEVICE,
// so torch.distributed collectives (RCCL over xGMI) can all-reduce /
// broadcast engine state in place — the learner-DP and local-SGD paths.
// The tensor aliases engine memory: keep the engine alive while using it.
static torch::Tensor device_tensor(int64_t h, int64_t which) {
    Engine& e = get(h);
    long n;
    float* p = slab_ptr(e, which, n);
    return torch::from_blob(
        p, {n}, torch::TensorOptions().dtype(torch::kFloat32)
                    .device(torch::kCUDA, e.device_));
}
static void capture(int64_t h, int64_t n) { get(h).capture((int)n); }
static void replay(int64_t h, int64_t iters) { get(h).replay((int)iters); }
static void replay_async(int64_t h, int64_t iters) {
    get(h).replay_async((int)iters);
}
static void sync(int64_t h) { get(h).sync(); }

static py::dict counters(int64_t h) {
    Counters c = get(h).read_counters();
    py::dict d;
    d["beta_t"] = c.beta_t;
    d["adam_t_actor"] = c.adam_t_actor;
    d["adam_t_critic"] = c.adam_t_critic;
    d["size"] = c.size;
    d["pos"] = c.pos;
    d["max_priority"] = c.max_priority;
    d["loss_critic"] = c.loss_critic;
    d["loss_actor"] = c.loss_actor;
    return d;
}

// ---- GPU-resident rollout (device Pendulum + K11 noise + K15 fold) ----
static void rollout_alloc(int64_t h, int64_t M, int64_t n_steps,
                          int64_t horizon, double gamma, int64_t noise_kind,
                          double eps, double ou_theta, double ou_sigma,
                          double ou_mu, int64_t seed) {
    get(h).rollout_alloc((int)M, (int)n_steps, (int)horizon, (float)gamma,
                         (int)noise_kind, (float)eps, (float)ou_theta,
                         (float)ou_sigma, (float)ou_mu, (uint64_t)seed);
}

static void rollout_run(int64_t h, int64_t episodes, bool reset,
                        bool use_graph) {
    get(h).rollout_run((int)episodes, reset, use_graph);
}

static void rollout_set_state(int64_t h, torch::Tensor th,
                              torch::Tensor thdot) {
    Engine& e = get(h);
    auto t1 = f32_contig(th), t2 = f32_contig(thdot);
    TORCH_CHECK(t1.numel() == t2.numel(), "th/thdot size mismatch");
    e.rollout_set_state(t1.data_ptr<float>(), t2.data_ptr<float>(),
                        (int)t1.numel());
}

static py::dict rollout_info(int64_t h) {
    Engine& e = get(h);
    py::dict d;
    d["M"] = e.roll_.M;
    d["n"] = e.roll_.n;
    d["horizon"] = e.roll_.horizon;
    d["emits_per_episode"] = (long)e.roll_emits_per_ep * e.roll_.M;
    d["env_steps_per_episode"] = (long)e.roll_.horizon * e.roll_.M;
    return d;
}

static torch::Tensor actor_forward(int64_t h, torch::Tensor x) {
    Engine& e = get(h);
    auto t = f32_contig(x);
    int n = (int)(t.numel() / e.cfg.obs);
    auto out = torch::empty({n, (long)e.cfg.act}, torch::kFloat32);
    e.actor_forward(t.data_ptr<float>(), out.data_ptr<float>(), n);
    return out;
}

// Debug/parity access: copy a named device buffer to a CPU tensor.
static torch::Tensor read_buffer(int64_t h, std::string name) {
    Engine& e = get(h);
    const int B = e.cfg.batch, O = e.cfg.obs, A = e.cfg.act,
              H = e.cfg.hidden, K = e.cfg.atoms;
    // the persistent path double-buffers bs by step parity; resolve the
    // buffer the LAST completed step actually used
    if (name == "bs" && e.use_persistent()) {
        Counters c = e.read_counters();
        if (c.beta_t > 0 && ((c.beta_t - 1) & 1))
            name = "bs_b";
    }
    struct Ent { const void* p; std::vector<long> shape; bool is_long; bool is_double; };
    std::unordered_map<std::string, Ent> m = {
        {"bs", {e.bs, {B, O}, false, false}},
        {"bs_b", {e.bs_b, {B, O}, false, false}},
        {"ba", {e.ba, {B, A}, false, false}},
        {"br", {e.br, {B}, false, false}},
        {"bs2", {e.bs2, {B, O}, false, false}},
        {"bd", {e.bd, {B}, false, false}},
        {"bw", {e.bw, {B}, false, false}},
        {"pri", {e.pri, {B}, false, false}},
        {"bidx", {e.bidx, {B}, true, false}},
        {"a2", {e.a2, {B, A}, false, false}},
        {"p_t", {e.p_t, {B, K}, false, false}},
        {"m_proj", {e.m_proj, {B, K}, false, false}},
        {"q", {e.q, {B, K}, false, false}},
        {"pq", {e.pq, {B, K}, false, false}},
        {"a_out", {e.a_out, {B, A}, false, false}},
        {"dlog", {e.dlog, {B, K}, false, false}},
        {"c_h1", {e.c_h1, {B, H}, false, false}},
        {"c_h2", {e.c_h2, {B, H}, false, false}},
        {"c_h3", {e.c_h3, {B, H}, false, false}},
        {"at_h1", {e.at_h1, {B, H}, false, false}},
        {"tstamp", {e.tstamp, {64}, true, false}},
        {"sum_tree", {e.sum_tree, {2 * e.tree_cap}, false, true}},
        {"min_tree", {e.min_tree, {2 * e.tree_cap}, false, true}},
    };
    auto it = m.find(name);
    if (it == m.end()) throw std::runtime_error("unknown buffer " + name);
    auto& ent = it->second;
    long n = 1;
    for (long s : ent.shape) n *= s;
    torch::Tensor out;
    if (ent.is_long) {
        out = torch::empty(ent.shape, torch::kInt64);
        HIP_CHECK(hipMemcpy(out.data_ptr<int64_t>(), ent.p, n * 8,
                            hipMemcpyDeviceToHost));
    } else if (ent.is_double) {
        out = torch::empty(ent.shape, torch::kFloat64);
        HIP_CHECK(hipMemcpy(out.data_ptr<double>(), ent.p, n * 8,
                            hipMemcpyDeviceToHost));
    } else {
        out = torch::empty(ent.shape, torch::kFloat32);
        HIP_CHECK(hipMemcpy(out.data_ptr<float>(), ent.p, n * 4,
                            hipMemcpyDeviceToHost));
    }
    return out;
}

}  // namespace d4pg

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
    mod.def("create", &d4pg::create);
    mod.def("destroy", &d4pg::destroy);
    mod.def("info", &d4pg::info);
    mod.def("load_slab", &d4pg::load_slab);
    mod.def("store_slab", &d4pg::store_slab);
    mod.def("synth_fill", &d4pg::synth_fill);
    mod.def("ingest", &d4pg::ingest);
    mod.def("set_seed", &d4pg::set_seed);
    mod.def("set_schedule", &d4pg::set_schedule);
    mod.def("replay_state", &d4pg::replay_state);
    mod.def("load_replay_state", &d4pg::load_replay_state);
    mod.def("step", &d4pg::step);
    mod.def("step_part", &d4pg::step_part);
    mod.def("device_tensor", &d4pg::device_tensor);
    mod.def("capture", &d4pg::capture);
    mod.def("replay", &d4pg::replay);
    mod.def("replay_async", &d4pg::replay_async);
    mod.def("sync", &d4pg::sync);
    mod.def("counters", &d4pg::counters);
    mod.def("rollout_alloc", &d4pg::rollout_alloc);
    mod.def("rollout_run", &d4pg::rollout_run);
    mod.def("rollout_set_state", &d4pg::rollout_set_state);
    mod.def("rollout_info", &d4pg::rollout_info);
    mod.def("actor_forward", &d4pg::actor_forward);
    mod.def("read_buffer", &d4pg::read_buffer);
}
