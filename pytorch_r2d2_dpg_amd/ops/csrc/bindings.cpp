// Python bindings for the MI355X HIP kernels (K1-K7).
// Pure C++ TU: talks to the .hip translation units through extern "C"
// launchers taking raw pointers + the current HIP stream from torch.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <hip/hip_runtime.h>

#include <tuple>
#include <unordered_map>
#include <ATen/hip/HIPContext.h>

// ---- extern "C" launchers from the .hip TUs ------------------------------
extern "C" int launch_lstm_fwd_bf16(const void*, const void*, const void*, const void*,
                                    void*, void*, void*, float*, int, int, int, int,
                                    hipStream_t);
extern "C" int launch_lstm_fwd_f32(const void*, const void*, const void*, const void*,
                                   void*, void*, void*, float*, int, int, int, int,
                                   hipStream_t);
extern "C" int launch_lstm_bwd_bf16(const void*, const void*, const void*, const void*,
                                    const float*, const void*, const void*, void*, void*,
                                    void*, int, int, int, hipStream_t);
extern "C" int launch_lstm_bwd_f32(const void*, const void*, const void*, const void*,
                                   const float*, const void*, const void*, void*, void*,
                                   void*, int, int, int, hipStream_t);
extern "C" void launch_td_targets(const float*, const float*, const float*, const float*,
                                  const int*, float*, float*, float*, int, int, int, int,
                                  float, float, int, float, hipStream_t);
extern "C" void launch_polyak_multi(const unsigned long long*, const long*, int, long,
                                    float, int, hipStream_t);
extern "C" void launch_sumtree_sample(const float*, int, int, unsigned long long,
                                      unsigned long long, long*, float*, hipStream_t);
extern "C" void launch_sumtree_update(float*, int, const long*, const float*, int,
                                      hipStream_t);
extern "C" void launch_fp8_encode(const float*, unsigned char*, long, float, hipStream_t);
extern "C" void launch_fp8_gather_decode(const unsigned char*, const long*, void*, int,
                                         long, int, float, hipStream_t);
extern "C" int launch_lstm_fwd_px_bf16(const void*, const void*, const void*, const void*,
                                       void*, void*, void*, float*, unsigned long long*,
                                       unsigned int*, int, int, int, int, int, hipStream_t);
extern "C" int launch_lstm_fwd_px2_bf16(const void*, const void*, const void*, const void*,
                                        void*, void*, void*, float*, int, int,
                                        unsigned long long*,
                                        const void*, const void*, const void*, const void*,
                                        void*, void*, void*, float*, int, int,
                                        unsigned long long*,
                                        unsigned int*, int, int, int, hipStream_t);
extern "C" int launch_lstm_bwd_px_bf16(const void*, const void*, const void*, const void*,
                                       const float*, const void*, const void*, void*, void*,
                                       void*, unsigned char*, unsigned long long*,
                                       unsigned int*, int, int, int, int, hipStream_t);
extern "C" int launch_lstm_bwd_px2_bf16(
    const void*, const void*, const void*, const void*, const float*, const void*,
    const void*, void*, void*, void*, unsigned char*, unsigned long long*, int,
    const void*, const void*, const void*, const void*, const float*, const void*,
    const void*, void*, void*, void*, unsigned char*, unsigned long long*, int,
    unsigned int*, int, int, int, hipStream_t);

namespace {

hipStream_t cur_stream() {
    return at::hip::getCurrentHIPStream().stream();
}

// Column-split kernels spin on cross-block exchanges, so two of them
// running concurrently contend for CU residency (measured 3x slower than
// back-to-back). Chain every px launch behind the previous one with a
// per-device event — px kernels serialize across ALL streams while the
// surrounding GEMMs/elementwise still overlap them.
bool g_px_serialize = false;  // default off: measured best at B<=256

hipEvent_t px_chain_event(int dev) {
    static std::unordered_map<int, hipEvent_t> events;
    auto it = events.find(dev);
    if (it == events.end()) {
        hipEvent_t ev;
        (void)hipEventCreateWithFlags(&ev, hipEventDisableTiming);
        it = events.emplace(dev, ev).first;
    }
    return it->second;
}

void px_chain_wait(hipStream_t st) {
    if (g_px_serialize)
        (void)hipStreamWaitEvent(st, px_chain_event(at::cuda::current_device()), 0);
}

void px_chain_record(hipStream_t st) {
    if (g_px_serialize)
        (void)hipEventRecord(px_chain_event(at::cuda::current_device()), st);
}

void set_px_serialize(bool on) { g_px_serialize = on; }

bool stream_capturing();  // defined with the pack cache below

// ---- CU-budget accounting for column-split grids ---------------------------
// Every px kernel requires ALL its blocks co-resident (cross-block spins), so
// the blocks of all CONCURRENTLY-RUNNING px grids must sum to <= 224 of the
// 256 CUs. The host tracks enqueued-and-possibly-running px launches via
// events; each new launch gets the remaining budget and the launcher picks a
// smaller CG (fewer blocks) under contention. If nothing fits, the caller's
// stream first waits on every in-flight px event (back-to-back px is fast;
// CONCURRENT over-subscribed px livelocks until the spin bound). GIL
// serializes all access. R2D2_PX_NO_BUDGET=1 restores the fixed 224 limit.
struct PxInflight { hipEvent_t ev; int blocks; hipStream_t st; };
long g_px_launches = 0, g_px_drains = 0, g_px_reserve_min = 224;
std::vector<PxInflight> g_px_inflight;       // current device only (per-proc 1 GPU)
std::vector<hipEvent_t> g_px_event_pool;
bool g_px_budget_enabled = []() {
    const char* e = getenv("R2D2_PX_NO_BUDGET");
    return !(e && e[0] && e[0] != '0');
}();

// What can actually run CONCURRENTLY with a new launch on stream `cur`:
// at most ONE kernel per OTHER stream (streams execute in order), so the
// consumed capacity is the per-other-stream MAX of in-flight cu_eq, not
// the sum of the queue. (Summing the queue was measured catastrophic: the
// host enqueues many steps ahead of the GPU, so dozens of already-ordered
// launches looked "in flight" and ~86% of launches drained, serializing
// the overlap — 6.3 vs 5.2 ms/step.)
int px_budget_reserve(hipStream_t cur) {
    if (!g_px_budget_enabled) return 224;
    // during hipGraph capture: event record/query would invalidate the
    // capture — the graph replays a fixed schedule anyway, so run unbudgeted
    if (stream_capturing()) return 224;
    // sweep completed entries
    for (size_t i = 0; i < g_px_inflight.size();) {
        if (hipEventQuery(g_px_inflight[i].ev) == hipSuccess) {
            g_px_event_pool.push_back(g_px_inflight[i].ev);
            g_px_inflight[i] = g_px_inflight.back();
            g_px_inflight.pop_back();
        } else {
            ++i;
        }
    }
    int used = 0;
    for (size_t i = 0; i < g_px_inflight.size(); ++i) {
        if (g_px_inflight[i].st == cur) continue;
        int mx = g_px_inflight[i].blocks;
        bool counted = false;
        for (size_t j = 0; j < i; ++j)
            if (g_px_inflight[j].st == g_px_inflight[i].st &&
                g_px_inflight[j].st != cur) { counted = true; break; }
        if (counted) continue;
        for (size_t j = i + 1; j < g_px_inflight.size(); ++j)
            if (g_px_inflight[j].st == g_px_inflight[i].st &&
                g_px_inflight[j].blocks > mx)
                mx = g_px_inflight[j].blocks;
        used += mx;
    }
    return 224 - used;
}

// make `st` wait for every OTHER-stream in-flight px grid, then return the
// full budget (same-stream grids are already ordered before the launch)
int px_budget_drain(hipStream_t st) {
    ++g_px_drains;
    for (auto& f : g_px_inflight)
        if (f.st != st)
            (void)hipStreamWaitEvent(st, f.ev, 0);
    return 224;
}

void px_budget_commit(int blocks, hipStream_t st) {
    ++g_px_launches;
    if (!g_px_budget_enabled) return;
    if (stream_capturing()) return;  // see px_budget_reserve
    hipEvent_t ev;
    if (!g_px_event_pool.empty()) {
        ev = g_px_event_pool.back();
        g_px_event_pool.pop_back();
    } else {
        (void)hipEventCreateWithFlags(&ev, hipEventDisableTiming);
    }
    (void)hipEventRecord(ev, st);
    g_px_inflight.push_back({ev, blocks, st});
}

// Per-device error word for the column-split kernel's bounded spins.
torch::Tensor& px_err_word(const torch::Device& dev) {
    static std::unordered_map<int, torch::Tensor> words;
    int idx = dev.index() < 0 ? 0 : dev.index();
    auto it = words.find(idx);
    if (it == words.end())
        it = words.emplace(idx, torch::zeros({1}, torch::dtype(torch::kInt32).device(dev))).first;
    return it->second;
}

std::tuple<int64_t, int64_t, int64_t> px_budget_stats() {
    auto r = std::make_tuple((int64_t)g_px_launches, (int64_t)g_px_drains,
                             (int64_t)g_px_inflight.size());
    return r;
}

// Read-and-clear the column-split spin-timeout error word (0 = healthy).
int64_t lstm_px_error() {
    int64_t total = 0;
    // check word on the current device only (tests run single-GPU)
    if (torch::cuda::is_available()) {
        auto& w = px_err_word(torch::Device(torch::kCUDA, at::cuda::current_device()));
        total = w.item<int32_t>();
        w.zero_();
    }
    return total;
}

// Pack a [R, C] weight into the MFMA-fragment-major layout the LSTM kernels
// load with lane-linear coalesced instructions: [R/16][C/KSTEP][64][KPL],
// where lane l of block (rb, cb) holds row rb*16+(l&15), cols
// cb*KSTEP+(l>>4)*KPL ... +KPL. KSTEP/KPL: bf16 32/8, f32 4/1.
torch::Tensor pack_fragment_major(const torch::Tensor& w) {
    const auto R = w.size(0), C = w.size(1);
    const int kstep = w.scalar_type() == torch::kBFloat16 ? 32 : 4;
    const int kpl = kstep / 4;
    return w.view({R / 16, 16, C / kstep, 4, kpl})
        .permute({0, 2, 3, 1, 4})
        .contiguous();
}

// ---- packed-weight cache ---------------------------------------------------
// The fragment-major packs are pure functions of the weight VALUES, which
// change only at optimizer/Polyak writes; the learner bumps the version once
// per step after those writes (learner.py::_bump_weight_version) and every
// LSTM entry point reuses the cached pack until then — the flagship step
// packs the critic Wh 4x and its transpose 1-2x otherwise.
// Stream-ordering contract (docs/STREAMS.md): a cache MISS enqueues the
// repack on the caller's current stream; consumers on other streams are
// already ordered behind it by the learner's burn-in -> ev_ba fencing (the
// first pack of each step happens on the main stream before ev_ba). Entries
// are held by the map, so the caching allocator never recycles them.
// Keyed by data_ptr: weights are long-lived parameters whose storage is
// stable; a version bump also clears entries whose shape changed.
// Entries PIN the source tensor (`src`): while cached, its storage cannot
// be freed, so a later allocation can never reuse the data_ptr key — the
// classic stale-hit hazard for tensors created/dropped outside the learner
// (unit tests). A hit additionally requires BOTH the learner's explicit
// version (covers the K4 Polyak raw-pointer writes torch can't see) and
// the tensor's autograd version counter (covers in-place torch mutation
// without a bump) to match.
struct PackEntry {
    uint64_t gver;
    uint32_t tver;
    torch::Tensor src;
    torch::Tensor packed;
};
uint64_t g_weight_version = 0;
std::unordered_map<const void*, PackEntry> g_pack_cache;    // w -> pack(w)
std::unordered_map<const void*, PackEntry> g_pack_cache_T;  // w -> pack(w^T)

void bump_weight_version() {
    ++g_weight_version;
    if (g_pack_cache.size() > 64) g_pack_cache.clear();       // bound pinning
    if (g_pack_cache_T.size() > 64) g_pack_cache_T.clear();
}

bool stream_capturing() {
    hipStreamCaptureStatus st = hipStreamCaptureStatusNone;
    (void)hipStreamIsCapturing(cur_stream(), &st);
    return st != hipStreamCaptureStatusNone;
}

torch::Tensor pack_cached(const torch::Tensor& w, bool transpose) {
    // under hipGraph capture the repack ops must be RECORDED (the optimizer
    // inside the graph mutates weights without a host-side version bump), so
    // bypass the cache there
    if (stream_capturing())
        return transpose ? pack_fragment_major(w.t().contiguous())
                         : pack_fragment_major(w);
    auto& cache = transpose ? g_pack_cache_T : g_pack_cache;
    const void* key = w.data_ptr();
    const uint32_t tver = w.unsafeGetTensorImpl()->version_counter().current_version();
    auto it = cache.find(key);
    if (it != cache.end() && it->second.gver == g_weight_version &&
        it->second.tver == tver && it->second.src.data_ptr() == key &&
        it->second.packed.numel() == w.numel())
        return it->second.packed;
    auto packed = transpose ? pack_fragment_major(w.t().contiguous())
                            : pack_fragment_major(w);
    cache[key] = {g_weight_version, tver, w, packed};
    return packed;
}

void check_dev_contig(const torch::Tensor& t, const char* name) {
    TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
    TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

// ---- K1: fused LSTM sequence forward -------------------------------------
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor>
lstm_seq_fwd(torch::Tensor gates_x, torch::Tensor h0, torch::Tensor c0,
             torch::Tensor wh, bool stash = true) {
    check_dev_contig(gates_x, "gates_x");
    check_dev_contig(h0, "h0");
    check_dev_contig(c0, "c0");
    check_dev_contig(wh, "wh");
    const auto T = gates_x.size(0);
    const auto B = gates_x.size(1);
    const auto H = h0.size(1);
    TORCH_CHECK(gates_x.size(2) == 4 * H, "gates_x last dim must be 4H");
    TORCH_CHECK(H % 64 == 0, "hidden size must be a multiple of 64 (got ", H, ")");
    auto dt = gates_x.scalar_type();
    TORCH_CHECK(dt == torch::kBFloat16 || dt == torch::kFloat32,
                "lstm_seq_fwd supports bf16/fp32 only");
    TORCH_CHECK(h0.scalar_type() == dt && wh.scalar_type() == dt && c0.scalar_type() == dt,
                "dtype mismatch");

    auto wh_p = pack_cached(wh, /*transpose=*/false);
    auto h_seq = torch::empty({T, B, H}, gates_x.options());
    auto c_final = torch::empty({B, H}, gates_x.options());
    auto gates_act = torch::empty({stash ? T : 0, B, 4 * H}, gates_x.options());
    auto c_seq = torch::empty({stash ? T : 0, B, H},
                              gates_x.options().dtype(torch::kFloat32));

    int rc;
    if (dt == torch::kBFloat16) {
        // column-split kernel first (CG-way less Wh streaming per CU); it
        // declines geometries where its blocks can't all be co-resident
        const int RT = (int)((B + 15) / 16);
        auto hx = torch::zeros({2LL, RT, 16, H / 2},
                               gates_x.options().dtype(torch::kInt64));
        auto& err = px_err_word(gates_x.device());
        px_chain_wait(cur_stream());
        int budget = px_budget_reserve(cur_stream());
        auto try_px = [&](int bud) {
            return launch_lstm_fwd_px_bf16(
                gates_x.data_ptr(), h0.data_ptr(), c0.data_ptr(), wh_p.data_ptr(),
                h_seq.data_ptr(), c_final.data_ptr(), gates_act.data_ptr(),
                stash ? c_seq.data_ptr<float>() : nullptr,
                (unsigned long long*)hx.data_ptr<int64_t>(),
                (unsigned int*)err.data_ptr<int32_t>(),
                (int)T, (int)B, (int)H, stash ? 1 : 0, bud, cur_stream());
        };
        rc = try_px(budget);
        if (rc == -2)  // would over-subscribe: serialize behind in-flight px
            rc = try_px(px_budget_drain(cur_stream()));
        if (rc > 0) {
            px_budget_commit(rc, cur_stream());
            px_chain_record(cur_stream());
            rc = 0;
        }
        if (rc != 0)
            rc = launch_lstm_fwd_bf16(gates_x.data_ptr(), h0.data_ptr(), c0.data_ptr(),
                                      wh_p.data_ptr(), h_seq.data_ptr(), c_final.data_ptr(),
                                      gates_act.data_ptr(),
                                      stash ? c_seq.data_ptr<float>() : nullptr,
                                      (int)T, (int)B, (int)H, stash ? 1 : 0, cur_stream());
    }
    else
        rc = launch_lstm_fwd_f32(gates_x.data_ptr(), h0.data_ptr(), c0.data_ptr(),
                                 wh_p.data_ptr(), h_seq.data_ptr(), c_final.data_ptr(),
                                 gates_act.data_ptr(),
                                 stash ? c_seq.data_ptr<float>() : nullptr,
                                 (int)T, (int)B, (int)H, stash ? 1 : 0, cur_stream());
    TORCH_CHECK(rc == 0, "unsupported LSTM geometry H=", H);
    return {h_seq, c_final, gates_act, c_seq};
}

// ---- K1-pair: two independent LSTM unrolls in ONE launch ------------------
// Same B/H; T and stash may differ per set (set with stash=true gets its
// activations + fp32 cell states written for the K2 backward — used to pair
// the no-grad target-actor unroll with the grad-tracked online critic fwd).
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor,
           torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor>
lstm_seq_fwd_pair(torch::Tensor gx0, torch::Tensor h00, torch::Tensor c00,
                  torch::Tensor wh0, torch::Tensor gx1, torch::Tensor h01,
                  torch::Tensor c01, torch::Tensor wh1,
                  bool stash0 = false, bool stash1 = false) {
    for (auto* t : {&gx0, &h00, &c00, &wh0, &gx1, &h01, &c01, &wh1})
        check_dev_contig(*t, "pair arg");
    const auto T0 = gx0.size(0);
    const auto T1 = gx1.size(0);
    const auto B = gx0.size(1);
    const auto H = h00.size(1);
    TORCH_CHECK(gx1.size(1) == B && h01.size(1) == H, "pair shape mismatch");
    TORCH_CHECK(gx0.scalar_type() == torch::kBFloat16, "pair path is bf16-only");
    auto wp0 = pack_cached(wh0, /*transpose=*/false);
    auto wp1 = pack_cached(wh1, /*transpose=*/false);
    const int RT = (int)((B + 15) / 16);
    auto hx0 = torch::zeros({2LL, RT, 16, H / 2}, gx0.options().dtype(torch::kInt64));
    auto hx1 = torch::zeros({2LL, RT, 16, H / 2}, gx0.options().dtype(torch::kInt64));
    auto hs0 = torch::empty({T0, B, H}, gx0.options());
    auto hs1 = torch::empty({T1, B, H}, gx0.options());
    auto cf0 = torch::empty({B, H}, gx0.options());
    auto cf1 = torch::empty({B, H}, gx0.options());
    auto ga0 = torch::empty({stash0 ? T0 : 0, B, 4 * H}, gx0.options());
    auto ga1 = torch::empty({stash1 ? T1 : 0, B, 4 * H}, gx0.options());
    auto cs0 = torch::empty({stash0 ? T0 : 0, B, H},
                            gx0.options().dtype(torch::kFloat32));
    auto cs1 = torch::empty({stash1 ? T1 : 0, B, H},
                            gx0.options().dtype(torch::kFloat32));
    auto& err = px_err_word(gx0.device());
    int budget = px_budget_reserve(cur_stream());
    auto try_px = [&](int bud) {
        return launch_lstm_fwd_px2_bf16(
            gx0.data_ptr(), h00.data_ptr(), c00.data_ptr(), wp0.data_ptr(),
            hs0.data_ptr(), cf0.data_ptr(), ga0.data_ptr(),
            stash0 ? cs0.data_ptr<float>() : nullptr, stash0 ? 1 : 0, (int)T0,
            (unsigned long long*)hx0.data_ptr<int64_t>(),
            gx1.data_ptr(), h01.data_ptr(), c01.data_ptr(), wp1.data_ptr(),
            hs1.data_ptr(), cf1.data_ptr(), ga1.data_ptr(),
            stash1 ? cs1.data_ptr<float>() : nullptr, stash1 ? 1 : 0, (int)T1,
            (unsigned long long*)hx1.data_ptr<int64_t>(),
            (unsigned int*)err.data_ptr<int32_t>(), (int)B, (int)H, bud,
            cur_stream());
    };
    int rc = try_px(budget);
    if (rc == -2)
        rc = try_px(px_budget_drain(cur_stream()));
    TORCH_CHECK(rc > 0, "pair geometry unsupported (fall back to two calls)");
    px_budget_commit(rc, cur_stream());
    return {hs0, cf0, ga0, cs0, hs1, cf1, ga1, cs1};
}

// ---- K2: fused LSTM sequence backward -------------------------------------
// Takes the UNTRANSPOSED weight wh [4H, H] (the parameter tensor itself) so
// the transposed fragment-major pack can be cached across calls keyed on the
// parameter's stable data_ptr (pack_cached).
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor>
lstm_seq_bwd(torch::Tensor dh_seq, torch::Tensor dhT, torch::Tensor dcT,
             torch::Tensor gates_act, torch::Tensor c_seq, torch::Tensor c0,
             torch::Tensor wh) {
    check_dev_contig(dh_seq, "dh_seq");
    check_dev_contig(dhT, "dhT");
    check_dev_contig(dcT, "dcT");
    check_dev_contig(gates_act, "gates_act");
    check_dev_contig(c_seq, "c_seq");
    check_dev_contig(c0, "c0");
    check_dev_contig(wh, "wh");
    const auto T = dh_seq.size(0);
    const auto B = dh_seq.size(1);
    const auto H = dh_seq.size(2);
    TORCH_CHECK(wh.size(0) == 4 * H && wh.size(1) == H, "wh must be [4H, H]");
    auto dt = dh_seq.scalar_type();

    auto whT_p = pack_cached(wh, /*transpose=*/true);
    auto dgates = torch::empty({T, B, 4 * H}, dh_seq.options());
    auto dh0 = torch::empty({B, H}, dh_seq.options());
    auto dc0 = torch::empty({B, H}, dh_seq.options());

    int rc;
    if (dt == torch::kBFloat16) {
        const int RT = (int)((B + 15) / 16);
        // ZEROS, not empty: the granule-exchange backward reinterprets this
        // as tagged u64 granules (8 B per 2 bf16 payload = 2x the slab
        // image) — and garbage must never alias a valid tag
        auto slab = torch::zeros({2LL, RT, 16 * 4 * H * 4},
                                 dh_seq.options().dtype(torch::kUInt8));
        auto flags = torch::zeros({2LL, RT, 8}, dh_seq.options().dtype(torch::kInt64));
        auto& err = px_err_word(dh_seq.device());
        px_chain_wait(cur_stream());
        int budget = px_budget_reserve(cur_stream());
        auto try_px = [&](int bud) {
            return launch_lstm_bwd_px_bf16(
                dh_seq.data_ptr(), dhT.data_ptr(), dcT.data_ptr(), gates_act.data_ptr(),
                c_seq.data_ptr<float>(), c0.data_ptr(), whT_p.data_ptr(),
                dgates.data_ptr(), dh0.data_ptr(), dc0.data_ptr(),
                slab.data_ptr<uint8_t>(), (unsigned long long*)flags.data_ptr<int64_t>(),
                (unsigned int*)err.data_ptr<int32_t>(),
                (int)T, (int)B, (int)H, bud, cur_stream());
        };
        rc = try_px(budget);
        if (rc == -2)
            rc = try_px(px_budget_drain(cur_stream()));
        if (rc > 0) {
            px_budget_commit(rc, cur_stream());
            px_chain_record(cur_stream());
            rc = 0;
        }
        if (rc != 0)
            rc = launch_lstm_bwd_bf16(dh_seq.data_ptr(), dhT.data_ptr(), dcT.data_ptr(),
                                      gates_act.data_ptr(), c_seq.data_ptr<float>(),
                                      c0.data_ptr(), whT_p.data_ptr(), dgates.data_ptr(),
                                      dh0.data_ptr(), dc0.data_ptr(),
                                      (int)T, (int)B, (int)H, cur_stream());
    }
    else
        rc = launch_lstm_bwd_f32(dh_seq.data_ptr(), dhT.data_ptr(), dcT.data_ptr(),
                                 gates_act.data_ptr(), c_seq.data_ptr<float>(),
                                 c0.data_ptr(), whT_p.data_ptr(), dgates.data_ptr(),
                                 dh0.data_ptr(), dc0.data_ptr(),
                                 (int)T, (int)B, (int)H, cur_stream());
    TORCH_CHECK(rc == 0, "unsupported LSTM geometry H=", H);
    return {dgates, dh0, dc0};
}

// ---- K2-pair: two independent LSTM backwards in ONE launch ----------------
// Same B/H; T may differ. Each set gets its own slab/flags workspace. Used
// for the critic-loss backward + the actor-chain critic backward, which
// become ready at nearly the same point of the learner step.
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor,
           torch::Tensor, torch::Tensor, torch::Tensor>
lstm_seq_bwd_pair(torch::Tensor dh0s, torch::Tensor dhT0, torch::Tensor dcT0,
                  torch::Tensor ga0, torch::Tensor cs0, torch::Tensor c00,
                  torch::Tensor wh0,
                  torch::Tensor dh1s, torch::Tensor dhT1, torch::Tensor dcT1,
                  torch::Tensor ga1, torch::Tensor cs1, torch::Tensor c01,
                  torch::Tensor wh1) {
    for (auto* t : {&dh0s, &dhT0, &dcT0, &ga0, &c00, &wh0,
                    &dh1s, &dhT1, &dcT1, &ga1, &c01, &wh1})
        check_dev_contig(*t, "bwd pair arg");
    const auto T0 = dh0s.size(0);
    const auto T1 = dh1s.size(0);
    const auto B = dh0s.size(1);
    const auto H = dh0s.size(2);
    TORCH_CHECK(dh1s.size(1) == B && dh1s.size(2) == H, "bwd pair shape mismatch");
    TORCH_CHECK(dh0s.scalar_type() == torch::kBFloat16, "bwd pair is bf16-only");
    TORCH_CHECK(wh0.size(0) == 4 * H && wh0.size(1) == H, "wh must be [4H, H]");
    auto wp0 = pack_cached(wh0, /*transpose=*/true);
    auto wp1 = pack_cached(wh1, /*transpose=*/true);
    const int RT = (int)((B + 15) / 16);
    auto opts = dh0s.options();
    auto dg0 = torch::empty({T0, B, 4 * H}, opts);
    auto dg1 = torch::empty({T1, B, 4 * H}, opts);
    auto dh00 = torch::empty({B, H}, opts);
    auto dh01 = torch::empty({B, H}, opts);
    auto dc00 = torch::empty({B, H}, opts);
    auto dc01 = torch::empty({B, H}, opts);
    // zeros: see lstm_seq_bwd (tagged-granule workspace, 2x the slab image)
    auto slab0 = torch::zeros({2LL, RT, 16 * 4 * H * 4}, opts.dtype(torch::kUInt8));
    auto slab1 = torch::zeros({2LL, RT, 16 * 4 * H * 4}, opts.dtype(torch::kUInt8));
    auto flags0 = torch::zeros({2LL, RT, 8}, opts.dtype(torch::kInt64));
    auto flags1 = torch::zeros({2LL, RT, 8}, opts.dtype(torch::kInt64));
    auto& err = px_err_word(dh0s.device());
    int budget = px_budget_reserve(cur_stream());
    auto try_px = [&](int bud) {
        return launch_lstm_bwd_px2_bf16(
            dh0s.data_ptr(), dhT0.data_ptr(), dcT0.data_ptr(), ga0.data_ptr(),
            cs0.data_ptr<float>(), c00.data_ptr(), wp0.data_ptr(),
            dg0.data_ptr(), dh00.data_ptr(), dc00.data_ptr(),
            slab0.data_ptr<uint8_t>(), (unsigned long long*)flags0.data_ptr<int64_t>(),
            (int)T0,
            dh1s.data_ptr(), dhT1.data_ptr(), dcT1.data_ptr(), ga1.data_ptr(),
            cs1.data_ptr<float>(), c01.data_ptr(), wp1.data_ptr(),
            dg1.data_ptr(), dh01.data_ptr(), dc01.data_ptr(),
            slab1.data_ptr<uint8_t>(), (unsigned long long*)flags1.data_ptr<int64_t>(),
            (int)T1,
            (unsigned int*)err.data_ptr<int32_t>(), (int)B, (int)H, bud,
            cur_stream());
    };
    int rc = try_px(budget);
    if (rc == -2)
        rc = try_px(px_budget_drain(cur_stream()));
    TORCH_CHECK(rc > 0, "bwd pair geometry unsupported (fall back to two calls)");
    px_budget_commit(rc, cur_stream());
    return {dg0, dh00, dc00, dg1, dh01, dc01};
}

// ---- K3: fused n-step TD targets + priorities -----------------------------
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor>
td_targets(torch::Tensor rewards, torch::Tensor dones, torch::Tensor q_boot,
           torch::Tensor q_online, double gamma, int64_t n_step, double eta,
           bool rescale = false, double rescale_eps = 1e-3,
           c10::optional<torch::Tensor> valid = c10::nullopt) {
    check_dev_contig(rewards, "rewards");
    check_dev_contig(dones, "dones");
    check_dev_contig(q_boot, "q_boot");
    check_dev_contig(q_online, "q_online");
    const auto B = q_online.size(0);
    const auto T = q_online.size(1);
    const auto t_ext = rewards.size(1);
    TORCH_CHECK(rewards.scalar_type() == torch::kFloat32, "rewards must be fp32");
    const int* valid_ptr = nullptr;
    if (valid.has_value()) {
        check_dev_contig(*valid, "valid");
        TORCH_CHECK(valid->scalar_type() == torch::kInt32 && valid->size(0) == B,
                    "valid must be int32 [B]");
        valid_ptr = valid->data_ptr<int>();
    }
    auto y = torch::empty({B, T}, q_online.options());
    auto td_abs = torch::empty({B, T}, q_online.options());
    auto prio = torch::empty({B}, q_online.options());
    launch_td_targets(rewards.data_ptr<float>(), dones.data_ptr<float>(),
                      q_boot.data_ptr<float>(), q_online.data_ptr<float>(),
                      valid_ptr, y.data_ptr<float>(), td_abs.data_ptr<float>(),
                      prio.data_ptr<float>(), (int)B, (int)T, (int)t_ext,
                      (int)n_step, (float)gamma, (float)eta,
                      rescale ? 1 : 0, (float)rescale_eps, cur_stream());
    return {y, td_abs, prio};
}

// ---- K4: fused multi-tensor Polyak ----------------------------------------
void polyak_multi(torch::Tensor ptr_table, torch::Tensor off_table, int64_t total,
                  double tau, bool is_bf16) {
    check_dev_contig(ptr_table, "ptr_table");
    check_dev_contig(off_table, "off_table");
    int n = (int)(off_table.size(0) - 1);
    launch_polyak_multi(
        reinterpret_cast<const unsigned long long*>(ptr_table.data_ptr<int64_t>()),
        reinterpret_cast<const long*>(off_table.data_ptr<int64_t>()), n,
        (long)total, (float)tau, is_bf16 ? 1 : 0, cur_stream());
}

// ---- K5: sum-tree stratified sample ---------------------------------------
std::tuple<torch::Tensor, torch::Tensor>
sumtree_sample(torch::Tensor tree, int64_t cap, int64_t batch, int64_t seed,
               int64_t step) {
    check_dev_contig(tree, "tree");
    TORCH_CHECK(tree.scalar_type() == torch::kFloat32, "tree must be fp32");
    auto idx = torch::empty({batch}, tree.options().dtype(torch::kInt64));
    auto prob = torch::empty({batch}, tree.options());
    launch_sumtree_sample(tree.data_ptr<float>(), (int)cap, (int)batch,
                          (unsigned long long)seed, (unsigned long long)step,
                          (long*)idx.data_ptr<int64_t>(), prob.data_ptr<float>(),
                          cur_stream());
    return {idx, prob};
}

// ---- K6: sum-tree leaf update + level-synchronous propagate ---------------
void sumtree_update(torch::Tensor tree, int64_t cap, torch::Tensor idx,
                    torch::Tensor prio) {
    check_dev_contig(tree, "tree");
    check_dev_contig(idx, "idx");
    check_dev_contig(prio, "prio");
    TORCH_CHECK(idx.scalar_type() == torch::kInt64 && prio.scalar_type() == torch::kFloat32,
                "idx must be int64, prio fp32");
    launch_sumtree_update(tree.data_ptr<float>(), (int)cap,
                          (const long*)idx.data_ptr<int64_t>(),
                          prio.data_ptr<float>(), (int)idx.size(0), cur_stream());
}

// ---- K7: fp8 e4m3 encode / fused gather+decode ----------------------------
void fp8_encode(torch::Tensor src, torch::Tensor dst, double scale) {
    check_dev_contig(src, "src");
    check_dev_contig(dst, "dst");
    TORCH_CHECK(src.scalar_type() == torch::kFloat32 && dst.scalar_type() == torch::kUInt8,
                "fp8_encode: src fp32 -> dst u8");
    TORCH_CHECK(src.numel() == dst.numel(), "numel mismatch");
    launch_fp8_encode(src.data_ptr<float>(), dst.data_ptr<uint8_t>(), src.numel(),
                      (float)scale, cur_stream());
}

torch::Tensor fp8_gather_decode(torch::Tensor store, torch::Tensor idx,
                                torch::ScalarType out_dtype, double inv_scale) {
    check_dev_contig(store, "store");
    check_dev_contig(idx, "idx");
    TORCH_CHECK(store.scalar_type() == torch::kUInt8, "store must be u8 (fp8 e4m3)");
    const auto B = idx.size(0);
    long row = 1;
    for (int d = 1; d < store.dim(); ++d) row *= store.size(d);
    std::vector<int64_t> shape;
    shape.push_back(B);
    for (int d = 1; d < store.dim(); ++d) shape.push_back(store.size(d));
    auto out = torch::empty(shape, store.options().dtype(out_dtype));
    launch_fp8_gather_decode(store.data_ptr<uint8_t>(),
                             (const long*)idx.data_ptr<int64_t>(), out.data_ptr(),
                             out_dtype == torch::kBFloat16 ? 1 : 0, row, (int)B,
                             (float)inv_scale, cur_stream());
    return out;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("lstm_seq_fwd", &lstm_seq_fwd, "fused LSTM sequence forward (K1)",
          py::arg("gates_x"), py::arg("h0"), py::arg("c0"), py::arg("wh"),
          py::arg("stash") = true);
    m.def("lstm_seq_bwd", &lstm_seq_bwd, "fused LSTM sequence backward (K2)");
    m.def("td_targets", &td_targets, "fused n-step TD targets + priorities (K3)",
          py::arg("rewards"), py::arg("dones"), py::arg("q_boot"), py::arg("q_online"),
          py::arg("gamma"), py::arg("n_step"), py::arg("eta"),
          py::arg("rescale") = false, py::arg("rescale_eps") = 1e-3,
          py::arg("valid") = py::none());
    m.def("polyak_multi", &polyak_multi, "fused multi-tensor Polyak update (K4)");
    m.def("sumtree_sample", &sumtree_sample, "sum-tree stratified sample (K5)");
    m.def("sumtree_update", &sumtree_update, "sum-tree leaf update + propagate (K6)");
    m.def("fp8_encode", &fp8_encode, "fp8 e4m3 encode (K7)");
    m.def("fp8_gather_decode", &fp8_gather_decode, "fp8 fused gather+decode (K7)",
          py::arg("store"), py::arg("idx"), py::arg("out_dtype"),
          py::arg("inv_scale") = 1.0);
    m.def("lstm_px_error", &lstm_px_error,
          "read-and-clear the column-split LSTM spin-timeout error word");
    m.def("lstm_seq_fwd_pair", &lstm_seq_fwd_pair,
          "two independent LSTM unrolls in one launch (per-set T/stash)",
          py::arg("gx0"), py::arg("h00"), py::arg("c00"), py::arg("wh0"),
          py::arg("gx1"), py::arg("h01"), py::arg("c01"), py::arg("wh1"),
          py::arg("stash0") = false, py::arg("stash1") = false);
    m.def("lstm_seq_bwd_pair", &lstm_seq_bwd_pair,
          "two independent LSTM backwards in one launch (per-set T)");
    m.def("set_px_serialize", &set_px_serialize,
          "serialize column-split kernels device-wide (off by default)");
    m.def("px_budget_stats", &px_budget_stats,
          "(px launches, budget drains, inflight entries)");
    m.def("bump_weight_version", &bump_weight_version,
          "invalidate the fragment-major weight-pack cache (call after every "
          "optimizer/Polyak write)");
}
