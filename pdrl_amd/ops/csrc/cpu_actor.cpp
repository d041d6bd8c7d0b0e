// CPU batched actor inference for the worker processes.
//
// The actor plane is CPU worker processes (reference topology kept:
// SURVEY.md §2.5 — actor parallelism is the reference's only parallelism).
// Its hot loop is `model.act` per tick: in eager PyTorch that is ~15 small
// op dispatches on (M≤8, H=64) tensors — dispatch-overhead bound, and the
// measured single-box ingest ceiling. This extension runs the whole act —
// body GEMM+ReLU, one LSTM cell step, the logits head, categorical
// sampling and log-prob — as ONE C++ call over the batch of envs, on the
// same transposed weight layout as the HIP kernels / eager core
// (networks/models.py SeqLSTMCore), so the worker swaps it in without any
// model surgery. Discrete policies only (the continuous actor also needs
// tanh-Gaussian sampling; its workers keep the eager path).
//
// Vector-friendly loop order: inner loops stream rows of the transposed
// weights (unit stride) so -O3 autovectorizes them; no torch ops inside.
#include <pybind11/pybind11.h>
#include <torch/extension.h>

#include <cmath>
#include <cstdint>
#include <vector>

namespace {

inline float sigf(float x) { return 1.0f / (1.0f + std::exp(-x)); }

inline uint64_t xorshift64(uint64_t& s) {
  s ^= s << 13;
  s ^= s >> 7;
  s ^= s << 17;
  return s;
}

}  // namespace

// obs (M,F), hx/cx (M,H) fp32; weights in the SeqLSTMCore transposed layout;
// logits live in head columns [0, A). rng: int64[1] state (advanced here).
// Returns (action (M,1) int64, logits (M,A), log_prob (M,1), h (M,H), c (M,H)).
// optional ``outs`` (5 preallocated tensors: action i64 (M,1),
// logits (M,A), logp (M,1), h_out (M,H), c_out (M,H)) skip the per-call
// allocations — the vectorized worker reuses ping-ponged buffers and
// cached numpy views. h_out/c_out MUST NOT alias hx/cx (the kernel reads
// state while writing the new one).
std::vector<at::Tensor> act_batch_discrete(
    const at::Tensor& obs, const at::Tensor& hx, const at::Tensor& cx,
    const at::Tensor& body_w, const at::Tensor& body_b,
    const at::Tensor& w_ih, const at::Tensor& w_hh, const at::Tensor& b_g,
    const at::Tensor& heads_w, const at::Tensor& heads_b, int64_t A,
    at::Tensor& rng,
    const c10::optional<std::vector<at::Tensor>>& outs = c10::nullopt) {
  TORCH_CHECK(obs.device().is_cpu() && obs.dtype() == at::kFloat);
  TORCH_CHECK(obs.is_contiguous() && hx.is_contiguous() && cx.is_contiguous());
  const int M = obs.size(0), F = obs.size(1), H = hx.size(1);
  const int G = 4 * H, D = heads_w.size(1);
  TORCH_CHECK(body_w.size(0) == F && body_w.size(1) == H,
              "single-body discrete core expected");
  TORCH_CHECK(A <= D);

  auto opt = obs.options();
  const bool reuse = outs.has_value();
  TORCH_CHECK(!reuse || outs->size() == 5, "outs must hold 5 tensors");
  auto action = reuse ? (*outs)[0] : at::empty({M, 1}, opt.dtype(at::kLong));
  auto logits = reuse ? (*outs)[1] : at::empty({M, (long)A}, opt);
  auto logp = reuse ? (*outs)[2] : at::empty({M, 1}, opt);
  auto h_out = reuse ? (*outs)[3] : at::empty({M, H}, opt);
  auto c_out = reuse ? (*outs)[4] : at::empty({M, H}, opt);
  TORCH_CHECK(!reuse || (h_out.data_ptr() != hx.data_ptr() &&
                         c_out.data_ptr() != cx.data_ptr()),
              "out state buffers must not alias the input state");
  // release the GIL for the compute: worker threads (tests / in-process
  // fleets) must not starve the python threads sharing the interpreter
  pybind11::gil_scoped_release nogil;

  const float* ob = obs.data_ptr<float>();
  const float* hp = hx.data_ptr<float>();
  const float* cp = cx.data_ptr<float>();
  const float* bw = body_w.data_ptr<float>();
  const float* bb = body_b.data_ptr<float>();
  const float* wih = w_ih.data_ptr<float>();
  const float* whh = w_hh.data_ptr<float>();
  const float* bg = b_g.data_ptr<float>();
  const float* hw = heads_w.data_ptr<float>();
  const float* hb = heads_b.data_ptr<float>();
  int64_t* act_p = action.data_ptr<int64_t>();
  float* lg_p = logits.data_ptr<float>();
  float* lp_p = logp.data_ptr<float>();
  float* ho_p = h_out.data_ptr<float>();
  float* co_p = c_out.data_ptr<float>();

  std::vector<float> xb(H), gates(G), hnew(H), prob(A);
  uint64_t s = (uint64_t)rng.data_ptr<int64_t>()[0];

  for (int m = 0; m < M; ++m) {
    // body GEMM + ReLU (stream rows of body_w: unit stride over j)
    for (int j = 0; j < H; ++j) xb[j] = bb[j];
    for (int k = 0; k < F; ++k) {
      const float x = ob[(long)m * F + k];
      const float* row = bw + (long)k * H;
      for (int j = 0; j < H; ++j) xb[j] += x * row[j];
    }
    for (int j = 0; j < H; ++j) xb[j] = xb[j] > 0.f ? xb[j] : 0.f;

    // gates = xb @ w_ih + h @ w_hh + b_g
    for (int g = 0; g < G; ++g) gates[g] = bg[g];
    for (int k = 0; k < H; ++k) {
      const float xk = xb[k];
      const float* row = wih + (long)k * G;
      for (int g = 0; g < G; ++g) gates[g] += xk * row[g];
    }
    for (int k = 0; k < H; ++k) {
      const float hk = hp[(long)m * H + k];
      const float* row = whh + (long)k * G;
      for (int g = 0; g < G; ++g) gates[g] += hk * row[g];
    }

    // cell update (gate order i, f, g, o — matches the eager core)
    for (int j = 0; j < H; ++j) {
      const float gi = sigf(gates[j]);
      const float gf = sigf(gates[H + j]);
      const float gg = std::tanh(gates[2 * H + j]);
      const float go = sigf(gates[3 * H + j]);
      const float c = gf * cp[(long)m * H + j] + gi * gg;
      const float h = go * std::tanh(c);
      co_p[(long)m * H + j] = c;
      ho_p[(long)m * H + j] = h;
      hnew[j] = h;
    }

    // logits head (columns [0, A) of the packed heads)
    float* lg = lg_p + (long)m * A;
    for (int a = 0; a < A; ++a) lg[a] = hb[a];
    for (int k = 0; k < H; ++k) {
      const float hk = hnew[k];
      const float* row = hw + (long)k * D;
      for (int a = 0; a < A; ++a) lg[a] += hk * row[a];
    }

    // categorical sample + log-prob
    float mx = lg[0];
    for (int a = 1; a < A; ++a) mx = std::max(mx, lg[a]);
    float Z = 0.f;
    for (int a = 0; a < A; ++a) {
      prob[a] = std::exp(lg[a] - mx);
      Z += prob[a];
    }
    const float u = (float)((xorshift64(s) >> 11) *
                            (1.0 / 9007199254740992.0));  // [0,1)
    float acc = 0.f;
    int chosen = A - 1;
    for (int a = 0; a < A; ++a) {
      acc += prob[a] / Z;
      if (u < acc) {
        chosen = a;
        break;
      }
    }
    act_p[m] = chosen;
    lp_p[m] = lg[chosen] - mx - std::log(Z);
  }
  rng.data_ptr<int64_t>()[0] = (int64_t)s;
  return {action, logits, logp, h_out, c_out};
}

// Continuous policies. mode 0 = PPO-C scheme: Normal(tanh(mu),
// softplus(std)+1e-4), unbounded sample (reference models.py:103-118);
// mode 1 = SAC-C scheme: tanh-squashed reparameterized Gaussian with
// log_std clamped to [-20, 2] (reference models.py:162-231). Heads:
// mu = cols [0,A), second head = cols [A,2A) (a value head may follow).
// Returns (action (M,A), logits (M,2A) = [mu|second_raw], logp (M,1), h, c).
std::vector<at::Tensor> act_batch_gaussian(
    const at::Tensor& obs, const at::Tensor& hx, const at::Tensor& cx,
    const at::Tensor& body_w, const at::Tensor& body_b,
    const at::Tensor& w_ih, const at::Tensor& w_hh, const at::Tensor& b_g,
    const at::Tensor& heads_w, const at::Tensor& heads_b, int64_t A,
    int64_t mode, at::Tensor& rng,
    const c10::optional<std::vector<at::Tensor>>& outs = c10::nullopt) {
  TORCH_CHECK(obs.device().is_cpu() && obs.dtype() == at::kFloat);
  TORCH_CHECK(obs.is_contiguous() && hx.is_contiguous() && cx.is_contiguous());
  const int M = obs.size(0), F = obs.size(1), H = hx.size(1);
  const int G = 4 * H, D = heads_w.size(1);
  TORCH_CHECK(2 * A <= D);
  constexpr float kHalfLog2Pi = 0.91893853320467274f;

  auto opt = obs.options();
  const bool reuse = outs.has_value();
  TORCH_CHECK(!reuse || outs->size() == 5, "outs must hold 5 tensors");
  auto action = reuse ? (*outs)[0] : at::empty({M, (long)A}, opt);
  auto logits = reuse ? (*outs)[1] : at::empty({M, (long)(2 * A)}, opt);
  auto logp = reuse ? (*outs)[2] : at::empty({M, 1}, opt);
  auto h_out = reuse ? (*outs)[3] : at::empty({M, H}, opt);
  auto c_out = reuse ? (*outs)[4] : at::empty({M, H}, opt);
  TORCH_CHECK(!reuse || (h_out.data_ptr() != hx.data_ptr() &&
                         c_out.data_ptr() != cx.data_ptr()),
              "out state buffers must not alias the input state");
  // release the GIL for the compute: worker threads (tests / in-process
  // fleets) must not starve the python threads sharing the interpreter
  pybind11::gil_scoped_release nogil;

  const float* ob = obs.data_ptr<float>();
  const float* hp = hx.data_ptr<float>();
  const float* cp = cx.data_ptr<float>();
  const float* bw = body_w.data_ptr<float>();
  const float* bb = body_b.data_ptr<float>();
  const float* wih = w_ih.data_ptr<float>();
  const float* whh = w_hh.data_ptr<float>();
  const float* bg = b_g.data_ptr<float>();
  const float* hw = heads_w.data_ptr<float>();
  const float* hb = heads_b.data_ptr<float>();
  float* act_p = action.data_ptr<float>();
  float* lg_p = logits.data_ptr<float>();
  float* lp_p = logp.data_ptr<float>();
  float* ho_p = h_out.data_ptr<float>();
  float* co_p = c_out.data_ptr<float>();

  std::vector<float> xb(H), gates(G), hnew(H), head(2 * A);
  uint64_t s = (uint64_t)rng.data_ptr<int64_t>()[0];
  auto uni = [&]() {
    return (float)(((xorshift64(s) >> 11) + 1) * (1.0 / 9007199254740993.0));
  };

  for (int m = 0; m < M; ++m) {
    for (int j = 0; j < H; ++j) xb[j] = bb[j];
    for (int k = 0; k < F; ++k) {
      const float x = ob[(long)m * F + k];
      const float* row = bw + (long)k * H;
      for (int j = 0; j < H; ++j) xb[j] += x * row[j];
    }
    for (int j = 0; j < H; ++j) xb[j] = xb[j] > 0.f ? xb[j] : 0.f;

    for (int g = 0; g < G; ++g) gates[g] = bg[g];
    for (int k = 0; k < H; ++k) {
      const float xk = xb[k];
      const float* row = wih + (long)k * G;
      for (int g = 0; g < G; ++g) gates[g] += xk * row[g];
    }
    for (int k = 0; k < H; ++k) {
      const float hk = hp[(long)m * H + k];
      const float* row = whh + (long)k * G;
      for (int g = 0; g < G; ++g) gates[g] += hk * row[g];
    }
    for (int j = 0; j < H; ++j) {
      const float gi = sigf(gates[j]);
      const float gf = sigf(gates[H + j]);
      const float gg = std::tanh(gates[2 * H + j]);
      const float go = sigf(gates[3 * H + j]);
      const float c = gf * cp[(long)m * H + j] + gi * gg;
      const float h = go * std::tanh(c);
      co_p[(long)m * H + j] = c;
      ho_p[(long)m * H + j] = h;
      hnew[j] = h;
    }

    for (int a = 0; a < 2 * A; ++a) head[a] = hb[a];
    for (int k = 0; k < H; ++k) {
      const float hk = hnew[k];
      const float* row = hw + (long)k * D;
      for (int a = 0; a < 2 * A; ++a) head[a] += hk * row[a];
    }
    for (int a = 0; a < 2 * A; ++a) lg_p[(long)m * 2 * A + a] = head[a];

    float lp = 0.f;
    for (int a = 0; a < A; ++a) {
      const float mu_raw = head[a];
      const float second = head[A + a];
      // standard normal via Box-Muller
      const float n =
          std::sqrt(-2.0f * std::log(uni())) *
          std::cos(6.283185307179586f * uni());
      float act_v;
      if (mode == 0) {  // PPO-C: Normal(tanh(mu), softplus(std)+1e-4)
        const float mean = std::tanh(mu_raw);
        const float sd =
            std::log1p(std::exp(second)) + 1e-4f;  // softplus
        act_v = mean + sd * n;
        lp += -0.5f * n * n - std::log(sd) - kHalfLog2Pi;
      } else {  // SAC-C: tanh-squashed, log_std clamped [-20, 2]
        const float ls = std::min(std::max(second, -20.0f), 2.0f);
        const float sd = std::exp(ls);
        const float z = mu_raw + sd * n;
        act_v = std::tanh(z);
        lp += -0.5f * n * n - ls - kHalfLog2Pi -
              std::log(1.0f - act_v * act_v + 1e-7f);
      }
      act_p[(long)m * A + a] = act_v;
    }
    lp_p[m] = lp;
  }
  rng.data_ptr<int64_t>()[0] = (int64_t)s;
  return {action, logits, logp, h_out, c_out};
}


// ------------------------------------------------------------------ //
// Batched native env physics (vectorized worker fast path): one call
// steps all M envs, writing obs/rew/done and updating state/steps
// in-place. Dynamics replicate pdrl_amd/envs/{cartpole,mountain_car}.py
// exactly (same float64 state, same formulas; parity-tested vs the
// python envs in tests/test_envs.py). Resets stay in Python (per-env
// RNG ownership).

// CartPole-v1 (envs/cartpole.py:54-84): state (M,4) f64, act (M) f32
// (0/1), steps (M) i64; done = terminated || steps >= max_steps.
std::vector<at::Tensor> cartpole_step_batch(at::Tensor& state,
                                            const at::Tensor& act,
                                            at::Tensor& steps, long max_steps,
                                            const c10::optional<std::vector<at::Tensor>>& outs = c10::nullopt) {
  const long M = state.size(0);
  const bool reuse = outs.has_value();
  TORCH_CHECK(!reuse || outs->size() == 3, "outs must hold 3 tensors");
  auto obs = reuse ? (*outs)[0] : at::empty({M, 4}, act.options());
  auto rew = reuse ? (*outs)[1] : at::empty({M}, act.options());
  auto done = reuse ? (*outs)[2] : at::empty({M}, act.options());
  pybind11::gil_scoped_release nogil;
  double* st = state.data_ptr<double>();
  const float* ac = act.data_ptr<float>();
  int64_t* sp = steps.data_ptr<int64_t>();
  float* ob = obs.data_ptr<float>();
  float* rw = rew.data_ptr<float>();
  float* dn = done.data_ptr<float>();
  constexpr double kGrav = 9.8, kMassPole = 0.1, kTotalMass = 1.1;
  constexpr double kLen = 0.5, kPoleMassLen = 0.05, kForceMag = 10.0;
  constexpr double kTau = 0.02, kXThr = 2.4;
  const double kThetaThr = 12.0 * 2.0 * M_PI / 360.0;
  for (long m = 0; m < M; ++m) {
    double x = st[m * 4], x_dot = st[m * 4 + 1];
    double th = st[m * 4 + 2], th_dot = st[m * 4 + 3];
    const double force = (ac[m] >= 0.5f) ? kForceMag : -kForceMag;
    const double cos_t = std::cos(th), sin_t = std::sin(th);
    const double temp =
        (force + kPoleMassLen * th_dot * th_dot * sin_t) / kTotalMass;
    const double th_acc =
        (kGrav * sin_t - cos_t * temp) /
        (kLen * (4.0 / 3.0 - kMassPole * cos_t * cos_t / kTotalMass));
    const double x_acc = temp - kPoleMassLen * th_acc * cos_t / kTotalMass;
    x += kTau * x_dot;
    x_dot += kTau * x_acc;
    th += kTau * th_dot;
    th_dot += kTau * th_acc;
    st[m * 4] = x;
    st[m * 4 + 1] = x_dot;
    st[m * 4 + 2] = th;
    st[m * 4 + 3] = th_dot;
    sp[m] += 1;
    const bool term = (x < -kXThr) || (x > kXThr) || (th < -kThetaThr) ||
                      (th > kThetaThr);
    ob[m * 4] = (float)x;
    ob[m * 4 + 1] = (float)x_dot;
    ob[m * 4 + 2] = (float)th;
    ob[m * 4 + 3] = (float)th_dot;
    rw[m] = 1.0f;
    dn[m] = (term || sp[m] >= max_steps) ? 1.0f : 0.0f;
  }
  return {obs, rew, done};
}

// MountainCarContinuous-v0 (envs/mountain_car.py:44-64): state (M,2)
// f64, act (M) f32 in [-1,1].
std::vector<at::Tensor> mcc_step_batch(at::Tensor& state,
                                       const at::Tensor& act,
                                       at::Tensor& steps, long max_steps,
                                       const c10::optional<std::vector<at::Tensor>>& outs = c10::nullopt) {
  const long M = state.size(0);
  const bool reuse = outs.has_value();
  TORCH_CHECK(!reuse || outs->size() == 3, "outs must hold 3 tensors");
  auto obs = reuse ? (*outs)[0] : at::empty({M, 2}, act.options());
  auto rew = reuse ? (*outs)[1] : at::empty({M}, act.options());
  auto done = reuse ? (*outs)[2] : at::empty({M}, act.options());
  pybind11::gil_scoped_release nogil;
  double* st = state.data_ptr<double>();
  const float* ac = act.data_ptr<float>();
  int64_t* sp = steps.data_ptr<int64_t>();
  float* ob = obs.data_ptr<float>();
  float* rw = rew.data_ptr<float>();
  float* dn = done.data_ptr<float>();
  constexpr double kMinPos = -1.2, kMaxPos = 0.6, kMaxSpeed = 0.07;
  constexpr double kGoalPos = 0.45, kGoalVel = 0.0, kPower = 0.0015;
  for (long m = 0; m < M; ++m) {
    double pos = st[m * 2], vel = st[m * 2 + 1];
    const double force =
        std::min(std::max((double)ac[m], -1.0), 1.0);
    vel += force * kPower - 0.0025 * std::cos(3.0 * pos);
    vel = std::min(std::max(vel, -kMaxSpeed), kMaxSpeed);
    pos += vel;
    pos = std::min(std::max(pos, kMinPos), kMaxPos);
    if (pos <= kMinPos && vel < 0.0) vel = 0.0;
    st[m * 2] = pos;
    st[m * 2 + 1] = vel;
    sp[m] += 1;
    const bool term = (pos >= kGoalPos) && (vel >= kGoalVel);
    ob[m * 2] = (float)pos;
    ob[m * 2 + 1] = (float)vel;
    double r = -0.1 * force * force;
    if (term) r += 100.0;
    rw[m] = (float)r;
    dn[m] = (term || sp[m] >= max_steps) ? 1.0f : 0.0f;
  }
  return {obs, rew, done};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("act_batch_discrete", &act_batch_discrete,
        "batched CPU actor step: body+LSTM+logits+sample in one call",
        pybind11::arg("obs"), pybind11::arg("hx"), pybind11::arg("cx"),
        pybind11::arg("body_w"), pybind11::arg("body_b"),
        pybind11::arg("w_ih"), pybind11::arg("w_hh"), pybind11::arg("b_g"),
        pybind11::arg("heads_w"), pybind11::arg("heads_b"),
        pybind11::arg("A"), pybind11::arg("rng"),
        pybind11::arg("outs") = pybind11::none());
  m.def("act_batch_gaussian", &act_batch_gaussian,
        "batched CPU actor step for Gaussian policies (PPO-C / SAC-C)",
        pybind11::arg("obs"), pybind11::arg("hx"), pybind11::arg("cx"),
        pybind11::arg("body_w"), pybind11::arg("body_b"),
        pybind11::arg("w_ih"), pybind11::arg("w_hh"), pybind11::arg("b_g"),
        pybind11::arg("heads_w"), pybind11::arg("heads_b"),
        pybind11::arg("A"), pybind11::arg("mode"), pybind11::arg("rng"),
        pybind11::arg("outs") = pybind11::none());
  m.def("cartpole_step_batch", &cartpole_step_batch,
        "vectorized CartPole-v1 physics (native env fast path)",
        pybind11::arg("state"), pybind11::arg("act"), pybind11::arg("steps"),
        pybind11::arg("max_steps"), pybind11::arg("outs") = pybind11::none());
  m.def("mcc_step_batch", &mcc_step_batch,
        "vectorized MountainCarContinuous-v0 physics",
        pybind11::arg("state"), pybind11::arg("act"), pybind11::arg("steps"),
        pybind11::arg("max_steps"), pybind11::arg("outs") = pybind11::none());
}
