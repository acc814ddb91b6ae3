// Batched root-parallel MCTS for the rollback planner — CDNA4 (gfx950).
//
// One wave64 per tree; 1024 trees run concurrently (BASELINE.json config 4:
// "1024 parallel sims").  Tree arenas live in HBM as flat arrays; each wave
// runs select / expand / rollout / backup sequentially over its own arena
// (lane 0 drives the walk — tree ops are inherently divergent — while the
// reward evaluation vectorises over file groups across lanes).
//
// The MDP, reward (-(data_loss + 0.1*downtime)) and xorshift32 rollout RNG
// mirror nerrf_amd/planner/{rewards,mcts}.py exactly; tests compare plans and
// root statistics against that CPU reference.
#include "common.h"
#include "mcts_params.h"

namespace nerrf {

#define MAX_GROUPS 16
#define MAX_ACTIONS (2 + MAX_GROUPS)
#define MAX_DEPTH 16
#define A_STOP 0
#define A_KILL 1
#define A_REVERT_BASE 2

__device__ __forceinline__ unsigned xorshift32_dev(unsigned s) {
  s ^= s << 13;
  s ^= s >> 17;
  s ^= s << 5;
  return s;
}

// Mirrors rewards.simulate_plan (fp32).
__device__ float eval_plan(const float* gscore, const float* gmb,
                           const float* gfiles, float proc_score,
                           float remaining_clean_mb,
                           const int* actions, int n_act,
                           const PlannerParamsDev& p) {
  float reverted[MAX_GROUPS];
  for (int g = 0; g < p.n_groups; ++g) reverted[g] = 0.0f;
  float fp_mb = 0.0f, downtime = 0.0f, ongoing = 0.0f;
  bool alive = true;
  for (int i = 0; i < n_act; ++i) {
    const int a = actions[i];
    if (a == A_STOP) break;
    if (a == A_KILL) {
      if (alive) {
        downtime += p.kill_time_s;
        alive = false;
      }
      continue;
    }
    const int gi = a - A_REVERT_BASE;
    if (gi < 0 || gi >= p.n_groups || reverted[gi] > 0.0f) continue;
    const float dt = p.revert_time_s * fmaxf(gfiles[gi], 1.0f);
    if (alive) ongoing += p.attack_rate_mbps * dt * proc_score;
    downtime += dt;
    reverted[gi] = 1.0f;
    fp_mb += (1.0f - gscore[gi]) * gmb[gi] * 0.05f;
  }
  if (alive) {
    ongoing += p.attack_rate_mbps * p.horizon_s * proc_score;
    downtime += p.horizon_s;
  }
  float loss = 0.0f;
  for (int g = 0; g < p.n_groups; ++g)
    loss += (1.0f - reverted[g]) * gscore[g] * gmb[g];
  loss += fminf(ongoing, remaining_clean_mb);
  return -(loss + p.downtime_weight * downtime + p.fp_weight * fp_mb);
}

// Arena layout per tree (cap nodes):
//   parent[cap] i32 | action[cap] i32 | visits[cap] i32 | value[cap] f32 |
//   children[cap * n_actions] i32
__global__ void mcts_kernel(
    const float* __restrict__ gscore,   // [G]
    const float* __restrict__ gmb,      // [G]
    const float* __restrict__ gfiles,   // [G]
    float proc_score, float remaining_clean_mb,
    PlannerParamsDev p,
    int n_trees, int cap,
    int* __restrict__ arena_i,          // [n_trees, 3*cap + cap*n_actions]
    float* __restrict__ arena_f,        // [n_trees, cap]
    int* __restrict__ root_visits,      // [n_trees, n_actions]
    float* __restrict__ root_value) {   // [n_trees, n_actions]
  const int wave = (blockIdx.x * blockDim.x + threadIdx.x) / NERRF_WAVE;
  const int lane = threadIdx.x % NERRF_WAVE;
  if (wave >= n_trees) return;
  if (lane != 0) return;  // lane 0 drives; tree walk is scalar by nature

  int* parent = arena_i + (long)wave * (3 * cap + cap * p.n_actions);
  int* action = parent + cap;
  int* visits = action + cap;
  int* children = visits + cap;
  float* value = arena_f + (long)wave * cap;

  // init arena
  for (int i = 0; i < cap; ++i) {
    parent[i] = -1;
    action[i] = -1;
    visits[i] = 0;
    value[i] = 0.0f;
  }
  for (int i = 0; i < cap * p.n_actions; ++i) children[i] = -1;
  int n_nodes = 1;

  unsigned rng = (p.seed * 2654435761u + (unsigned)wave * 40503u + 1u);
  int acts[MAX_DEPTH + 1];

  for (int si = 0; si < p.sims_per_tree; ++si) {
    // ---- selection + expansion ----
    int node = 0, depth = 0;
    while (depth < p.max_depth) {
      int* kids = children + node * p.n_actions;
      int untried = -1;
      for (int a = 0; a < p.n_actions; ++a)
        if (kids[a] < 0) { untried = a; break; }
      if (untried >= 0) {
        if (n_nodes >= cap) break;  // arena full: treat as leaf
        const int nw = n_nodes++;
        parent[nw] = node;
        action[nw] = untried;
        kids[untried] = nw;
        node = nw;
        ++depth;
        break;
      }
      // UCB1 over fully-expanded children
      const float logn = __logf(fmaxf((float)visits[node], 1.0f));
      int best = 0;
      float best_u = -1e30f;
      bool found_unvisited = false;
      for (int a = 0; a < p.n_actions && !found_unvisited; ++a) {
        const int ch = kids[a];
        const int nv = visits[ch];
        float u;
        if (nv == 0) {
          best = a;
          found_unvisited = true;
          break;
        }
        u = value[ch] / (float)nv + p.ucb_c * __fsqrt_rn(logn / (float)nv);
        if (u > best_u) { best_u = u; best = a; }
      }
      node = kids[best];
      ++depth;
      if (action[node] == A_STOP) break;
    }
    // ---- rollout ----
    int n_act = 0;
    {  // reconstruct path actions root->node
      int chain[MAX_DEPTH];
      int cl = 0;
      for (int nd = node; nd != 0 && cl < MAX_DEPTH; nd = parent[nd]) chain[cl++] = action[nd];
      for (int i = cl - 1; i >= 0; --i) acts[n_act++] = chain[i];
    }
    rng = xorshift32_dev(rng ^ ((unsigned)si * 747796405u + 2891336453u));
    unsigned r = rng;
    while (n_act < p.max_depth && (n_act == 0 || acts[n_act - 1] != A_STOP)) {
      r = xorshift32_dev(r);
      const int a = (int)(r % (unsigned)p.n_actions);
      acts[n_act++] = a;
      if (a == A_STOP) break;
    }
    const float reward = eval_plan(gscore, gmb, gfiles, proc_score,
                                   remaining_clean_mb, acts, n_act, p);
    // ---- backup ----
    for (int nd = node; nd >= 0; nd = parent[nd]) {
      visits[nd] += 1;
      value[nd] += reward;
    }
  }

  // export root child statistics
  int* kids = children;  // node 0
  for (int a = 0; a < p.n_actions; ++a) {
    const int ch = kids[a];
    root_visits[(long)wave * p.n_actions + a] = (ch >= 0) ? visits[ch] : 0;
    root_value[(long)wave * p.n_actions + a] = (ch >= 0) ? value[ch] : 0.0f;
  }
}

// Batch plan evaluator (test harness for eval_plan parity with the CPU ref).
__global__ void eval_plans_kernel(
    const float* __restrict__ gscore, const float* __restrict__ gmb,
    const float* __restrict__ gfiles, float proc_score,
    float remaining_clean_mb, PlannerParamsDev p,
    const int* __restrict__ plans,  // [n_plans, plan_len]
    int n_plans, int plan_len, float* __restrict__ out) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n_plans) return;
  int acts[MAX_DEPTH + 1];
  const int n = plan_len < MAX_DEPTH ? plan_len : MAX_DEPTH;
  for (int j = 0; j < n; ++j) acts[j] = plans[(long)i * plan_len + j];
  out[i] = eval_plan(gscore, gmb, gfiles, proc_score, remaining_clean_mb, acts, n, p);
}

void launch_mcts(const float* gscore, const float* gmb, const float* gfiles,
                 float proc_score, float remaining_clean_mb,
                 const PlannerParamsDev& p, int n_trees, int cap,
                 int* arena_i, float* arena_f, int* root_visits,
                 float* root_value, hipStream_t s) {
  const int waves_per_block = 4;
  const int block = waves_per_block * NERRF_WAVE;
  const int grid = (n_trees + waves_per_block - 1) / waves_per_block;
  mcts_kernel<<<grid, block, 0, s>>>(gscore, gmb, gfiles, proc_score,
                                     remaining_clean_mb, p, n_trees, cap,
                                     arena_i, arena_f, root_visits, root_value);
}

void launch_eval_plans(const float* gscore, const float* gmb,
                       const float* gfiles, float proc_score,
                       float remaining_clean_mb, const PlannerParamsDev& p,
                       const int* plans, int n_plans, int plan_len, float* out,
                       hipStream_t s) {
  const int block = 256;
  const int grid = (n_plans + block - 1) / block;
  eval_plans_kernel<<<grid, block, 0, s>>>(gscore, gmb, gfiles, proc_score,
                                           remaining_clean_mb, p, plans,
                                           n_plans, plan_len, out);
}

}  // namespace nerrf
