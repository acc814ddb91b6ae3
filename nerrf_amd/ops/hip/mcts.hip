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
#define MAX_ACTIONS (3 + MAX_GROUPS)
#define MAX_DEPTH 16
#define A_STOP 0
#define A_KILL 1
#define A_RESTORE 2
#define A_REVERT_BASE 3

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
  float fp_mb = 0.0f, downtime = 0.0f, ongoing = 0.0f, staleness = 0.0f;
  bool alive = true, restored = false;
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
    if (a == A_RESTORE) {
      if (restored) continue;
      const float dt = p.restore_time_s;
      if (alive) ongoing += p.attack_rate_mbps * dt * proc_score;
      downtime += dt;
      for (int g = 0; g < p.n_groups; ++g) reverted[g] = 1.0f;
      staleness = p.restore_loss_mb;
      restored = true;
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
    // degradation charge scales with belief the process is malicious
    downtime += p.horizon_s * proc_score;
  }
  float loss = 0.0f;
  for (int g = 0; g < p.n_groups; ++g)
    loss += (1.0f - reverted[g]) * gscore[g] * gmb[g];
  loss += fminf(ongoing, remaining_clean_mb);
  loss += staleness;
  return -(loss + p.downtime_weight * downtime + p.fp_weight * fp_mb);
}

// Wave-vectorized eval_plan: group g's stats live in lane g's registers
// (loaded once per tree instead of per simulation); per-action group
// lookups are shuffles; the final loss reduces by a SEQUENTIAL shuffle
// chain so the fp32 accumulation order matches eval_plan / the CPU
// reference bit-for-bit.  All 64 lanes execute the (wave-uniform) control
// flow; only the group-indexed state is lane-local.
__device__ float eval_plan_wave(float my_score, float my_mb, float my_files,
                                float proc_score, float remaining_clean_mb,
                                const int* actions, int n_act,
                                const PlannerParamsDev& p, int lane) {
  float my_rev = 0.0f;
  float fp_mb = 0.0f, downtime = 0.0f, ongoing = 0.0f, staleness = 0.0f;
  bool alive = true, restored = false;
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
    if (a == A_RESTORE) {
      if (restored) continue;
      const float dt = p.restore_time_s;
      if (alive) ongoing += p.attack_rate_mbps * dt * proc_score;
      downtime += dt;
      my_rev = 1.0f;
      staleness = p.restore_loss_mb;
      restored = true;
      continue;
    }
    const int gi = a - A_REVERT_BASE;
    if (gi < 0 || gi >= p.n_groups) continue;
    if (__shfl(my_rev, gi, NERRF_WAVE) > 0.0f) continue;
    const float files_g = __shfl(my_files, gi, NERRF_WAVE);
    const float score_g = __shfl(my_score, gi, NERRF_WAVE);
    const float mb_g = __shfl(my_mb, gi, NERRF_WAVE);
    const float dt = p.revert_time_s * fmaxf(files_g, 1.0f);
    if (alive) ongoing += p.attack_rate_mbps * dt * proc_score;
    downtime += dt;
    if (lane == gi) my_rev = 1.0f;
    fp_mb += (1.0f - score_g) * mb_g * 0.05f;
  }
  if (alive) {
    ongoing += p.attack_rate_mbps * p.horizon_s * proc_score;
    downtime += p.horizon_s * proc_score;
  }
  const float my_term = (1.0f - my_rev) * my_score * my_mb;
  float loss = 0.0f;
  for (int g = 0; g < p.n_groups; ++g)  // sequential order == scalar order
    loss += __shfl(my_term, g, NERRF_WAVE);
  loss += fminf(ongoing, remaining_clean_mb);
  loss += staleness;
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

  // All 64 lanes run the (wave-uniform) tree walk redundantly; arena WRITES
  // are lane-0-guarded, and the per-simulation reward evaluates wave-wide
  // with group state lane-resident (eval_plan_wave) — the v1 kernel parked
  // 63/64 lanes for the whole search.
  int* parent = arena_i + (long)wave * (3 * cap + cap * p.n_actions);
  int* action = parent + cap;
  int* visits = action + cap;
  int* children = visits + cap;
  float* value = arena_f + (long)wave * cap;

  const float my_score = (lane < p.n_groups) ? gscore[lane] : 0.0f;
  const float my_mb = (lane < p.n_groups) ? gmb[lane] : 0.0f;
  const float my_files = (lane < p.n_groups) ? gfiles[lane] : 0.0f;

  // init arena (lane-parallel)
  for (int i = lane; i < cap; i += NERRF_WAVE) {
    parent[i] = -1;
    action[i] = -1;
    visits[i] = 0;
    value[i] = 0.0f;
  }
  for (int i = lane; i < cap * p.n_actions; i += NERRF_WAVE) children[i] = -1;
  int n_nodes = 1;

  unsigned rng = (p.seed * 2654435761u + (unsigned)wave * 40503u + 1u);
  int acts[MAX_DEPTH + 1];

  for (int si = 0; si < p.sims_per_tree; ++si) {
    // ---- selection + expansion (redundant on every lane) ----
    int node = 0, depth = 0;
    while (depth < p.max_depth) {
      int* kids = children + node * p.n_actions;
      int untried = -1;
      for (int a = 0; a < p.n_actions; ++a)
        if (kids[a] < 0) { untried = a; break; }
      if (untried >= 0) {
        if (n_nodes >= cap) break;  // arena full: treat as leaf
        const int nw = n_nodes++;
        if (lane == 0) {
          parent[nw] = node;
          action[nw] = untried;
          kids[untried] = nw;
        }
        // lanes track the walk in registers; memory catches up via lane 0
        node = nw;
        ++depth;
        acts[depth - 1] = untried;  // kept in sync below
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
      acts[depth] = best;
      ++depth;
      if (best == A_STOP) break;
    }
    const int path_len = depth;  // acts[0:depth] = root->node actions
    // ---- rollout ----
    int n_act = path_len;
    rng = xorshift32_dev(rng ^ ((unsigned)si * 747796405u + 2891336453u));
    unsigned r = rng;
    while (n_act < p.max_depth && (n_act == 0 || acts[n_act - 1] != A_STOP)) {
      r = xorshift32_dev(r);
      const int a = (int)(r % (unsigned)p.n_actions);
      acts[n_act++] = a;
      if (a == A_STOP) break;
    }
    const float reward = eval_plan_wave(my_score, my_mb, my_files, proc_score,
                                        remaining_clean_mb, acts, n_act, p,
                                        lane);
    // ---- backup (lane 0 writes; all lanes know the path in registers) ----
    if (lane == 0) {
      for (int nd = node; nd >= 0; nd = parent[nd]) {
        visits[nd] += 1;
        value[nd] += reward;
      }
    }
  }

  // export root child statistics
  if (lane == 0) {
    int* kids = children;  // node 0
    for (int a = 0; a < p.n_actions; ++a) {
      const int ch = kids[a];
      root_visits[(long)wave * p.n_actions + a] = (ch >= 0) ? visits[ch] : 0;
      root_value[(long)wave * p.n_actions + a] = (ch >= 0) ? value[ch] : 0.0f;
    }
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
