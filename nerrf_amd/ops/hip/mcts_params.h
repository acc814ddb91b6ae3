// Shared host/device planner parameter block (bindings.cpp + mcts.hip).
#pragma once

namespace nerrf {

struct PlannerParamsDev {
  int n_groups;
  int n_actions;
  int max_depth;
  int sims_per_tree;
  float downtime_weight;
  float revert_time_s;
  float kill_time_s;
  float fp_weight;
  float attack_rate_mbps;
  float horizon_s;
  float restore_time_s;
  float restore_loss_mb;
  float ucb_c;
  unsigned seed;
};

}  // namespace nerrf
